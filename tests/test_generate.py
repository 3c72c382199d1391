"""KV-cache generation vs the full-recompute oracle (fp32 CPU, tp=1)."""
import pytest
import torch

from torchdistpackage_amd.inference import generate
from torchdistpackage_amd.models.gpt2 import GPT2Config, GPT2Model
from torchdistpackage_amd.models.llama import LlamaModel, llama_tiny


def naive_generate(model, idx, n):
    """O(T^2) oracle: full forward every step, greedy."""
    tokens = idx
    for _ in range(n):
        logits = model(tokens)["logits"]
        nxt = logits[:, -1].argmax(-1)
        tokens = torch.cat([tokens, nxt[:, None]], dim=1)
    return tokens


@pytest.fixture
def gpt2_tiny_model():
    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=512, n_layer=2, n_head=4, dim=128,
                     max_seq=64)
    return GPT2Model(cfg).eval()


def test_gpt2_generate_matches_full_recompute(gpt2_tiny_model):
    m = gpt2_tiny_model
    torch.manual_seed(1)
    idx = torch.randint(0, 512, (2, 7))
    want = naive_generate(m, idx, 10)
    got = generate(m, idx, 10)
    assert torch.equal(got[:, :7], idx)
    assert torch.equal(got, want)


def test_llama_generate_matches_full_recompute():
    torch.manual_seed(0)
    m = LlamaModel(llama_tiny()).eval()
    torch.manual_seed(2)
    idx = torch.randint(0, 512, (2, 5))
    want = naive_generate(m, idx, 8)
    got = generate(m, idx, 8)
    assert torch.equal(got, want)


def test_generate_sampling_shapes_and_range(gpt2_tiny_model):
    m = gpt2_tiny_model
    torch.manual_seed(3)
    idx = torch.randint(0, 512, (3, 4))
    out = generate(m, idx, 6, greedy=False, temperature=0.8, top_k=20)
    assert out.shape == (3, 10)
    assert out.min() >= 0 and out.max() < 512


def test_generate_single_token_prompt(gpt2_tiny_model):
    m = gpt2_tiny_model
    idx = torch.tensor([[5]])
    want = naive_generate(m, idx, 4)
    assert torch.equal(generate(m, idx, 4), want)


def test_speculative_equals_greedy(gpt2_tiny_model):
    """Speculative decode must reproduce the target's greedy output
    EXACTLY, for any draft."""
    from torchdistpackage_amd.inference.generate import speculative_generate
    target = gpt2_tiny_model
    torch.manual_seed(42)
    draft = GPT2Model(GPT2Config(vocab_size=512, n_layer=1, n_head=2,
                                 dim=64, max_seq=64)).eval()
    torch.manual_seed(5)
    idx = torch.randint(0, 512, (1, 6))
    want = generate(target, idx, 12)
    for k in (1, 3, 4, 7):
        got = speculative_generate(target, draft, idx, 12, k=k)
        assert torch.equal(got, want), k
    # draft == target: every proposal accepted, still exact
    got = speculative_generate(target, target, idx, 12, k=4)
    assert torch.equal(got, want)


def test_speculative_llama():
    from torchdistpackage_amd.inference.generate import speculative_generate
    torch.manual_seed(0)
    target = LlamaModel(llama_tiny()).eval()
    torch.manual_seed(9)
    draft = LlamaModel(llama_tiny()).eval()
    idx = torch.randint(0, 512, (1, 5))
    want = generate(target, idx, 10)
    got = speculative_generate(target, draft, idx, 10, k=3)
    assert torch.equal(got, want)
