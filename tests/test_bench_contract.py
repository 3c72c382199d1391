"""Driver-contract test: bench.py must run under torch.distributed.run and
print ONE valid JSON line with the agreed keys, for the parallel layouts the
driver launches (dp2 at N=2; dp2*pp2*tp2 at N=8)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _run_bench(nproc, port, extra):
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
           "--master-port", str(port), os.path.join(REPO, "bench.py"),
           "--gpus", str(nproc), "--steps", "2", "--warmup", "1"] + extra
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, f"expected ONE json line, got: {out.stdout[-500:]}"
    payload = json.loads(lines[0])
    assert REQUIRED <= set(payload), REQUIRED - set(payload)
    assert payload["n_gpus"] == nproc
    assert payload["value"] > 0 and payload["ms_per_step"] > 0
    return payload


def test_bench_json_contract_dp2():
    p = _run_bench(2, 29601, ["--model", "tiny", "--batch", "2"])
    assert p["config"]["parallelism"] == "dp2_pp1_tp1"


@pytest.mark.slow
def test_bench_json_contract_8rank():
    p = _run_bench(8, 29602, ["--model", "tiny", "--batch", "2"])
    assert p["config"]["parallelism"] == "dp2_pp2_tp2"


@pytest.mark.slow
def test_bench_json_contract_zero_tp():
    """ZeRO + TP composition through the real bench flow (the r01-advisor
    bug class: SP grads must survive ZeRO's bucketing)."""
    os.environ["TDPA_PARALLEL"] = "2,1,2"
    try:
        p = _run_bench(4, 29603, ["--model", "tiny", "--batch", "4",
                                  "--zero"])
    finally:
        os.environ.pop("TDPA_PARALLEL", None)
    assert p["config"]["parallelism"] == "dp2_pp1_tp2"
