"""In-tree build of the gfx950 HIP extension.

Drives torch.utils.cpp_extension's hipcc path (``.hip`` sources compile with
hipcc directly — no hipify) and drops ``_tdpa_hip.*.so`` next to this file so
the built artifact travels with the repo snapshot to the GPU box.

Usage: ``python -m torchdistpackage_amd.ops.build`` or
``__graft_entry__.build()``.
"""

from __future__ import annotations

import os


HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")

SOURCES = [
    "bindings.cpp",
    "norms.hip",
    "elementwise.hip",
    "attention.hip",
    "attention_v2.hip",
    "cross_entropy.hip",
    "probe.hip",
    "gemm.hip",
    "rope_swiglu.hip",
    "gemv.hip",
]


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", str(min(os.cpu_count() or 4, 16)))
    from torch.utils.cpp_extension import load

    build_dir = os.path.join(HERE, "_build")
    os.makedirs(build_dir, exist_ok=True)
    mod = load(
        name="_tdpa_hip",
        sources=[os.path.join(CSRC, s) for s in SOURCES],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        build_directory=build_dir,
        verbose=verbose,
        is_python_module=False,  # just build; we load it ourselves
    )
    # copy the built .so next to this file (in-tree, snapshot-visible)
    import glob
    import shutil
    sos = glob.glob(os.path.join(build_dir, "_tdpa_hip*.so"))
    assert sos, f"no .so produced in {build_dir}"
    dest = os.path.join(HERE, os.path.basename(sos[0]))
    shutil.copy2(sos[0], dest)
    if verbose:
        print(f"[ops.build] built {dest}")
    return dest


if __name__ == "__main__":
    build()
