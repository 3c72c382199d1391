"""Installable packaging (reference parity: /root/reference/setup.py:4-10).

The HIP extension is built in-tree (python -m torchdistpackage_amd.ops.build)
rather than at install time, so the built .so stays next to its sources.
"""
from setuptools import find_packages, setup

setup(
    name="torchdistpackage_amd",
    version="0.1.0",
    description="MI355X-native mixed-parallel training toolkit "
                "(RCCL over xGMI + gfx950 HIP kernels)",
    packages=find_packages(include=["torchdistpackage_amd",
                                    "torchdistpackage_amd.*"]),
    package_data={"torchdistpackage_amd.ops": ["csrc/*", "*.so"]},
    python_requires=">=3.10",
)
