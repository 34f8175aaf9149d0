"""Build the gfx950 HIP kernel pack in-tree: paddlenlp_amd/ops/_C*.so.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(__graft_entry__.build() sets the env var and calls this.)
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "paddlenlp_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in sorted(os.listdir(CSRC))
    if f.endswith((".hip", ".cpp")) and not f.endswith("_hip.hip") and not f.endswith("_hip.cpp")
]

setup(
    name="paddlenlp_amd_ops",
    ext_modules=[
        CUDAExtension(
            name="paddlenlp_amd.ops._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                # NOTE: no -ffast-math — the flash-attention kernels rely on
                # -INFINITY masking semantics; -fno-math-errno is safe.
                "nvcc": [
                    "-O3",
                    "-std=c++17",
                    "--offload-arch=gfx950",
                    "-fno-math-errno",
                ],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
