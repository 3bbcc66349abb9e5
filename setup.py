"""In-tree build of the dlrover_amd HIP extension for gfx950 (MI355X).

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built _hip_ops.so lands inside dlrover_amd/ops/ so it ships with the
repo snapshot to GPU boxes (gpurun) without any JIT cache.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "dlrover_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "rmsnorm.hip"),
    os.path.join(CSRC, "rope.hip"),
    os.path.join(CSRC, "swiglu.hip"),
    os.path.join(CSRC, "adamw.hip"),
    os.path.join(CSRC, "softmax.hip"),
    os.path.join(CSRC, "cross_entropy.hip"),
    os.path.join(CSRC, "mfma_probe.hip"),
    os.path.join(CSRC, "attention.hip"),
    os.path.join(CSRC, "attention_v3.hip"),
    os.path.join(CSRC, "attention_bwd.hip"),
    os.path.join(CSRC, "attention_bwd_v3.hip"),
]

setup(
    name="dlrover_amd_ops",
    ext_modules=[
        CUDAExtension(
            name="dlrover_amd.ops._hip_ops",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
