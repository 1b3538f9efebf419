"""Build the saturn_amd HIP extension in-tree for MI355X (gfx950).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built `saturn_amd/_C*.so` is git-ignored but ships with the gpurun
snapshot (the GPU box gets the binary without a JIT cache).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("saturn_amd", "ops", "csrc")

ext = CUDAExtension(
    name="saturn_amd._C",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "fused_optim.hip"),
        os.path.join(CSRC, "layernorm.hip"),
        os.path.join(CSRC, "cross_entropy.hip"),
        os.path.join(CSRC, "rope.hip"),
        os.path.join(CSRC, "gelu.hip"),
        os.path.join(CSRC, "attention.hip"),
        os.path.join(CSRC, "add3.hip"),
        os.path.join(CSRC, "swiglu.hip"),
        os.path.join(CSRC, "embedding.hip"),
        os.path.join(CSRC, "dropout.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

comm_ext = CUDAExtension(
    name="saturn_amd._comm",
    sources=[os.path.join("saturn_amd", "comm", "csrc", "rccl_comm.cpp")],
    libraries=["rccl"],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="saturn_amd",
    version="0.1.0",
    packages=[
        "saturn_amd",
        "saturn_amd.core",
        "saturn_amd.library",
        "saturn_amd.solver",
        "saturn_amd.engine",
        "saturn_amd.trial_runner",
        "saturn_amd.executors",
        "saturn_amd.parallel",
        "saturn_amd.ops",
        "saturn_amd.comm",
        "saturn_amd.models",
        "saturn_amd.utils",
    ],
    ext_modules=[ext, comm_ext],
    cmdclass={"build_ext": BuildExtension},
)
