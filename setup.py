"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting deepspeed_amd/ops/_C*.so is git-ignored but ships to the GPU
box with the repo snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("deepspeed_amd", "ops", "csrc")

ext = CUDAExtension(
    name="deepspeed_amd.ops._C",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "cpu_adam.cpp"),
        os.path.join(CSRC, "aio.cpp"),
        os.path.join(CSRC, "shm_comm.cpp"),
        os.path.join(CSRC, "adam.hip"),
        os.path.join(CSRC, "optim.hip"),
        os.path.join(CSRC, "fp_quant.hip"),
        os.path.join(CSRC, "transpose.hip"),
        os.path.join(CSRC, "softmax_dropout.hip"),
        os.path.join(CSRC, "paged_decode.hip"),
        os.path.join(CSRC, "quantize.hip"),
        os.path.join(CSRC, "attention.hip"),
        os.path.join(CSRC, "attention_bwd.hip"),
        os.path.join(CSRC, "cross_entropy.hip"),
        os.path.join(CSRC, "norms.hip"),
        os.path.join(CSRC, "rope.hip"),
        os.path.join(CSRC, "swiglu.hip"),
        os.path.join(CSRC, "spatial.hip"),
        os.path.join(CSRC, "token_ops.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17", "-fopenmp", "-mavx2", "-mfma"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
    extra_link_args=["-fopenmp"],
)

setup(
    name="deepspeed_amd",
    version="0.1.0",
    packages=["deepspeed_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
