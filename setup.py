"""In-tree build of the gfx950 HIP extension (distrifuser_amd._C).

Build with:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(the arch defaults to gfx950 here if unset). The .so lands inside the package
so it travels to GPU boxes with the repo snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "distrifuser_amd", "csrc")

ext = CUDAExtension(
    name="distrifuser_amd._C",
    sources=[
        os.path.join(CSRC, "bindings.hip"),
        os.path.join(CSRC, "groupnorm.hip"),
        os.path.join(CSRC, "geglu.hip"),
        os.path.join(CSRC, "scheduler.hip"),
        os.path.join(CSRC, "attention.hip"),
        os.path.join(CSRC, "conv.hip"),
        os.path.join(CSRC, "vae_attn.hip"),
        os.path.join(CSRC, "layernorm.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++20"],
        "nvcc": ["-O3", "-std=c++20"],  # hipcc flags on ROCm
    },
)

setup(
    name="distrifuser_amd",
    version="0.1.0",
    description="MI355X-native displaced-patch-parallel diffusion inference",
    packages=["distrifuser_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
    python_requires=">=3.10",
)
