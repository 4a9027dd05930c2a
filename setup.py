"""Build the MI355X HIP extension IN-TREE:
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The .so lands in learning_at_home_amd/ops/ so it travels with the repo
snapshot to GPU boxes (no JIT cache dependence)."""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

HIP_DIR = os.path.join("learning_at_home_amd", "ops", "hip")

ext = CUDAExtension(
    name="learning_at_home_amd.ops._lah_hip",
    sources=[
        os.path.join(HIP_DIR, "bindings.cpp"),
        os.path.join(HIP_DIR, "ffn_gemm.hip"),
        os.path.join(HIP_DIR, "ffn_gemm_tn.hip"),
        os.path.join(HIP_DIR, "gating.hip"),
        os.path.join(HIP_DIR, "dispatch.hip"),
        os.path.join(HIP_DIR, "optim.hip"),
        os.path.join(HIP_DIR, "attention.hip"),
        os.path.join(HIP_DIR, "ce.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="learning_at_home_amd",
    version="0.1.0",
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
