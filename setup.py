"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting alphafold2_amd/_hip_ops*.so travels with the repo snapshot
to GPU boxes (it is git-ignored but not gpurun-ignored).
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_DIR = os.path.join("alphafold2_amd", "ops", "hip")

ext = CUDAExtension(
    name="alphafold2_amd._hip_ops",
    sources=[
        os.path.join(HIP_DIR, "bindings.cpp"),
        os.path.join(HIP_DIR, "layernorm.hip"),
        os.path.join(HIP_DIR, "geglu.hip"),
        os.path.join(HIP_DIR, "distbucket.hip"),
        os.path.join(HIP_DIR, "attention.hip"),
        os.path.join(HIP_DIR, "gatemul.hip"),
        os.path.join(HIP_DIR, "pcgemm.hip"),
        os.path.join(HIP_DIR, "ffgemm.hip"),
        os.path.join(HIP_DIR, "wgrad.hip"),
        os.path.join(HIP_DIR, "pairrep.hip"),
        os.path.join(HIP_DIR, "ipacore.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="alphafold2_amd",
    version="0.1.0",
    description="MI355X-native AlphaFold2-style framework "
                "(PyTorch-ROCm + gfx950 HIP kernels + RCCL)",
    packages=find_packages(exclude=("tests",)),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
