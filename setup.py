"""Build the metis_amd gfx950 HIP extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python3 setup.py build_ext --inplace

The built `metis_amd/_hip_ops*.so` is git-ignored but travels to the GPU
box with the gpurun snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "metis_amd", "ops", "hip")

sources = [
    os.path.join(HIP_DIR, "bindings.cpp"),
    os.path.join(HIP_DIR, "layernorm.hip"),
    os.path.join(HIP_DIR, "adamw.hip"),
    os.path.join(HIP_DIR, "attention.hip"),
    os.path.join(HIP_DIR, "attention_bwd.hip"),
    os.path.join(HIP_DIR, "gemm.hip"),
    os.path.join(HIP_DIR, "gemm8.hip"),
    os.path.join(HIP_DIR, "rmsnorm.hip"),
    os.path.join(HIP_DIR, "rope.hip"),
    os.path.join(HIP_DIR, "swiglu.hip"),
    os.path.join(HIP_DIR, "cross_entropy.hip"),
    os.path.join(HIP_DIR, "relayout.hip"),
    os.path.join(HIP_DIR, "qkv_rope.hip"),
    os.path.join(HIP_DIR, "decode.hip"),
    os.path.join(HIP_DIR, "lt_gemm.cpp"),
]
sources = [s for s in sources if os.path.exists(s)]

setup(
    name="metis_amd_hip_ops",
    ext_modules=[
        CUDAExtension(
            name="metis_amd._hip_ops",
            sources=sources,
            libraries=["hipblaslt"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
