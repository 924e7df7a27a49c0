"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built ft_sgemm_amd/_C*.so travels with the repo snapshot to GPU boxes
(it is git-ignored but not gpurun-ignored).
"""

import glob
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

setup(
    name="ft_sgemm_amd",
    version="0.3.0",
    packages=["ft_sgemm_amd"],
    ext_modules=[
        CUDAExtension(
            name="ft_sgemm_amd._C",
            sources=[
                "csrc/torch_ext.cpp",
                "csrc/dispatch.hip",
                "csrc/rocblas_path.hip",
            ] + sorted(p for p in glob.glob("csrc/generated/kernel_*.hip")
                       if not p.endswith("_hip.hip")),
            include_dirs=[os.path.join(ROOT, "csrc")],
            libraries=["rocblas", "roctx64"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
