"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package (hyperspace_amd/_hip*.so) so it
travels with repo snapshots.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

setup(
    name="hyperspace_amd",
    version="0.1.0",
    description="MI355X-native covering-index acceleration engine",
    packages=["hyperspace_amd"],
    ext_modules=[
        CUDAExtension(
            name="hyperspace_amd._hip",
            sources=[
                "hyperspace_amd/csrc/binding.cpp",
                "hyperspace_amd/csrc/kernels/kernels.hip",
            ],
            include_dirs=[os.path.join(ROOT, "hyperspace_amd", "csrc")],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
