"""In-tree build of the dblink_amd native extension (gfx950 HIP + host C++).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces ``dblink_amd/_C*.so`` next to the Python package so the built
artifact travels with the source tree (no site-packages install, no JIT
cache dependence).
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "dblink_amd", "ops", "csrc")

ext = CUDAExtension(
    name="dblink_amd._C",
    sources=[
        os.path.join(CSRC, "ext.cpp"),
        os.path.join(CSRC, "sim_pairs_cpu.cpp"),
        os.path.join(CSRC, "link_dense_cpu.cpp"),
        os.path.join(CSRC, "kernels.hip"),
    ],
    include_dirs=[CSRC],
    extra_compile_args={
        "cxx": ["-O3", "-fopenmp", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
    extra_link_args=["-fopenmp"],
)

setup(
    name="dblink_amd",
    version="0.1.0",
    packages=find_packages(include=["dblink_amd", "dblink_amd.*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
