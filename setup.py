"""Build the in-tree gfx950 HIP extension: python setup.py build_ext --inplace.

Compiles for gfx950 ONLY (MI355X / CDNA4); no CUDA path, no multi-arch
fatbins. The built .so lands inside dmosopt_amd/ so it travels with the
source tree to GPU boxes.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import find_packages, setup
from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "dmosopt_amd", "ops", "hip")

# libhdf5 location (this image ships it under /opt/conda)
HDF5_PREFIX = os.environ.get("DMOSOPT_AMD_HDF5_PREFIX", "/opt/conda")

ext = CUDAExtension(
    name="dmosopt_amd._hipops",
    sources=[
        os.path.join("dmosopt_amd", "ops", "hip", "bindings.cpp"),
        os.path.join("dmosopt_amd", "ops", "hip", "matern.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "cholesky.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "pareto.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "variation.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "hv_mc.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "cmaes_update.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "sceua_stage.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "moea_ops.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "hv_exact.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "matern_bf16.hip"),
        os.path.join("dmosopt_amd", "ops", "hip", "agemoea_survival.hip"),
    ],
    include_dirs=[HIP_DIR],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

h5ext = CppExtension(
    name="dmosopt_amd._h5core",
    sources=[os.path.join("dmosopt_amd", "storage", "h5cpp", "h5core.cpp")],
    include_dirs=[os.path.join(HDF5_PREFIX, "include")],
    library_dirs=[os.path.join(HDF5_PREFIX, "lib")],
    libraries=["hdf5"],
    extra_compile_args=["-O2", "-std=c++17"],
    extra_link_args=[f"-Wl,-rpath,{os.path.join(HDF5_PREFIX, 'lib')}"],
)

setup(
    name="dmosopt_amd",
    version="0.1.0",
    description=(
        "MI355X-native multi-objective adaptive surrogate optimization "
        "(MO-ASMO) with gfx950 HIP kernels"
    ),
    packages=find_packages(include=["dmosopt_amd", "dmosopt_amd.*"]),
    ext_modules=[ext, h5ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
    entry_points={
        "console_scripts": [
            "dmosopt-analyze=dmosopt_amd.cli.analyze:main",
            "dmosopt-train=dmosopt_amd.cli.train:main",
            "dmosopt-onestep=dmosopt_amd.cli.onestep:main",
        ]
    },
)
