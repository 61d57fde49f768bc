"""Build the geomx_amd native extension (_geops) for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The extension is built IN-TREE (geomx_amd/_geops.*.so) so it travels to
GPU machines with the source snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import (BuildExtension,  # noqa: E402
                                       CppExtension, CUDAExtension)

here = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="geomx_amd._geops",
    sources=[
        "geomx_amd/csrc/geops.cpp",
        "geomx_amd/csrc/kernels.hip",
        "geomx_amd/csrc/conv.hip",
        "geomx_amd/csrc/convwrw.hip",
        "geomx_amd/csrc/convwrw2.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

# host-side IO runtime (record-file reader) — plain C++, no HIP
io_ext = CppExtension(
    name="geomx_amd._geoio",
    sources=["geomx_amd/csrc/recordio.cpp"],
    extra_compile_args=["-O3", "-std=c++17"],
)

setup(
    name="geomx_amd",
    version="0.1.0",
    packages=["geomx_amd", "geomx_amd.kvstore", "geomx_amd.ops",
              "geomx_amd.parallel", "geomx_amd.models", "geomx_amd.utils"],
    ext_modules=[ext, io_ext],
    cmdclass={"build_ext": BuildExtension},
)
