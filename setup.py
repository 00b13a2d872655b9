"""In-tree build of the CDNA4 HIP extension (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built _tcsdn_hip.so lands inside traffic_classifier_sdn_amd/ops/ so the
repo snapshot that travels to a GPU box carries it (no JIT cache reliance).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import pybind11
from setuptools import Extension
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "traffic_classifier_sdn_amd", "csrc")

setup(
    name="traffic_classifier_sdn_amd",
    version="0.1.0",
    packages=["traffic_classifier_sdn_amd"],
    ext_modules=[
        CUDAExtension(
            name="traffic_classifier_sdn_amd.ops._tcsdn_hip",
            sources=[
                os.path.join(CSRC, "ext.cpp"),
                os.path.join(CSRC, "predict_kernels.hip"),
                os.path.join(CSRC, "knn_mfma.hip"),
                os.path.join(CSRC, "fit_kernels.hip"),
            ],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        ),
        # plain C++ extension: no torch/HIP dependency, loads standalone
        Extension(
            name="traffic_classifier_sdn_amd.flow._tcsdn_native",
            sources=[os.path.join(CSRC, "flowtable.cpp")],
            include_dirs=[pybind11.get_include()],
            extra_compile_args=["-O3", "-std=c++17"],
            language="c++",
        ),
    ],
    cmdclass={"build_ext": BuildExtension},
)
