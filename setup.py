"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package sources
(hetu_galvatron_amd/ops/_galvatron_hip.*.so) so it travels with repo
snapshots to GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""
import os
from glob import glob

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import (  # noqa: E402
    BuildExtension, CppExtension, CUDAExtension)

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = sorted(f for f in glob(os.path.join(HERE,
             "hetu_galvatron_amd/ops/csrc/*.hip"))
             if not f.endswith("_hip.hip"))  # torch hipify artifacts

setup(
    name="hetu_galvatron_amd_ext",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="hetu_galvatron_amd.ops._galvatron_hip",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
            },
        ),
        CppExtension(
            name="hetu_galvatron_amd._galvatron_dp_core",
            sources=["hetu_galvatron_amd/csrc_cpu/dp_core.cpp"],
            extra_compile_args={"cxx": ["-O3", "-std=c++17"]},
        ),
        CppExtension(
            name="hetu_galvatron_amd._galvatron_dataset_helpers",
            sources=["hetu_galvatron_amd/csrc_cpu/dataset_helpers.cpp"],
            extra_compile_args={"cxx": ["-O3", "-std=c++17"]},
        ),
    ],
    cmdclass={"build_ext": BuildExtension},
)
