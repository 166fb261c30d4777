"""Build for kfac_pytorch_amd.

`python setup.py build_ext --inplace` compiles the gfx950 HIP extension
in-tree (kfac_pytorch_amd/ops/_kfac_hip*.so) by driving hipcc directly --
no hipify, no CUDA build path.
"""

import os
import sys

from setuptools import find_packages, setup
from setuptools.command.build_ext import build_ext as _build_ext

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)


class HipBuildExt(_build_ext):
    def run(self):
        from kfac_pytorch_amd.ops.build import build_all
        build_all(force=False)

    def build_extensions(self):  # pragma: no cover
        pass


setup(
    name="kfac_pytorch_amd",
    version="0.1.0",
    description="MI355X-native distributed K-FAC for PyTorch-ROCm",
    packages=find_packages(include=["kfac_pytorch_amd",
                                    "kfac_pytorch_amd.*"]),
    package_data={"kfac_pytorch_amd.ops": ["*.so", "csrc/*.hip",
                                           "csrc_rccl/*.hip",
                                           "csrc_solver/*.hip"]},
    cmdclass={"build_ext": HipBuildExt},
    python_requires=">=3.9",
)
