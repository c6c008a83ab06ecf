"""In-tree build of the MI355X HIP extension (gfx950 only).

    python setup.py build_ext --inplace
        (or)  python -m pytorch_r2d2_dpg_amd.ops.build

torch.utils.cpp_extension would HIPIFY the sources on ROCm; this framework
is native HIP, so the build drives hipcc directly (ops/build.py). The .so
lands in pytorch_r2d2_dpg_amd/ops/ so it travels with the source tree.
"""

import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from setuptools.command.build_ext import build_ext as _build_ext


class HipccBuildExt(_build_ext):
    def run(self):
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        from pytorch_r2d2_dpg_amd.ops.build import build

        build()


setup(
    name="pytorch_r2d2_dpg_amd",
    version="0.1.0",
    packages=["pytorch_r2d2_dpg_amd"],
    cmdclass={"build_ext": HipccBuildExt},
)
