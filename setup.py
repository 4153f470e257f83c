"""setup.py: builds the native core in-tree (reference: Horovod's
CMake-driven setup.py; here a single hipcc pass — see build.py)."""
import os
import subprocess
import sys

from setuptools import setup
from setuptools.command.build_ext import build_ext
from setuptools import Extension


class HipBuildExt(build_ext):
    def run(self):
        subprocess.check_call([sys.executable,
                               os.path.join(os.path.dirname(__file__),
                                            "build.py")])


setup(
    name="horovod_amd",
    version="0.1.0",
    description="MI355X-native distributed deep learning (Horovod-compatible)",
    packages=["horovod_amd", "horovod_amd.common", "horovod_amd.torch",
              "horovod_amd.torch.elastic", "horovod_amd.runner",
              "horovod_amd.data", "horovod_amd.models", "horovod_amd.ops",
              "horovod_amd.parallel", "horovod_amd.utils",
              "horovod_amd.spark", "horovod_amd.ray"],
    ext_modules=[Extension("horovod_amd._core", sources=[])],
    cmdclass={"build_ext": HipBuildExt},
    scripts=["bin/hvdrun", "bin/horovodrun"],
    python_requires=">=3.8",
)
