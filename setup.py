"""Build the in-tree HIP kernel library: `python setup.py build_ext --inplace`
(or simply `python -c "from scalerl_amd.ops import build_kernels; build_kernels()"`).

The kernels are plain HIP compiled with hipcc --offload-arch=gfx950 into
scalerl_amd/ops/_hip_ops.so (no torch-extension ABI, loaded via ctypes)."""

import sys

from setuptools import find_packages, setup
from setuptools.command.build_ext import build_ext


class BuildHip(build_ext):
    def run(self):
        sys.path.insert(0, ".")
        from scalerl_amd.ops import _backend
        _backend.build(verbose=True)


setup(
    name="scalerl-amd",
    version="0.1.0",
    description="MI355X-native distributed deep-RL engine",
    packages=find_packages(include=["scalerl_amd*"]),
    package_data={"scalerl_amd.ops": ["_hip_ops.so", "csrc/*"]},
    cmdclass={"build_ext": BuildHip},
    python_requires=">=3.10",
)
