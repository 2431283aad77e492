"""Package build for skdist-mi355x (reference analog: sk-dist setup.py).

The HIP extension is compiled in-tree by ``python -m skdist_amd.ops.build``
(hipcc --offload-arch=gfx950); ``build_ext`` delegates to it so
``pip install -e .`` / ``python setup.py build_ext --inplace`` work on a
ROCm machine.  The pure-Python package installs anywhere; GPU ops then
require the prebuilt ``_skdist_hip.so`` next to ``skdist_amd/ops``.
"""

from setuptools import Command, find_packages, setup


class BuildHip(Command):
    user_options = []

    def initialize_options(self):
        pass

    def finalize_options(self):
        pass

    def run(self):
        from skdist_amd.ops.build import build

        build()


setup(
    name="skdist-mi355x",
    version="0.1.0",
    description=(
        "MI355X-native distributed meta-estimator engine with the "
        "sk-dist scikit-learn-compatible API"
    ),
    packages=find_packages(include=["skdist_amd", "skdist_amd.*"]),
    package_data={"skdist_amd.ops": ["*.so", "csrc/*"]},
    python_requires=">=3.9",
    install_requires=["numpy", "scipy", "scikit-learn", "pandas"],
    cmdclass={"build_hip": BuildHip},
)
