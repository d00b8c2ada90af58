"""Build/install for pytensor_federated_amd.

``python setup.py build_ext --inplace`` (or ``pip install -e .``) compiles
the CDNA4 HIP extension in-tree via hipcc (gfx950 only -- see
pytensor_federated_amd/ops/build.py).
"""
import sys

from setuptools import Command, find_packages, setup


class BuildHip(Command):
    description = "compile the gfx950 HIP extension in-tree"
    user_options = []

    def initialize_options(self):
        pass

    def finalize_options(self):
        pass

    def run(self):
        sys.path.insert(0, ".")
        from pytensor_federated_amd.ops.build import build

        build(force=True)


setup(
    name="pytensor-federated-amd",
    version="0.2.0",
    description="MI355X-native federated logp/gradient engine",
    packages=find_packages(include=["pytensor_federated_amd*"]),
    package_data={"pytensor_federated_amd.ops": ["*.so", "csrc/*.hip"]},
    python_requires=">=3.10",
    install_requires=["numpy", "psutil", "grpcio", "nest_asyncio"],
    cmdclass={"build_ext": BuildHip, "build_hip": BuildHip},
)
