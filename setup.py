"""Package build: `pip install -e .` or `python setup.py build_ext --inplace`.

Builds the C++ scheduler core in-tree (the gfx950 HIP extension is built via
hivedscheduler_amd/ops/build.py, which needs PyTorch's hipcc driver).
"""
import os
import sys

from setuptools import find_packages, setup
from setuptools.command.build_ext import build_ext
from setuptools import Extension


class CoreBuildExt(build_ext):
    def run(self):
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        from hivedscheduler_amd.core import build as core_build

        core_build.build()


setup(
    name="hivedscheduler-amd",
    version="0.1.0",
    description="MI355X-native topology-aware gang scheduler for Kubernetes",
    packages=find_packages(include=["hivedscheduler_amd*"]),
    package_data={"hivedscheduler_amd": ["*.so", "core/*.hpp", "core/*.cpp", "ops/*.hip"]},
    python_requires=">=3.8",
    install_requires=["pyyaml", "requests"],
    extras_require={
        "server": ["fastapi", "uvicorn", "prometheus_client"],
    },
    ext_modules=[Extension("hivedscheduler_amd.hivedcore", sources=[])],
    cmdclass={"build_ext": CoreBuildExt},
    entry_points={
        "console_scripts": [
            "hivedscheduler-amd=hivedscheduler_amd.__main__:main",
            "hivedscheduler-amd-agent=hivedscheduler_amd.agent.__main__:main",
        ]
    },
)
