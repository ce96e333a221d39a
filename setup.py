"""Packaging for DeepConsensus-AMD (reference setup.py parity: console
entry point `deepconsensus`, package metadata)."""
import os

from setuptools import find_packages, setup


def read_version():
    init = os.path.join(
        os.path.dirname(__file__), "deepconsensus_amd", "__init__.py"
    )
    for line in open(init):
        if line.startswith("__version__"):
            return line.split('"')[1]
    return "0.0.0"


setup(
    name="deepconsensus-amd",
    version=read_version(),
    description=(
        "MI355X-native consensus calling for PacBio CCS reads: "
        "PyTorch-ROCm + hand-written HIP/CDNA4 kernels + RCCL over xGMI"
    ),
    packages=find_packages(include=["deepconsensus_amd*"]),
    package_data={
        "deepconsensus_amd.ops": ["hip/*.hip", "_build/*.so"],
        "deepconsensus_amd.preprocess": ["*.cpp"],
    },
    python_requires=">=3.10",
    install_requires=["numpy", "torch"],
    entry_points={
        "console_scripts": [
            "deepconsensus=deepconsensus_amd.cli:main",
        ]
    },
)
