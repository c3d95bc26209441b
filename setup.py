"""Editable install + in-tree HIP extension build.

    pip install -e .            # package only (CPU golden engine)
    python setup.py build_hip   # compile the gfx950 extension in-tree

The extension deliberately builds IN-TREE (ops/_build) rather than into
site-packages so a repo snapshot carries its own .so to GPU machines.
"""

import sys

from setuptools import Command, find_packages, setup


class BuildHip(Command):
    description = "compile the CDNA4 (gfx950) HIP extension in-tree"
    user_options = []

    def initialize_options(self):
        pass

    def finalize_options(self):
        pass

    def run(self):
        from nn_distributed_training_amd.ops.build import build

        build(verbose=True)
        print("HIP extension built in-tree.")


setup(
    name="nn_distributed_training_amd",
    version="0.1.0",
    description=(
        "MI355X-native decentralized consensus training (DiNNO/DSGD/"
        "DSGT) with hand-written CDNA4 HIP kernels and RCCL neighbor "
        "exchange"
    ),
    packages=find_packages(
        include=["nn_distributed_training_amd*"]
    ),
    package_data={
        "nn_distributed_training_amd.ops": ["hip/*.hip", "hip/*.h"],
    },
    python_requires=">=3.9",
    install_requires=["torch", "networkx", "numpy", "scipy", "pyyaml"],
    cmdclass={"build_hip": BuildHip},
)
