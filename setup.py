"""pip-installable package stub (reference: setup.py:1-9).

The HIP extension is built IN-TREE (csrc/build) via
`python -c 'import __graft_entry__; __graft_entry__.build()'` so the
.so stays with the repo; this setup only installs the Python package.
"""

from setuptools import find_packages, setup

setup(
    name="shallowspeed_amd",
    version="0.1.0",
    description="MI355X-native minimal distributed-training engine "
                "(DP + PP over RCCL/xGMI, HIP/CDNA4 kernels)",
    packages=find_packages(include=["shallowspeed_amd*"]),
    python_requires=">=3.10",
    install_requires=["torch"],
)
