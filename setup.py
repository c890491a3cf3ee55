"""Packaging for multiverso_amd (and the `multiverso` drop-in shim —
reference binding/python/setup.py parity).

The gfx950 HIP extension is built in-tree (so the .so ships with the
source tree): `python __graft_entry__.py build` or
`python -c "from multiverso_amd import ops; ops.build()"`.
"""

from setuptools import find_packages, setup

setup(
    name="multiverso-amd",
    version="0.1.0",
    description=("MI355X-native parameter-server training framework with "
                 "Microsoft Multiverso's capabilities (C/Python/Lua/C# API "
                 "surface, table/checkpoint format) on RCCL over xGMI"),
    packages=(find_packages(include=["multiverso_amd", "multiverso_amd.*"])
              + ["multiverso"]),
    package_dir={"multiverso": "binding/python/multiverso"},
    package_data={"multiverso_amd.ops": ["csrc/*", "_build/*.so"],
                  "multiverso_amd.capi": ["*.h", "*.hpp", "*.cpp", "*.so"]},
    python_requires=">=3.8",
    install_requires=["torch"],
)
