"""Packaging (reference setup.py). The HIP extensions build in-tree via
``python -m ray_lightning_amd.ops.build`` (gfx950); this setup ships the
Python package and the pre-built .so files when present."""
from setuptools import find_packages, setup

setup(
    name="ray_lightning_amd",
    packages=find_packages(include=["ray_lightning_amd*"]),
    package_data={"ray_lightning_amd.ops": ["*.so", "csrc/*.hip"]},
    version="0.1.0",
    author="ray_lightning_amd authors",
    description="MI355X-native actor-launched distributed training "
                "strategies (RayStrategy / RayShardedStrategy / "
                "HorovodRayStrategy) with a native RCCL/HIP stack",
    long_description="See README.md",
    url="https://example.invalid/ray_lightning_amd",
    install_requires=["torch", "numpy", "cloudpickle"],
)
