"""Shim for toolchains whose setuptools predates PEP 621 ([project]
tables landed in setuptools 61; this image ships 59). Mirrors
pyproject.toml so `pip wheel . --no-build-isolation` produces a correct
wheel offline. Build the native engine first (`make -C native`) so the
in-tree hipflux/_native*.so is bundled via package-data.
"""

from setuptools import setup

setup(
    name="selkies-amd",
    version="0.1.0",
    description=("MI355X-native low-latency HTML5 remote desktop "
                 "streaming framework"),
    python_requires=">=3.10",
    install_requires=[
        "aiohttp>=3.9",
        "numpy>=1.24",
        "psutil",
        "prometheus_client",
    ],
    packages=["selkies_amd", "selkies_amd.webrtc", "hipflux"],
    package_data={
        "selkies_amd": ["web/*"],
        "hipflux": ["*.so"],
    },
    include_package_data=True,
    entry_points={
        "console_scripts": [
            "selkies = selkies_amd.__main__:main",
            "selkies-resize = selkies_amd.display_utils:resize_entrypoint",
            "selkies-gpu-probe = selkies_amd.gpu_probe:main",
            "selkies-mux = selkies_amd.mp4:mux_entrypoint",
        ],
    },
)
