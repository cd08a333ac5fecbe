"""fei_amd packaging.

Console scripts mirror the reference's entry points (setup.py:38-41) plus
the memdir/memorychain CLIs. The gfx950 kernel library is built in-tree by
``python -m fei_amd.ops.build`` (hipcc) — deliberately NOT a setuptools
ext_module so the .so stays next to the sources and travels with repo
snapshots instead of site-packages.
"""

from setuptools import find_packages, setup

setup(
    name="fei-amd",
    version="0.1.0",
    description="MI355X-native code-assistant framework (local HIP inference engine)",
    packages=find_packages(include=["fei_amd", "fei_amd.*"]),
    python_requires=">=3.10",
    install_requires=[
        "requests",
        "flask",
        "numpy",
    ],
    extras_require={
        "ui": ["rich", "textual"],
    },
    entry_points={
        "console_scripts": [
            "fei=fei_amd.ui.cli:main",
            "memdir=fei_amd.memdir.cli:main",
            "memdir-server=fei_amd.memdir.run_server:main",
            "memorychain=fei_amd.memorychain.cli:main",
            "fei-api=fei_amd.serve.api:main",
        ],
    },
    package_data={"fei_amd.ops": ["csrc/*.hip", "csrc/*.h", "*.so"],
                  "fei_amd.engine": ["fei16k.model", "fei16k.vocab"]},
)
