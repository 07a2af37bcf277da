"""cordum-mi355x package setup.

The HIP extension is built IN-TREE (cordum_amd/ops/_build/) via
`python -m cordum_amd.ops` machinery or `__graft_entry__.build()` — not by
setup.py — so that the .so stays inside the repo checkout (it must travel
with repo snapshots, not live in site-packages).
"""
from setuptools import find_packages, setup

setup(
    name="cordum-mi355x",
    version="0.1.0",
    description="MI355X-native control plane for autonomous agent workflows",
    packages=find_packages(include=["cordum_amd", "cordum_amd.*"]),
    python_requires=">=3.10",
    install_requires=["pyyaml", "requests"],
    extras_require={
        "gateway": ["fastapi", "uvicorn"],
        "grpc": ["grpcio"],
    },
    entry_points={
        "console_scripts": [
            "cordumctl = cordum_amd.cli.cordumctl:main",
        ]
    },
)
