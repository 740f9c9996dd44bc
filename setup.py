"""Package metadata. The gfx950 HIP extension is built IN-TREE (so the .so
travels with source snapshots) via:

    PYTORCH_ROCM_ARCH=gfx950 python -m dgl_operator_amd.csrc.build

Install offline with `pip install -e . --no-build-isolation`.
"""
from setuptools import find_packages, setup

setup(
    name="dgl-operator-amd",
    version="0.2.0",
    description=(
        "MI355X-native distributed-GNN framework + DGLJob operator "
        "(HIP/CDNA4 kernels, RCCL over xGMI)"
    ),
    license="Apache-2.0",
    packages=find_packages(include=["dgl_operator_amd*"]),
    package_data={"dgl_operator_amd": ["_C.so", "csrc/*.hip", "csrc/*.cpp",
                                       "csrc/*.h"]},
    python_requires=">=3.10",
    install_requires=["torch", "pyyaml", "numpy"],
    extras_require={"operator": ["prometheus_client"]},
    entry_points={
        "console_scripts": [
            # the reference images expose dglrun/dglkerun as executables
            # (python/dglrun/exec/); same names after pip install
            "dglrun=dgl_operator_amd.tools.dglrun:main",
            "dglkerun=dgl_operator_amd.tools.dglkerun:main",
            "dgl-operator-manager=dgl_operator_amd.operator_plane.manager:main",
            "dgl-watcher-loop=dgl_operator_amd.operator_plane.watcher:main",
        ],
    },
)
