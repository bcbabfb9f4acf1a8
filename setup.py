"""Build the pvraft_amd HIP (gfx950 / CDNA4) extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting pvraft_amd/_C*.so stays inside the source tree so it travels
with repo snapshots (no JIT cache dependence).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

SOURCES = [
    "pvraft_amd/csrc/bindings.cpp",
    "pvraft_amd/csrc/knn_graph.hip",
    "pvraft_amd/csrc/gather_edge.hip",
    "pvraft_amd/csrc/voxel_corr.hip",
    "pvraft_amd/csrc/knn_corr.hip",
    "pvraft_amd/csrc/group_norm.hip",
    "pvraft_amd/csrc/edge_gnmp.hip",
    "pvraft_amd/csrc/knn_gnmp.hip",
    "pvraft_amd/csrc/cast_pack.hip",
    "pvraft_amd/csrc/pw_wgrad.hip",
    "pvraft_amd/csrc/pw_fwd.hip",
    "pvraft_amd/csrc/transpose.hip",
    "pvraft_amd/csrc/pv_corr_fused.hip",
    "pvraft_amd/csrc/topk_rows.hip",
    "pvraft_amd/csrc/corr_topk.hip",
    "pvraft_amd/csrc/gru_gates.hip",
    "pvraft_amd/csrc/seq_loss.hip",
]

setup(
    name="pvraft_amd",
    version="0.1.0",
    packages=["pvraft_amd"],
    ext_modules=[
        CUDAExtension(
            name="pvraft_amd._C",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
