"""pvraft_amd: an MI355X-native (CDNA4/gfx950) scene-flow framework with the
capabilities of PV-RAFT (CVPR 2021) -- point-voxel correlation fields for
scene-flow estimation on point clouds.

Compute path: PyTorch-ROCm autograd driver + hand-written HIP kernels for
the hot ops (pvraft_amd/ops/hip), RCCL over xGMI for multi-GPU data
parallelism (one process per GPU).
"""
__version__ = "0.1.0"
