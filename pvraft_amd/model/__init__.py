from .graph import Graph
from .setconv import SetConv
from .encoder import PointEncoder
from .corr import CorrBlock, CorrField
from .update import UpdateBlock, MotionEncoder, ConvGRU, FlowHead
from .refine import RefineHead
from .pvraft import PVRaft, PVRaftRefine, build_model

__all__ = [
    "Graph",
    "SetConv",
    "PointEncoder",
    "CorrBlock",
    "CorrField",
    "UpdateBlock",
    "MotionEncoder",
    "ConvGRU",
    "FlowHead",
    "RefineHead",
    "PVRaft",
    "PVRaftRefine",
    "build_model",
]
