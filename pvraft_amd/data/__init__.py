from .batch import Batch
from .base import SceneFlowDataset
from .flyingthings3d import FT3D
from .kitti import Kitti
from .synthetic import SyntheticSceneFlow, synthetic_batch
from .prefetch import CudaPrefetcher
