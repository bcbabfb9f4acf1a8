from .trainer import Trainer, VAL_ITERS
from .refine_trainer import RefineTrainer
from .predictor import Predictor
