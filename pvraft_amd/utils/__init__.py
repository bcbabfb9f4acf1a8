from .loss import compute_loss, sequence_loss
from .metrics import compute_epe, compute_epe_train
from .checkpoint import (
    save_checkpoint,
    load_checkpoint,
    checkpoint_dir,
    save_train_state,
    load_train_state,
)
from .logging import setup_logger, ScalarLogger
