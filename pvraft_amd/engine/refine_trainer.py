"""Refine-stage training engine (reference tools/engine_refine.py).

Same skeleton as Trainer; differences (engine_refine.py:50-62,110,142,199):
* model = PVRaftRefine with the backbone frozen (explicit
  requires_grad_(False) + the model's own no_grad forward -- doing cleanly
  what engine_refine.py:51-54 intends).
* stage-1 weights loaded with strict=False (the refine head is new).
* loss = compute_loss on the single refined flow (no sequence weighting).
* validation runs args.iters GRU iterations (32 in run.sh:3).
"""

from __future__ import annotations

import os

import torch

from pvraft_amd.model import PVRaftRefine
from pvraft_amd.parallel import broadcast_module
from pvraft_amd.utils import compute_loss, load_checkpoint

from .trainer import Trainer


class RefineTrainer(Trainer):
    loss_is_sequence = False

    def _make_model(self):
        model = PVRaftRefine.from_args(self.args)
        model.freeze_backbone()
        return model

    def _loss(self, est_flow, batch):
        return compute_loss(self._final_flow(est_flow), batch)

    def _eval_iters(self) -> int:
        return self.args.iters

    def _load_weights(self, weights: str):
        """Stage-1 weights: path or experiment name, strict=False."""
        path = weights
        if not os.path.isfile(path):
            path = os.path.join(
                self.args.root, "experiments", weights, "checkpoints", "best_checkpoint.params"
            )
        load_checkpoint(path, self.model, strict=False)
        # stage-1 epochs do not advance the refine schedule
        self.begin_epoch = 1
        broadcast_module(self.model)
        self.log.info(f"Loaded stage-1 weights from {path} (strict=False)")
