"""Checkpoint bundle + best-model saver / early stopper.

Replicates the reference's ``helpers.layers.append_save_and_load_fns`` +
``ModelSaver`` contract (call sites ``/root/reference/main.py:749-769``):

* checkpoints bundle {model state (incl. CosEMA mean + step via extra
  state), optimizer state, scheduler state, epoch, args/uid};
* ``ModelSaver(model, early_stop, rank, burn_in_interval, larger_is_better,
  max_early_stop_steps)``; ``saver(loss) -> bool`` (True => stop);
  ``restore() -> {'epoch': int}``; no saving during the burn-in; patience
  ``max_early_stop_steps``; rank-0 writes, every rank restores.
"""

import os
from typing import Optional

import torch

__all__ = ["CheckpointBundle", "ModelSaver", "get_name"]


def get_name(args) -> str:
    """Deterministic run name from args + uid (the reference's
    ``helpers.utils.get_name``)."""
    uid = getattr(args, "uid", "") or ""
    parts = [
        uid or "byol",
        getattr(args, "arch", "arch"),
        f"bs{getattr(args, 'batch_size', 0)}",
        f"r{getattr(args, 'num_replicas', 1)}",
        getattr(args, "optimizer", "opt"),
    ]
    return "_".join(str(p) for p in parts)


class CheckpointBundle:
    """save/load closures over {model, optimizer, scheduler, args}."""

    def __init__(self, model, optimizer, scheduler, args):
        self.model = model
        self.optimizer = optimizer
        self.scheduler = scheduler
        self.args = args

    def _module(self):
        return self.model.module if hasattr(self.model, "module") \
            else self.model

    def path(self) -> str:
        model_dir = getattr(self.args, "model_dir", ".models")
        os.makedirs(model_dir, exist_ok=True)
        return os.path.join(model_dir, get_name(self.args) + ".pt")

    def save(self, epoch: int, extra: Optional[dict] = None):
        payload = {
            "epoch": epoch,
            "model": self._module().state_dict(),
            "optimizer": self.optimizer.state_dict()
            if self.optimizer is not None else None,
            "scheduler": self.scheduler.state_dict()
            if self.scheduler is not None else None,
            "args": {k: v for k, v in vars(self.args).items()
                     if isinstance(v, (int, float, str, bool, type(None),
                                       list, tuple))},
        }
        if extra:
            payload.update(extra)
        tmp = self.path() + ".tmp"
        torch.save(payload, tmp)
        os.replace(tmp, self.path())

    def load(self) -> dict:
        path = self.path()
        if not os.path.isfile(path):
            return {"epoch": 1}
        payload = torch.load(path, map_location="cpu", weights_only=False)
        self._module().load_state_dict(payload["model"])
        if self.optimizer is not None and payload.get("optimizer"):
            self.optimizer.load_state_dict(payload["optimizer"])
        if self.scheduler is not None and payload.get("scheduler"):
            self.scheduler.load_state_dict(payload["scheduler"])
        return {"epoch": payload.get("epoch", 0) + 1,
                "best_loss": payload.get("best_loss")}


class ModelSaver:
    def __init__(self, bundle: CheckpointBundle, early_stop: bool = False,
                 rank: int = 0, burn_in_interval: int = 0,
                 larger_is_better: bool = False,
                 max_early_stop_steps: int = 10):
        self.bundle = bundle
        self.early_stop = early_stop
        self.rank = rank
        self.burn_in_interval = burn_in_interval
        self.larger_is_better = larger_is_better
        self.patience = max_early_stop_steps
        self.best: Optional[float] = None
        self.stale = 0
        self.epoch = 0

    def restore(self) -> dict:
        state = self.bundle.load()
        self.epoch = state["epoch"] - 1
        # re-seed best from the checkpoint so a resumed run does not
        # overwrite the best checkpoint with a worse post-restart epoch
        if state.get("best_loss") is not None:
            self.best = state["best_loss"]
        return state

    def __call__(self, loss: float) -> bool:
        """Record this epoch's test loss; save when best; return True to
        request an early stop."""
        self.epoch += 1
        if self.epoch <= self.burn_in_interval:
            return False
        improved = (self.best is None
                    or (loss > self.best if self.larger_is_better
                        else loss < self.best))
        if improved:
            self.best = loss
            self.stale = 0
            if self.rank == 0:
                self.bundle.save(self.epoch, extra={"best_loss": loss})
        else:
            self.stale += 1
        return self.early_stop and self.stale >= self.patience
