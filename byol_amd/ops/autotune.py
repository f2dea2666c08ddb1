"""Runtime per-shape conv dispatch autotuner (opt-in).

The default dispatch tables in :mod:`byol_amd.ops.conv` were measured at
bs=512-class M on MI355X.  For other batch sizes, image sizes, or
architectures, ``BYOL_MFMA_CONV1X1=autotune`` (and
``BYOL_MFMA_CONV3X3=autotune``) switches to cudnn.benchmark-style runtime
selection: the FIRST call of an unseen (op, shape) times the MFMA kernel
against MIOpen over a few iterations and caches the winner for the rest of
the process.  Set ``BYOL_AUTOTUNE_CACHE=/path.json`` to persist decisions
across runs (read at import of the first decision, appended on update).

The measurement itself synchronizes the device (once per shape), exactly
like ``torch.backends.cudnn.benchmark``'s find step.
"""

import json
import os
import time
from typing import Callable, Dict, Tuple

import torch

__all__ = ["Autotuner", "autotuner"]


class Autotuner:
    def __init__(self, cache_path: str = None, iters: int = 3,
                 timer: Callable[[Callable], float] = None):
        self.iters = iters
        self.cache_path = cache_path or os.environ.get(
            "BYOL_AUTOTUNE_CACHE")
        self.decisions: Dict[Tuple, bool] = {}
        self._timer = timer or self._time_cuda
        if self.cache_path and os.path.isfile(self.cache_path):
            try:
                with open(self.cache_path) as fh:
                    for entry in json.load(fh):
                        self.decisions[tuple(entry["key"])] = entry["ours"]
            except (OSError, ValueError, KeyError):
                pass

    # -- timing -----------------------------------------------------------
    def _time_cuda(self, fn: Callable) -> float:
        fn()  # warm (alloc, module caches)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(self.iters):
            fn()
        torch.cuda.synchronize()
        return time.perf_counter() - t0

    # -- decisions --------------------------------------------------------
    def choose(self, key: Tuple, ours: Callable, theirs: Callable) -> bool:
        """Return True if OUR kernel should run for this key (timing both
        on first sight, cached afterwards)."""
        hit = self.decisions.get(key)
        if hit is not None:
            return hit
        t_ours = self._timer(ours)
        t_theirs = self._timer(theirs)
        win = t_ours <= t_theirs
        self.decisions[key] = win
        self._persist()
        return win

    def _persist(self):
        if not self.cache_path:
            return
        try:
            payload = [{"key": list(k), "ours": v}
                       for k, v in sorted(self.decisions.items())]
            tmp = self.cache_path + ".tmp"
            with open(tmp, "w") as fh:
                json.dump(payload, fh, indent=0)
            os.replace(tmp, self.cache_path)
        except OSError:
            pass


_instance = None


def autotuner() -> Autotuner:
    global _instance
    if _instance is None:
        _instance = Autotuner()
    return _instance
