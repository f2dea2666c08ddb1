"""hipGraph capture/replay of the whole BYOL training step.

The eager step costs ~2k kernel launches (4 encoder passes x 53 convs + BN
pairs + backward + fused optimizer) with ~1-2 us of host/gap time per
launch boundary on MI355X.  Capturing the step as ONE hipGraph
(`torch.cuda.CUDAGraph` is hipGraph on ROCm) and replaying it removes that
host overhead entirely.

Two per-step dynamic scalars are routed through 1-float device tensors so
replays see fresh values without recapture:

* the CosEMA cosine-ramped decay (changes EVERY step) — the EMA kernel
  reads ``CosEMA._decay_dev`` (``csrc/ema.hip``);
* the LR (changes per scheduler epoch) — the fused LARS update kernel
  reads ``FlatLarsState.lr_dev`` (``csrc/lars.hip``).

``replay()`` refreshes both scalars, advances the host-side EMA step
counter (the captured Python increments only once, at capture), copies the
new minibatch into the graph's static input buffers, and launches the
graph.

Constraints (standard CUDA/hipGraph whole-network capture): static shapes,
no host syncs inside the step, allocations inside capture come from the
graph's private pool.  All byol_amd kernels satisfy this; MIOpen finds must
be warmed up first (the wrapper runs ``warmup_steps`` eager iterations
before capturing).  Reference parity: the captured step computes exactly
the eager step — capture changes scheduling, not numerics.
"""

from typing import Callable, Optional, Tuple

import torch

__all__ = ["GraphedTrainStep"]


class GraphedTrainStep:
    """Capture ``step_fn(a1, a2, labels) -> loss`` into a hipGraph.

    ``step_fn`` must run the FULL training step (zero_grad, forward, loss,
    backward, optimizer step) with no host-side data-dependent control
    flow.  ``model`` is the BYOL module (or a wrapper exposing
    ``.module``); ``lars`` the LARS optimizer with an attached flat space
    (its fused path must have engaged at least once before capture).
    """

    def __init__(self, model, lars, step_fn: Callable,
                 static_inputs: Tuple[torch.Tensor, ...],
                 warmup_steps: int = 2):
        byol = model.module if hasattr(model, "module") else model
        self.ema = byol.target_network
        self.lars = lars
        self.step_fn = step_fn
        self.static_inputs = static_inputs
        self.warmup_steps = warmup_steps
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.static_loss: Optional[torch.Tensor] = None
        self._last_lr: Optional[float] = None

    def _current_lr(self) -> float:
        return float(self.lars.optim.param_groups[0]["lr"])

    def capture(self) -> None:
        """Run eager warmup steps on the static buffers, then capture."""
        device = self.static_inputs[0].device
        for _ in range(self.warmup_steps):
            self.step_fn(*self.static_inputs)
        fused = getattr(self.lars, "_fused", None)
        if fused is None:
            raise RuntimeError(
                "fused LARS state missing after warmup — graph capture "
                "requires the fused flat-space step (GPU + SGD momentum)")
        # pre-fill the device scalars the captured kernels will read
        self.ema.ensure_decay_dev(device)
        self._last_lr = self._current_lr()
        fused.lr_dev.fill_(self._last_lr)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_loss = self.step_fn(*self.static_inputs)
        torch.cuda.synchronize()

    def replay(self, *inputs: torch.Tensor) -> torch.Tensor:
        """Copy ``inputs`` into the static buffers (pass nothing to reuse
        the resident batch), refresh the dynamic scalars, replay."""
        assert self.graph is not None, "capture() first"
        for dst, src in zip(self.static_inputs, inputs):
            if dst.data_ptr() != src.data_ptr():
                dst.copy_(src, non_blocking=True)
        # per-step cosine decay + host step counter (the captured python
        # ran exactly once; replays advance state here)
        self.ema._decay_dev.fill_(self.ema.current_decay())
        self.ema.step += 1
        lr = self._current_lr()
        if lr != self._last_lr:
            self.lars._fused.lr_dev.fill_(lr)
            self._last_lr = lr
        self.graph.replay()
        return self.static_loss
