"""BYOL model, MI355X-native flat-parameter design.

Capability parity with the reference BYOL module
(``/root/reference/main.py:133-276``): online encoder (ResNet minus fc),
projector MLP, predictor MLP, detached linear probe, and an EMA "target
network" with cosine-ramped decay over the *entire* parameter vector.

The reference realises the target network with a pack/swap/restore trick:
every forward packs all ~37M params to a vector, loads the EMA vector into the
modules, runs the pass, restores, then packs again for the EMA update — seven
full parameter traversals, ~1 GB of HBM copy traffic per step
(``/root/reference/main.py:212-227,255``).  On MI355X we instead keep every
parameter as a *view into one contiguous flat fp32 buffer*:

* the EMA update is ONE fused lerp over the flat buffer (two reads, one
  write — ~0.45 GB instead of ~1 GB, in one kernel launch instead of ~320);
* the target pass is ``torch.func.functional_call`` with parameter views into
  the EMA buffer — zero copies, zero kernel launches for the "swap";
* DDP gradient all-reduce and the fused LARS step operate on matching flat
  gradient storage (see ``byol_amd/parallel/ddp.py`` and
  ``byol_amd/ops/lars_step.py``).

Numerics are the reference's exactly:

* CosEMA decay ``tau = 1 - (1-tau_base) * (cos(pi*k/K)+1)/2`` stepped once per
  training-mode forward, primed once at construction (so ``mean`` starts at
  ``(1-tau_base) * params`` and ``step`` at 1), frozen in eval
  (``/root/reference/main.py:147-163,211-212``);
* target passes run the SAME module objects in the CURRENT train/eval mode, so
  BatchNorm uses batch stats and updates running stats during target passes in
  training, exactly as the reference's in-place param swap does;
* the target pass runs under ``no_grad`` — the reference builds a throwaway
  autograd graph there but detaches every target output before use
  (``/root/reference/objective.py:23-24``, ``main.py:252``), so outputs and
  BN-stat side effects are identical while we skip the dead graph.
"""

import math
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from .resnet import build_encoder

__all__ = ["BYOL", "CosEMA", "FlatParamSpace"]


class FlatParamSpace:
    """All parameters of a module as views into one contiguous flat buffer.

    Call :meth:`flatten` once, after any structural change (SyncBN conversion)
    and after the final device/dtype move.  Afterwards:

    * ``flat_params`` is the single fp32 tensor holding every parameter;
    * ``flat_grads`` is matching gradient storage; every ``p.grad`` is a view,
      so autograd accumulates straight into it (bucketed DDP all-reduce and
      the fused LARS step read it with zero repacking);
    * ``named_views(buffer)`` maps parameter names onto any same-sized flat
      tensor (used for the EMA target weights).
    """

    def __init__(self, module: nn.Module):
        self.module = module
        self.flat_params: Optional[torch.Tensor] = None
        self.flat_grads: Optional[torch.Tensor] = None
        # name -> (offset, numel, shape), in module.named_parameters() order
        self.layout: List[Tuple[str, int, int, torch.Size]] = []
        self.numel = 0

    def flatten(self) -> torch.Tensor:
        params = list(self.module.named_parameters())
        assert params, "module has no parameters"
        self.numel = sum(p.numel() for _, p in params)
        p0 = params[0][1]
        self.flat_params = torch.empty(
            self.numel, dtype=p0.dtype, device=p0.device)
        self.flat_grads = torch.zeros_like(self.flat_params)
        self.layout = []
        offset = 0
        for name, p in params:
            n = p.numel()
            self.layout.append((name, offset, n, p.shape))
            view = self.flat_params[offset:offset + n].view(p.shape)
            view.copy_(p.detach())
            p.data = view
            p.grad = self.flat_grads[offset:offset + n].view(p.shape)
            offset += n
        return self.flat_params

    @property
    def is_flat(self) -> bool:
        return self.flat_params is not None

    def named_views(self, buffer: torch.Tensor,
                    prefix: str = "") -> Dict[str, torch.Tensor]:
        """Views of ``buffer`` shaped like each parameter.

        ``prefix`` filters and strips a leading qualifier (e.g.
        ``"base_network."``) so the dict can feed ``functional_call`` on a
        submodule."""
        assert buffer.numel() == self.numel
        out = {}
        for name, offset, n, shape in self.layout:
            if prefix and not name.startswith(prefix):
                continue
            out[name[len(prefix):]] = buffer[offset:offset + n].view(shape)
        return out

    def verify(self) -> None:
        """Raise if any parameter is no longer a view of the flat buffer —
        catches the silent footgun of a device/memory-format move AFTER
        ``finalize()`` (``Module.to`` rebinds ``param.data`` to fresh
        storage, detaching it from the buffer the fused EMA/LARS/DDP ops
        update)."""
        base_ptr = self.flat_params.data_ptr()
        end_ptr = base_ptr + self.flat_params.numel() \
            * self.flat_params.element_size()
        for name, p in self.module.named_parameters():
            ptr = p.data_ptr()
            if not (base_ptr <= ptr < end_ptr):
                raise RuntimeError(
                    f"parameter {name!r} is detached from the flat buffer "
                    "— structural changes (.to, SyncBN conversion, memory-"
                    "format moves) must happen BEFORE finalize()")

    def zero_grads(self) -> None:
        self.flat_grads.zero_()

    def grad_segments(self) -> List[Tuple[str, int, int]]:
        return [(name, off, n) for name, off, n, _ in self.layout]


class CosEMA(nn.Module):
    """Cosine-ramped EMA of a flat parameter vector.

    Semantics of the reference CosEMA (``/root/reference/main.py:133-164``):
    ``decay(k) = 1 - (1 - base) * (cos(pi*k/K) + 1) / 2`` with
    ``mean <- (1-decay)*x + decay*mean``; ``step`` advances only in training
    mode; the first (priming) call sees ``mean == 0``.  The update here is one
    fused lerp over the flat buffer (HIP kernel on GPU, ATen lerp elsewhere)
    instead of the reference's pack-then-blend.
    """

    def __init__(self, total_steps: int, base_decay: float = 0.996):
        super().__init__()
        self.step = 0
        self.total_steps = max(int(total_steps), 1)
        self.base_decay = float(base_decay)
        self.register_buffer("mean", None)
        self._decay_dev = None  # 1-float device scalar for hipGraph capture

    def current_decay(self) -> float:
        return 1.0 - (1.0 - self.base_decay) * (
            math.cos(math.pi * self.step / self.total_steps) + 1.0) / 2.0

    def ensure_decay_dev(self, device) -> torch.Tensor:
        """Device scalar holding the current decay — the hipGraph replay
        wrapper refreshes it (and advances ``step``) between replays."""
        if self._decay_dev is None or self._decay_dev.device != device:
            self._decay_dev = torch.zeros(1, device=device,
                                          dtype=torch.float32)
        self._decay_dev.fill_(self.current_decay())
        return self._decay_dev

    @torch.no_grad()
    def forward(self, flat: torch.Tensor) -> torch.Tensor:
        if self.mean is None:
            self.mean = torch.zeros_like(flat)
        if self.training:
            from ..ops import ema as ema_ops
            if flat.is_cuda and torch.cuda.is_current_stream_capturing():
                # graph capture: decay comes from the device scalar and the
                # replay wrapper owns the host-side step advancement
                if self._decay_dev is None:
                    raise RuntimeError(
                        "CosEMA captured without ensure_decay_dev(); the "
                        "graph wrapper must create the decay scalar first")
                ema_ops.flat_ema_update(self.mean, flat, 0.0,
                                        decay_dev=self._decay_dev)
            else:
                ema_ops.flat_ema_update(self.mean, flat,
                                        self.current_decay())
                self.step += 1
        return flat

    # step/total/base ride the state_dict alongside the mean buffer so
    # resume restores the cosine ramp position (reference bundles these via
    # its checkpoint closure; /root/reference/main.py:749).
    def get_extra_state(self):
        return {"step": self.step, "total_steps": self.total_steps,
                "base_decay": self.base_decay}

    def set_extra_state(self, state):
        self.step = int(state["step"])
        self.total_steps = int(state["total_steps"])
        self.base_decay = float(state["base_decay"])


def _mlp_head(in_dim: int, latent_dim: int, out_dim: int) -> nn.Sequential:
    # projector/predictor shape of the reference
    # (/root/reference/main.py:194-205).  The BN1d carries the ReLU fused
    # into its HIP kernel; the Identity keeps the reference's Sequential
    # indices (head.0 linear, head.1 bn, head.3 linear).
    from ..ops.bn import FusedBatchNorm
    return nn.Sequential(
        nn.Linear(in_dim, latent_dim),
        FusedBatchNorm(latent_dim, relu=True),
        nn.Identity(),
        nn.Linear(latent_dim, out_dim),
    )


class _PredictionStack(nn.Module):
    """Encoder -> projector -> predictor as one callable, sharing BYOL's
    submodules, for ``functional_call`` on the EMA parameter views."""

    def __init__(self, base_network, head, predictor, repr_size):
        super().__init__()
        self.base_network = base_network
        self.head = head
        self.predictor = predictor
        self.repr_size = repr_size

    def forward(self, x):
        representation = self.base_network(x).view(-1, self.repr_size)
        projection = self.head(representation)
        prediction = self.predictor(projection)
        return representation, projection, prediction


class BYOL(nn.Module):
    """BYOL with online+EMA-target encoders, projector/predictor and a
    detached linear probe; forward contract identical to the reference
    (``/root/reference/main.py:167-276``): two augmentations in, a 13-entry
    dict of named tensors out."""

    def __init__(self, arch: str,
                 base_network_output_size: int,
                 projection_output_size: int,
                 classifier_output_size: int,
                 total_training_steps: int,
                 head_latent_size: int = 4096,
                 base_decay: float = 0.996,
                 in_channels: int = 3):
        super().__init__()
        self.base_network_output_size = base_network_output_size

        self.base_network = build_encoder(arch, in_channels=in_channels)
        if self.base_network.out_channels != base_network_output_size:
            raise ValueError(
                f"--representation-size {base_network_output_size} does not "
                f"match {arch} output {self.base_network.out_channels}")
        self.head = _mlp_head(base_network_output_size, head_latent_size,
                              projection_output_size)
        self.predictor = _mlp_head(projection_output_size, head_latent_size,
                                   projection_output_size)
        self.linear_classifier = nn.Linear(base_network_output_size,
                                           classifier_output_size)
        self.target_network = CosEMA(total_training_steps, base_decay)

        # populated by finalize(); not registered as a submodule
        self.__dict__["_flat_space"] = None
        self.__dict__["_pred_stack"] = None
        self.__dict__["_target_param_views"] = None

    # -- flat-parameter finalization -------------------------------------
    def finalize(self) -> "BYOL":
        """Flatten parameters and prime the EMA.  Call exactly once, after
        SyncBN conversion and the final device move (the reference primes its
        EMA inside ``__init__``; value-wise identical since conversion/move
        preserve parameter values)."""
        assert self._flat_space is None, "finalize() already called"
        space = FlatParamSpace(self)
        space.flatten()
        self.__dict__["_flat_space"] = space
        self.__dict__["_pred_stack"] = _PredictionStack(
            self.base_network, self.head, self.predictor,
            self.base_network_output_size)
        # Prime the EMA exactly as the reference does at construction
        # (training mode, step 0 -> mean = (1-base)*params, step -> 1).
        was_training = self.training
        self.train()
        self.target_network(space.flat_params)
        self.train(was_training)
        self._rebuild_target_views()
        return self

    def _rebuild_target_views(self):
        space = self._flat_space
        mean = self.target_network.mean
        views = {}
        for sub in ("base_network.", "head.", "predictor."):
            for rel, v in space.named_views(mean, prefix=sub).items():
                views[sub + rel] = v
        self.__dict__["_target_param_views"] = views

    @property
    def flat_space(self) -> FlatParamSpace:
        assert self._flat_space is not None, "call finalize() first"
        return self._flat_space

    # -- forward ----------------------------------------------------------
    def prediction(self, augmentation: torch.Tensor):
        representation = self.base_network(augmentation).view(
            -1, self.base_network_output_size)
        projection = self.head(representation)
        prediction = self.predictor(projection)
        return representation, projection, prediction

    def target_prediction(self, augmentation: torch.Tensor):
        """Run encoder->head->predictor with the EMA weights: zero-copy
        ``functional_call`` on parameter views into the EMA buffer (vs the
        reference's three full param-vector copies,
        ``/root/reference/main.py:214-227``).  Buffers are NOT substituted, so
        BN running stats behave exactly as the reference's in-place swap."""
        assert self._flat_space is not None, "call finalize() first"
        with torch.no_grad():
            return torch.func.functional_call(
                self._pred_stack, self._target_param_views, (augmentation,))

    def forward(self, augmentation1: torch.Tensor,
                augmentation2: torch.Tensor) -> Dict[str, torch.Tensor]:
        # parameters are frozen for the duration of one forward (LARS step,
        # EMA lerp and DDP broadcast all happen outside it), so the conv
        # weight-layout cache is valid exactly until the next forward
        from ..ops.conv import clear_weight_cache
        clear_weight_cache()
        online_representation1, online_projection1, online_prediction1 = \
            self.prediction(augmentation1)
        online_representation2, online_projection2, online_prediction2 = \
            self.prediction(augmentation2)
        target_representation1, target_projection1, target_prediction1 = \
            self.target_prediction(augmentation1)
        target_representation2, target_projection2, target_prediction2 = \
            self.target_prediction(augmentation2)

        # Probe sees detached representations; both views in train mode,
        # one in eval (/root/reference/main.py:249-252).
        repr_to_classifier = (
            torch.cat([online_representation1, online_representation2], 0)
            if self.training else online_representation1)
        linear_preds = self.linear_classifier(
            repr_to_classifier.clone().detach())

        # EMA update: ONE fused lerp over the flat buffer.
        self.target_network(self.flat_space.flat_params)

        return {
            "linear_preds": linear_preds,
            "online_representation1": online_representation1,
            "online_projection1": online_projection1,
            "online_prediction1": online_prediction1,
            "online_representation2": online_representation2,
            "online_projection2": online_projection2,
            "online_prediction2": online_prediction2,
            "target_representation1": target_representation1,
            "target_projection1": target_projection1,
            "target_prediction1": target_prediction1,
            "target_representation2": target_representation2,
            "target_projection2": target_projection2,
            "target_prediction2": target_prediction2,
        }

    # state_dict hooks: the flat views serialize/deserialize transparently
    # (load copies into the existing view tensors, keeping the flat space).
    def load_state_dict(self, state_dict, strict: bool = True, assign: bool = False):
        assert not assign, "assign=True would break the flat parameter space"
        result = super().load_state_dict(state_dict, strict=strict)
        if self._flat_space is not None and self.target_network.mean is not None:
            self._rebuild_target_views()
        return result
