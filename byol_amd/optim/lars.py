"""LARS (Layer-wise Adaptive Rate Scaling) optimizer wrapper.

Reference semantics (``/root/reference/optimizers/lars.py``):

* wraps an arbitrary inner optimizer (here usually SGD+momentum);
* before the inner step, per parameter: add weight decay to the grad, then —
  only for groups carrying an explicit ``ignore: False`` key (set by
  :func:`byol_amd.layers.add_weight_decay`) — scale the grad by
  ``trust_coef * ||p|| / (||grad|| + eps)`` when both norms are positive;
* the inner optimizer's ``weight_decay`` is zeroed during the step and
  restored after (decay already folded into the grad);
* delegates ``param_groups``/``state``/``state_dict``/``zero_grad``.

MI355X fast path: when the model parameters live in a
:class:`~byol_amd.models.byol.FlatParamSpace` and the inner optimizer is
SGD+momentum, the whole step (per-tensor norms, weight decay, trust scaling,
momentum, parameter update — the reference's ~160 tensors x ~6 ATen kernels)
collapses into three HIP kernel launches over the flat buffers
(:func:`byol_amd.ops.lars_step.fused_lars_momentum_step`).
"""

import contextlib
from typing import Optional

import torch
from torch.optim.optimizer import Optimizer

__all__ = ["LARS"]


class LARS(Optimizer):
    def __init__(self, optimizer, eps: float = 1e-8,
                 trust_coef: float = 0.001):
        if eps < 0.0:
            raise ValueError(f"invalid epsilon value: {eps}")
        if trust_coef < 0.0:
            raise ValueError(f"invalid trust coefficient: {trust_coef}")
        self.optim = optimizer
        self.eps = eps
        self.trust_coef = trust_coef
        self._fused = None  # lazily-attached FusedLARSMomentum or None

    def __getstate__(self):
        return (self.optim, {"eps": self.eps, "trust_coef": self.trust_coef})

    def __setstate__(self, state):
        self.optim, d = state
        self.eps = d["eps"]
        self.trust_coef = d["trust_coef"]
        self._fused = None

    def __repr__(self):
        return f"{self.__class__.__name__}({self.optim!r})"

    @property
    def param_groups(self):
        return self.optim.param_groups

    @property
    def state(self):
        return self.optim.state

    def state_dict(self):
        return self.optim.state_dict()

    def load_state_dict(self, state_dict):
        self.optim.load_state_dict(state_dict)

    def zero_grad(self, set_to_none: bool = True):
        # flat-space grads must stay as views; zero in place instead
        if getattr(self, "_flat_space", None) is not None:
            self._flat_space.zero_grads()
        else:
            self.optim.zero_grad(set_to_none=set_to_none)

    def add_param_group(self, param_group):
        self.optim.add_param_group(param_group)

    def attach_flat_space(self, flat_space):
        """Enable the fused flat step and flat zero_grad.  Verifies every
        parameter still views the flat buffer (catches post-``finalize``
        device/format moves early, instead of silently training a model
        whose weights the optimizer no longer touches)."""
        flat_space.verify()
        self._flat_space = flat_space

    @torch.no_grad()
    def apply_adaptive_lrs(self):
        """Precondition grads in place, batched with ``_foreach`` kernels
        (the reference does this one tensor and ~6 ATen calls at a time):

        1. every group: ``grad += weight_decay * p`` (decay folded into the
           grad so the inner optimizer must NOT apply it again — see
           :meth:`step`);
        2. groups tagged ``ignore: False`` by ``add_weight_decay``: scale
           each grad by ``trust_coef * ||p|| / (||grad|| + eps)``, with a
           neutral 1.0 whenever either norm is zero.

        In-place mutation (vs the reference's out-of-place rebind) keeps
        ``p.grad`` a view of the flat grad buffer.
        """
        for group in self.optim.param_groups:
            weight_decay = group["weight_decay"]
            adapt = group.get("ignore", None) is not None \
                and not group["ignore"]
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            grads = [p.grad for p in params]
            if weight_decay > 0:
                torch._foreach_add_(grads, params, alpha=weight_decay)
            if adapt:
                p_norms = torch.stack(torch._foreach_norm(params))
                g_norms = torch.stack(torch._foreach_norm(grads))
                ratios = torch.where(
                    (p_norms > 0) & (g_norms > 0),
                    self.trust_coef * p_norms / (g_norms + self.eps),
                    torch.ones_like(p_norms))
                torch._foreach_mul_(grads, list(ratios.unbind(0)))

    @contextlib.contextmanager
    def _suppress_inner_weight_decay(self):
        """The decay is already in the grads; zero the inner optimizer's
        ``weight_decay`` for the duration of its step, restore after."""
        saved = [g["weight_decay"] for g in self.optim.param_groups]
        for g in self.optim.param_groups:
            g["weight_decay"] = 0
        try:
            yield
        finally:
            for g, wd in zip(self.optim.param_groups, saved):
                g["weight_decay"] = wd

    def _try_fused_step(self) -> bool:
        space = getattr(self, "_flat_space", None)
        if space is None or not space.flat_params.is_cuda:
            return False
        from ..ops import lars_step as fused
        return fused.fused_lars_momentum_step(self, space)

    def step(self, *args, **kwargs):
        # GPU + flat params + SGD/momentum inner: the whole multi-tensor
        # step runs as three HIP kernel launches
        if self._try_fused_step():
            return None
        self.apply_adaptive_lrs()
        with self._suppress_inner_weight_decay():
            return self.optim.step(*args, **kwargs)
