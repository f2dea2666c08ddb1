"""Fused multi-tensor LARS + SGD-momentum step on the flat parameter space.

The reference's LARS step runs ~6 ATen kernels per parameter tensor (~160
tensors: two norm reductions, wd add, grad scale, momentum blend, param
update — ``/root/reference/optimizers/lars.py:84-127`` + torch SGD).  Here the
whole step is three HIP launches over the flat param/grad/momentum buffers:

1. ``lars_norms``  — per-segment squared norms of p and (g + wd*p), one
   pass, block-per-chunk with atomics into per-segment accumulators;
2. ``lars_ratio``  — tiny kernel computing the per-segment adaptive LR
   ``trust * ||p|| / (||g_eff|| + eps)`` (1 for skipped/degenerate segments);
3. ``lars_update`` — fused ``g_eff = (g + wd*p) * alr``; ``m = mu*m + g_eff``;
   ``p -= lr*m`` (matching torch SGD momentum semantics: first step sets the
   buffer to g_eff).

Semantics match the eager LARS wrapper bit-for-bit-modulo-reduction-order
(tests compare against it at fp32 tolerance).
"""

from typing import Optional

import torch

from . import require_extension

__all__ = ["fused_lars_momentum_step", "FlatLarsState"]


class FlatLarsState:
    """Per-optimizer cached descriptor of the flat segments + momentum buf."""

    def __init__(self, lars, space):
        inner = lars.optim
        flat = space.flat_params
        device = flat.device
        # map data_ptr -> (offset, numel) from the flat layout
        by_ptr = {}
        for name, off, n, shape in space.layout:
            p = flat[off:off + n]
            by_ptr[p.data_ptr()] = (off, n)
        offs, lens, wds, adapts = [], [], [], []
        for group in inner.param_groups:
            wd = float(group["weight_decay"])
            ignore = group.get("ignore", None)
            adapt = 1 if (ignore is not None and not ignore) else 0
            for p in group["params"]:
                key = p.data_ptr()
                if key not in by_ptr:
                    raise RuntimeError(
                        "optimizer param not in flat space; fused LARS "
                        "requires all params to be flat views")
                off, n = by_ptr[key]
                offs.append(off)
                lens.append(n)
                wds.append(wd)
                adapts.append(adapt)
        self.seg_off = torch.tensor(offs, dtype=torch.int64, device=device)
        self.seg_len = torch.tensor(lens, dtype=torch.int64, device=device)
        self.seg_wd = torch.tensor(wds, dtype=torch.float32, device=device)
        self.seg_adapt = torch.tensor(adapts, dtype=torch.int32,
                                      device=device)
        self.nseg = len(offs)
        # 1-float device scalar for hipGraph capture: the captured update
        # kernel reads lr from here; the replay wrapper refreshes it when
        # the scheduler changes the LR
        self.lr_dev = torch.zeros(1, dtype=torch.float32, device=device)
        self.momentum = torch.zeros_like(flat)
        self.momentum_initialized = False
        self.norm_acc = torch.zeros(2 * self.nseg, dtype=torch.float32,
                                    device=device)
        self.alr = torch.ones(self.nseg, dtype=torch.float32, device=device)
        # chunk table: (seg_idx, chunk_start_within_seg) flattened for a
        # fixed 1D grid; rebuilt only if layout changes (it never does)
        self.chunk = 65536
        seg_idx, seg_base = [], []
        for i, (o, n) in enumerate(zip(offs, lens)):
            for c in range(0, n, self.chunk):
                seg_idx.append(i)
                seg_base.append(c)
        self.chunk_seg = torch.tensor(seg_idx, dtype=torch.int32,
                                      device=device)
        self.chunk_base = torch.tensor(seg_base, dtype=torch.int64,
                                       device=device)
        self.nchunks = len(seg_idx)

    def state_dict(self):
        return {"momentum": self.momentum,
                "momentum_initialized": self.momentum_initialized}

    def load_state_dict(self, sd):
        self.momentum.copy_(sd["momentum"])
        self.momentum_initialized = bool(sd["momentum_initialized"])


def _inner_is_plain_sgd_momentum(inner) -> bool:
    if not isinstance(inner, torch.optim.SGD):
        return False
    for g in inner.param_groups:
        if (g.get("momentum", 0.0) == 0.0 or g.get("dampening", 0.0) != 0.0
                or g.get("nesterov", False) or g.get("maximize", False)):
            return False
    # single lr/momentum across groups is what the builder produces
    lrs = {g["lr"] for g in inner.param_groups}
    mus = {g["momentum"] for g in inner.param_groups}
    return len(lrs) == 1 and len(mus) == 1


def fused_lars_momentum_step(lars, space) -> bool:
    """Run the fused step if applicable; returns False to fall back."""
    inner = lars.optim
    if not _inner_is_plain_sgd_momentum(inner):
        return False
    ext = require_extension("fused LARS step")
    state: Optional[FlatLarsState] = getattr(lars, "_fused", None)
    if state is None:
        state = FlatLarsState(lars, space)
        lars._fused = state
        # if the inner optimizer already has momentum state (e.g. restored
        # from an eager-path checkpoint), import it
        _import_momentum(inner, space, state)
    lr = float(inner.param_groups[0]["lr"])
    mu = float(inner.param_groups[0]["momentum"])
    capturing = torch.cuda.is_current_stream_capturing()
    ext.lars_momentum_step(
        space.flat_params, space.flat_grads, state.momentum,
        state.norm_acc, state.alr,
        state.seg_off, state.seg_len, state.seg_wd, state.seg_adapt,
        state.chunk_seg, state.chunk_base,
        float(lars.trust_coef), float(lars.eps), lr, mu,
        1 if state.momentum_initialized else 0,
        # under capture the kernel reads lr from the device scalar so the
        # graph survives scheduler LR changes (wrapper refreshes it)
        state.lr_dev if capturing else None)
    state.momentum_initialized = True
    _export_momentum_lazy(inner, space, state)
    return True


def _import_momentum(inner, space, state):
    flat = space.flat_params
    by_ptr = {}
    for name, off, n, shape in space.layout:
        by_ptr[flat[off:off + n].data_ptr()] = (off, n)
    any_state = False
    for group in inner.param_groups:
        for p in group["params"]:
            st = inner.state.get(p, None)
            if st and st.get("momentum_buffer", None) is not None:
                off, n = by_ptr[p.data_ptr()]
                state.momentum[off:off + n].copy_(
                    st["momentum_buffer"].reshape(-1))
                any_state = True
    state.momentum_initialized = any_state


def _export_momentum_lazy(inner, space, state):
    """Keep torch-SGD state pointing at views of the flat momentum buffer so
    ``state_dict()``/checkpointing see the fused state with zero copies."""
    if getattr(state, "_exported", False):
        return
    flat = space.flat_params
    by_ptr = {}
    for name, off, n, shape in space.layout:
        by_ptr[flat[off:off + n].data_ptr()] = (off, n)
    for group in inner.param_groups:
        for p in group["params"]:
            off, n = by_ptr[p.data_ptr()]
            inner.state[p]["momentum_buffer"] = \
                state.momentum[off:off + n].view(p.shape)
    state._exported = True
