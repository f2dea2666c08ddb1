"""Per-phase timing instrumentation (the observability the reference lacks —
SURVEY.md section 5 "Tracing/profiling: none ... New framework: per-phase HIP
events (data/fwd/bwd/opt/collective), images/sec reporting").

``PhaseTimer`` brackets phases with HIP events on GPU (host perf_counter on
CPU) and accumulates per-phase totals; the engine prints a per-epoch summary
line when ``--perf-stats`` is set.  HIP kernels are all named, so
``rocprofv3 --kernel-trace --stats`` gives the per-kernel view
(tools/analyze_prof.py summarizes the resulting rocpd database).
"""

import time
from collections import defaultdict
from typing import Dict, Optional

import torch

__all__ = ["PhaseTimer"]


class PhaseTimer:
    PHASES = ("data", "forward", "loss", "backward", "optimizer")

    def __init__(self, enabled: bool = True, use_cuda: Optional[bool] = None):
        self.enabled = enabled
        self.use_cuda = (torch.cuda.is_available()
                         if use_cuda is None else use_cuda)
        self.totals: Dict[str, float] = defaultdict(float)
        self.samples = 0
        self._events = []  # (phase, start_ev, end_ev)
        self._t0 = None
        self._phase = None

    def start(self, phase: str):
        if not self.enabled:
            return
        self.stop()
        self._phase = phase
        if self.use_cuda:
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            self._t0 = ev
        else:
            self._t0 = time.perf_counter()

    def stop(self):
        if not self.enabled or self._phase is None:
            return
        if self.use_cuda:
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            self._events.append((self._phase, self._t0, ev))
        else:
            self.totals[self._phase] += time.perf_counter() - self._t0
        self._phase = None

    def add_samples(self, n: int):
        self.samples += n

    def epoch_summary(self, reset: bool = True) -> str:
        if not self.enabled:
            return ""
        if self.use_cuda and self._events:
            torch.cuda.synchronize()
            for phase, s, e in self._events:
                self.totals[phase] += s.elapsed_time(e) / 1000.0
            self._events.clear()
        total = sum(self.totals.values())
        # known phases in canonical order, then any ad-hoc ones (e.g. the
        # fused "step" phase of the round-2 engine loop)
        keys = [k for k in self.PHASES if k in self.totals]
        keys += [k for k in self.totals if k not in self.PHASES]
        parts = [f"{k}={self.totals.get(k, 0.0) * 1000:.1f}ms"
                 for k in keys]
        ips = self.samples / total if total > 0 else 0.0
        line = (f"[perf] {' '.join(parts)} total={total * 1000:.1f}ms "
                f"images/sec={ips:.1f}")
        if reset:
            self.totals.clear()
            self.samples = 0
        return line
