"""Warmup -> cosine LR scheduling.

Behavior pinned to the reference's contract (call sites
``/root/reference/main.py:289-291,763``; class surface
``/root/reference/optimizers/scheduler.py``): the engine steps the
scheduler once per EPOCH; LR ramps linearly 0 -> base over ``warmup_steps``
steps, then a cosine anneal (constructed separately with its own internal
step counter) takes over for the remaining ``epochs - warmup`` steps.

The implementation is our own: ``LinearWarmup`` writes the param-group LRs
directly from the captured ``initial_lr`` values (no LambdaLR subclass, no
closure serialization caveats), and ``Scheduler`` is a two-phase router
whose active phase is decided by the warmup's ``complete`` latch.  Both
``complete`` and the step counter ride in ``state_dict`` so a resumed run
continues in the correct phase.
"""

__all__ = ["Scheduler", "LinearWarmup"]


class LinearWarmup:
    """Linear LR ramp: factor step/warmup_steps for step < warmup_steps,
    then a constant 1.0 and ``complete = True`` (the router's handoff
    latch).

    Mirrors torch scheduler conventions so it is interchangeable with one
    in checkpoints: construction applies step 0 (LR starts at 0), each
    ``step()`` advances the counter then writes the LRs, ``initial_lr`` is
    recorded on (or reused from) the param groups.
    """

    def __init__(self, optimizer, warmup_steps, last_epoch=-1):
        self.optimizer = optimizer
        self.warmup_steps = warmup_steps
        self.complete = False
        self.last_epoch = last_epoch
        if last_epoch == -1:
            for group in optimizer.param_groups:
                group.setdefault("initial_lr", group["lr"])
        self.base_lrs = [g["initial_lr"] for g in optimizer.param_groups]
        self._last_lr = [g["lr"] for g in optimizer.param_groups]
        self.step()  # apply step 0, like torch schedulers at last_epoch=-1

    def _factor(self) -> float:
        if self.last_epoch < self.warmup_steps:
            return float(self.last_epoch) / float(max(1.0, self.warmup_steps))
        self.complete = True
        return 1.0

    def step(self):
        self.last_epoch += 1
        factor = self._factor()
        self._last_lr = []
        for group, base in zip(self.optimizer.param_groups, self.base_lrs):
            group["lr"] = base * factor
            self._last_lr.append(group["lr"])

    def get_last_lr(self):
        return list(self._last_lr)

    def state_dict(self):
        return {k: v for k, v in self.__dict__.items() if k != "optimizer"}

    def load_state_dict(self, state_dict):
        self.__dict__.update(state_dict)


class Scheduler:
    """Two-phase LR router: drives the warmup until its ``complete`` latch
    flips, then the main (cosine) schedule — which keeps its own private
    step counter, so the anneal starts from its step 1 at handoff.
    ``state_dict`` nests both phases' states under fixed keys."""

    def __init__(self, normal_scheduler, warmup_scheduler=None):
        self.warmup = warmup_scheduler
        self.sched = normal_scheduler

    def _active(self):
        in_warmup = self.warmup is not None and not self.warmup.complete
        return self.warmup if in_warmup else self.sched

    def step(self, *args, **kwargs):
        return self._active().step(*args, **kwargs)

    def get_last_lr(self):
        return self._active().get_last_lr()

    def state_dict(self):
        warmup_state = {} if self.warmup is None else self.warmup.state_dict()
        return {"warmup": warmup_state, "sched": self.sched.state_dict()}

    def load_state_dict(self, state_dict):
        if self.warmup is not None:
            self.warmup.load_state_dict(state_dict["warmup"])
        self.sched.load_state_dict(state_dict["sched"])
