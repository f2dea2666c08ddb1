"""Warmup + cosine LR scheduling with the reference's exact semantics
(``/root/reference/optimizers/scheduler.py``): a ``LinearWarmup`` LambdaLR
ramps 0 -> 1 over ``warmup_steps`` scheduler steps (the engine steps the
scheduler once per EPOCH), flips ``complete`` after, and a ``Scheduler``
container routes ``step``/``get_last_lr``/``state_dict`` to the warmup until
it completes, then to the cosine schedule."""

from torch.optim.lr_scheduler import LambdaLR

__all__ = ["Scheduler", "LinearWarmup"]


class Scheduler:
    """Container for a warmup scheduler followed by a normal scheduler."""

    def __init__(self, normal_scheduler, warmup_scheduler=None):
        self.warmup = warmup_scheduler
        self.sched = normal_scheduler

    def get_last_lr(self):
        if self.warmup is not None and not self.warmup.complete:
            return self.warmup.get_last_lr()
        return self.sched.get_last_lr()

    def state_dict(self):
        return {
            "warmup": self.warmup.state_dict() if self.warmup is not None else {},
            "sched": self.sched.state_dict(),
        }

    def load_state_dict(self, state_dict):
        if self.warmup:
            self.warmup.load_state_dict(state_dict["warmup"])
        self.sched.load_state_dict(state_dict["sched"])

    def step(self, *args, **kwargs):
        if self.warmup is not None and not self.warmup.complete:
            return self.warmup.step(*args, **kwargs)
        return self.sched.step(*args, **kwargs)


class LinearWarmup(LambdaLR):
    """Linear 0 -> 1 over ``warmup_steps`` scheduler steps, then constant 1
    (and ``complete=True``, which hands control to the cosine schedule)."""

    def __init__(self, optimizer, warmup_steps, last_epoch=-1):
        self.warmup_steps = warmup_steps
        self.complete = False
        super().__init__(optimizer, self.lr_lambda, last_epoch=last_epoch)

    def lr_lambda(self, step):
        if step < self.warmup_steps:
            return float(step) / float(max(1.0, self.warmup_steps))
        self.complete = True
        return 1.0

    # `complete` must survive checkpoint/resume: LambdaLR.state_dict drops
    # callables but keeps plain attributes; make sure both ride along.
    def state_dict(self):
        sd = {k: v for k, v in self.__dict__.items()
              if k not in ("optimizer", "lr_lambdas")}
        return sd

    def load_state_dict(self, state_dict):
        self.__dict__.update(
            {k: v for k, v in state_dict.items() if k != "lr_lambdas"})
