"""Unified metrics writer (the reference's ``helpers.grapher.Grapher``
contract: ``add_scalar``/``add_image``/``add_text``/``save``/``close``,
rank-0 only — call sites ``/root/reference/main.py:452-460,521,541-544``).

Backends:
* ``jsonl``  — always available: scalars/text to ``events.jsonl``, images as
  ``.npy`` thumbnails (no tensorboard/visdom in this environment);
* ``tensorboard`` — used when tensorboard/tensorboardX is importable;
* ``visdom`` — used when visdom is importable (flag parity with
  ``--visdom-url``); otherwise falls back to jsonl with a warning.
"""

import json
import os
import time
from typing import Optional

import numpy as np
import torch

__all__ = ["Grapher"]


class _JsonlBackend:
    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        self.logdir = logdir
        self._fh = open(os.path.join(logdir, "events.jsonl"), "a",
                        buffering=1)

    def add_scalar(self, tag, value, step):
        self._fh.write(json.dumps({
            "t": time.time(), "kind": "scalar", "tag": tag,
            "value": float(value), "step": int(step)}) + "\n")

    def add_image(self, tag, img, global_step=0):
        arr = img.detach().cpu().numpy() if torch.is_tensor(img) else \
            np.asarray(img)
        path = os.path.join(self.logdir,
                            f"{tag.replace('/', '_')}_{global_step}.npy")
        np.save(path, arr)
        self._fh.write(json.dumps({
            "t": time.time(), "kind": "image", "tag": tag, "path": path,
            "step": int(global_step)}) + "\n")

    def add_text(self, tag, text, step=0):
        self._fh.write(json.dumps({
            "t": time.time(), "kind": "text", "tag": tag, "text": str(text),
            "step": int(step)}) + "\n")

    def save(self):
        self._fh.flush()

    def close(self):
        self._fh.close()


class Grapher:
    def __init__(self, backend: str = "tensorboard", env: str = "",
                 server: Optional[str] = None, port: Optional[int] = None,
                 log_folder: str = "./runs", logdir: Optional[str] = None):
        logdir = logdir or os.path.join(log_folder, env or "run")
        self.backend_name = backend
        self.backend = None
        if backend == "tensorboard":
            try:
                from torch.utils.tensorboard import SummaryWriter
                self.backend = _TBBackend(SummaryWriter(log_dir=logdir))
            except Exception:
                self.backend = _JsonlBackend(logdir)
        elif backend == "visdom":
            try:
                import visdom  # noqa: F401
                self.backend = _VisdomBackend(env, server, port)
            except Exception:
                print(f"visdom unavailable; logging to {logdir}/events.jsonl")
                self.backend = _JsonlBackend(logdir)
        else:
            self.backend = _JsonlBackend(logdir)

    def add_scalar(self, tag, value, step):
        self.backend.add_scalar(tag, value, step)

    def add_image(self, tag, img, global_step=0):
        self.backend.add_image(tag, img, global_step)

    def add_text(self, tag, text, step=0):
        self.backend.add_text(tag, text, step)

    def save(self):
        self.backend.save()

    def close(self):
        self.backend.close()


class _TBBackend:
    def __init__(self, writer):
        self.writer = writer

    def add_scalar(self, tag, value, step):
        self.writer.add_scalar(tag, value, step)

    def add_image(self, tag, img, global_step=0):
        self.writer.add_image(tag, img, global_step)

    def add_text(self, tag, text, step=0):
        self.writer.add_text(tag, text, step)

    def save(self):
        self.writer.flush()

    def close(self):
        self.writer.close()


class _VisdomBackend:
    def __init__(self, env, server, port):
        import visdom
        self.viz = visdom.Visdom(server=server, port=port, env=env)

    def add_scalar(self, tag, value, step):
        self.viz.line(X=np.array([step]), Y=np.array([value]), win=tag,
                      update="append", opts={"title": tag})

    def add_image(self, tag, img, global_step=0):
        arr = img.detach().cpu().numpy() if torch.is_tensor(img) else img
        self.viz.image(arr, win=tag, opts={"title": tag})

    def add_text(self, tag, text, step=0):
        self.viz.text(str(text), win=tag)

    def save(self):
        pass

    def close(self):
        pass
