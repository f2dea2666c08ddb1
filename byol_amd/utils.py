"""Misc run-environment helpers (the reference's ``helpers.utils`` probes:
``get_aws_instance_id``/``get_slurm_id`` for run tagging,
``number_of_gpus`` for launch-mode choice — call sites
``/root/reference/main.py:128,775,800``)."""

import contextlib
import os
from typing import Optional

import torch

__all__ = ["get_slurm_id", "get_aws_instance_id", "number_of_gpus",
           "dummy_context"]


def get_slurm_id() -> Optional[str]:
    for var in ("SLURM_ARRAY_JOB_ID", "SLURM_JOB_ID", "SLURM_JOBID"):
        if os.environ.get(var):
            task = os.environ.get("SLURM_ARRAY_TASK_ID")
            return f"{os.environ[var]}_{task}" if task else os.environ[var]
    return None


def get_aws_instance_id() -> Optional[str]:
    """EC2 metadata probe (offline environments return None quickly)."""
    try:
        import urllib.request
        req = urllib.request.Request(
            "http://169.254.169.254/latest/meta-data/instance-id")
        with urllib.request.urlopen(req, timeout=0.2) as resp:
            return resp.read().decode()
    except Exception:
        return None


def number_of_gpus() -> int:
    return torch.cuda.device_count() if torch.cuda.is_available() else 0


def dummy_context():
    return contextlib.nullcontext()
