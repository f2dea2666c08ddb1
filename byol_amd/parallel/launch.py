"""Process topology & distributed init, with the reference's three launch
modes (``/root/reference/main.py:695-729,786-814``):

(a) ``num_replicas == 1``  -> direct ``run(0)``, no process group;
(b) single node, #GPUs == num_replicas -> ``mp.spawn``, one process per GPU,
    ``CUDA_VISIBLE_DEVICES`` pinned per rank;
(c) otherwise -> one process : one GPU (SLURM/multi-node), rank from
    ``--distributed-rank``, exactly one visible GPU asserted.

Extras over the reference: torchrun env (RANK/LOCAL_RANK/WORLD_SIZE) is
auto-detected, and CPU multi-process uses gloo so the distributed path is
testable without GPUs (the reference would fail on 'nccl' there).
Backend "nccl" IS RCCL on ROCm — the xGMI collective path.
"""

import os
from typing import Callable

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

__all__ = ["init_multiprocessing_and_cuda", "launch", "cleanup"]


def _endpoint(args) -> str:
    master = args.distributed_master or "127.0.0.1"
    prefix = "" if master.startswith("tcp://") else "tcp://"
    return f"{prefix}{master}:{args.distributed_port}"


def init_multiprocessing_and_cuda(rank: int, args):
    """Per-process device pinning, seeding and process-group init; divides
    the global batch size by the world size (reference semantics)."""
    if getattr(args, "multi_gpu_distributed", False):
        os.environ["CUDA_VISIBLE_DEVICES"] = str(rank)
        args.distributed_rank = rank

    # per-rank MIOpen user-DB copy: ranks sharing one sqlite DB contend on
    # its lock (main.py sets a single parent dir before spawn)
    udb = os.environ.get("MIOPEN_USER_DB_PATH")
    if udb and args.num_replicas > 1 and not udb.endswith(f".r{rank}"):
        import shutil
        rank_dir = f"{udb}.r{rank}"
        if os.path.isdir(udb):
            shutil.copytree(udb, rank_dir, dirs_exist_ok=True)
            os.environ["MIOPEN_USER_DB_PATH"] = rank_dir

    args.cuda = not args.no_cuda and torch.cuda.is_available()
    if args.cuda:
        # reference parity (cudnn.benchmark=True -> MIOpen exhaustive
        # find); with MIOPEN_FIND_MODE set (main.py defaults it to HYBRID
        # + the shipped tuned DB) the fast find path is used instead
        torch.backends.cudnn.benchmark = (
            os.environ.get("MIOPEN_FIND_MODE", "") == "")

    if args.seed is not None:
        import numpy as np
        np.random.seed(args.seed)
        torch.manual_seed(args.seed)
        if args.cuda:
            torch.cuda.manual_seed_all(args.seed)

    if args.num_replicas > 1:
        backend = "nccl" if args.cuda else "gloo"
        init_method = os.environ.get("MASTER_ADDR", _endpoint(args))
        if not init_method.startswith("tcp://"):
            port = os.environ.get("MASTER_PORT", str(args.distributed_port))
            init_method = f"tcp://{init_method}:{port}"
        dist.init_process_group(backend=backend, init_method=init_method,
                                world_size=args.num_replicas,
                                rank=args.distributed_rank)
        args.batch_size = args.batch_size // args.num_replicas
    return args


def cleanup():
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()


def _spawn_target(rank, run_fn, args):
    args.distributed_rank = rank
    run_fn(rank, args)


def launch(run_fn: Callable, args):
    """Dispatch to one of the three launch modes (or torchrun env)."""
    args.multi_gpu_distributed = False

    # torchrun / torch.distributed.run environment
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        args.num_replicas = int(os.environ["WORLD_SIZE"])
        args.distributed_rank = int(os.environ["RANK"])
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        return run_fn(args.distributed_rank, args)

    if args.num_replicas > 1:
        assert args.distributed_master is not None, \
            "Specify --distributed-master for DDP."
        os.environ["MASTER_ADDR"] = _endpoint(args)
        os.environ["MASTER_PORT"] = str(args.distributed_port)
        num_gpus = torch.cuda.device_count()
        if num_gpus == args.num_replicas:
            args.multi_gpu_distributed = True
            mp.spawn(_spawn_target, nprocs=args.num_replicas,
                     args=(run_fn, args))
        else:
            assert num_gpus <= 1, \
                "Only 1 GPU per process supported; filter with " \
                "CUDA_VISIBLE_DEVICES."
            run_fn(args.distributed_rank, args)
    else:
        run_fn(0, args)
