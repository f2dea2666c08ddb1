"""Training engine: setup orchestration + the shared train/eval minibatch
loop, replicating the reference's ``execute_graph``/``train``/``test``/``run``
semantics (``/root/reference/main.py:403-499,559-692,732-783``) on the
MI355X-native stack (flat-param BYOL, FlatDDP over RCCL, custom SyncBN,
fused loss/EMA/LARS HIP kernels)."""

import functools
import pprint
import time

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import layers
from ..data import build_train_and_test_transforms, get_loader
from ..models.byol import BYOL
from ..objective import loss_function
from ..optim import build_optimizer
from ..parallel import FlatDDP, convert_sync_batchnorm
from .grapher import Grapher
from .saver import CheckpointBundle, ModelSaver, get_name

__all__ = ["build_loader_model_grapher", "execute_graph", "train", "test",
           "run", "register_plots", "register_images", "make_grid"]


def _dummy_context():
    import contextlib
    return contextlib.nullcontext()


# ---------------------------------------------------------------------------
# setup
# ---------------------------------------------------------------------------

def build_loader_model_grapher(args):
    """transforms -> loader -> derived sizes -> BYOL -> (SyncBN) -> device ->
    lazy sanity pass -> weight init -> finalize flat space -> DDP -> grapher
    (order mirrors ``/root/reference/main.py:403-462``; finalize happens after
    all structural changes so the flat buffer is terminal)."""
    train_transform, test_transform = build_train_and_test_transforms(args)
    loader = get_loader(train_transform=train_transform,
                        test_transform=test_transform, **vars(args))

    args.input_shape = loader.input_shape
    args.num_train_samples = loader.num_train_samples // args.num_replicas
    args.num_test_samples = loader.num_test_samples  # test is not sharded
    args.num_valid_samples = loader.num_valid_samples // args.num_replicas
    args.steps_per_train_epoch = args.num_train_samples // args.batch_size
    args.total_train_steps = args.epochs * max(args.steps_per_train_epoch, 1)

    network = BYOL(arch=args.arch,
                   base_network_output_size=args.representation_size,
                   projection_output_size=args.projection_size,
                   classifier_output_size=loader.output_size,
                   total_training_steps=args.total_train_steps,
                   head_latent_size=args.head_latent_size,
                   base_decay=args.base_decay,
                   in_channels=loader.input_shape[0])
    if args.convert_to_sync_bn and args.num_replicas > 1:
        network = convert_sync_batchnorm(network)
    if args.cuda:
        network = network.cuda()
        if getattr(args, "channels_last", False):
            # NHWC: the tuned MIOpen solvers + our fused BN kernels' layout
            network.base_network.to(memory_format=torch.channels_last)
    network = layers.init_weights(network, init=args.weight_initialization)
    network.finalize()  # flat param space + EMA prime (terminal)
    lazy_generate_modules(network, loader.train_loader, args)

    if args.num_replicas > 1:
        network = FlatDDP(network,
                          bucket_cap_mb=getattr(args, "bucket_cap_mb", 32.0),
                          broadcast_buffers=getattr(
                              args, "broadcast_buffers", False))

    print("model has {:.3f} million parameters.".format(
        layers.number_of_parameters(network) / 1e6))

    grapher = None
    if args.distributed_rank == 0:
        if args.visdom_url is not None:
            grapher = Grapher("visdom", env=get_name(args),
                              server=args.visdom_url, port=args.visdom_port,
                              log_folder=args.log_dir)
        else:
            import os
            grapher = Grapher("tensorboard",
                              logdir=os.path.join(args.log_dir,
                                                  get_name(args)))
    return loader, network, grapher


def lazy_generate_modules(model, loader, args):
    """One eval-mode sanity forward: prints shapes/dtypes and hard-errors if
    augmented pixels leave [0, 1] (``/root/reference/main.py:465-499``)."""
    model.eval()
    for augmentation1, augmentation2, labels in loader:
        with torch.no_grad():
            print("augmentation1 = {} / {} | augmentation2 = {} / {} | "
                  "labels = {} / {}".format(
                      tuple(augmentation1.shape), augmentation1.dtype,
                      tuple(augmentation2.shape), augmentation2.dtype,
                      tuple(labels.shape), labels.dtype))
            a1_min, a1_max = augmentation1.min(), augmentation1.max()
            a2_min, a2_max = augmentation2.min(), augmentation2.max()
            print(f"aug1 in range [min: {a1_min}, max: {a1_max}] | "
                  f"aug2 in range [min: {a2_min}, max: {a2_max}]")
            # the pipeline relies on ToTensor scaling with NO mean/std
            # normalization (/root/reference/main.py:486-490) — pixels
            # outside [0,1] mean a transform is mis-wired, so fail hard
            for name, lo, hi in (("augmentation1", a1_min, a1_max),
                                 ("augmentation2", a2_min, a2_max)):
                if hi > 1.0 or lo < 0:
                    raise ValueError(
                        f"{name} pixels fall outside [0, 1] "
                        f"(min={float(lo):.4f}, max={float(hi):.4f}); the "
                        "augmentation stack must emit unnormalized "
                        "ToTensor-scaled images.")
            if args.cuda:
                augmentation1 = augmentation1.cuda(non_blocking=True)
                augmentation2 = augmentation2.cuda(non_blocking=True)
            _ = model(augmentation1, augmentation2)
            break
    if args.polyak_ema > 0:
        layers.polyak_ema_parameters(model, args.polyak_ema)


# ---------------------------------------------------------------------------
# plot/image registration (key-name conventions of the reference)
# ---------------------------------------------------------------------------

def make_grid(images: torch.Tensor, nrow: int = 8,
              normalize: bool = True) -> torch.Tensor:
    """Minimal image-grid builder (torchvision.utils.make_grid replacement)."""
    imgs = images.detach().float().cpu()
    if normalize:
        lo, hi = imgs.amin(dim=(1, 2, 3), keepdim=True), \
            imgs.amax(dim=(1, 2, 3), keepdim=True)
        imgs = (imgs - lo) / (hi - lo).clamp(min=1e-8)
    n, c, h, w = imgs.shape
    ncol = min(nrow, n)
    nrows = (n + ncol - 1) // ncol
    grid = torch.zeros(c, nrows * (h + 2) + 2, ncol * (w + 2) + 2)
    for idx in range(n):
        r, col = divmod(idx, ncol)
        grid[:, 2 + r * (h + 2):2 + r * (h + 2) + h,
             2 + col * (w + 2):2 + col * (w + 2) + w] = imgs[idx]
    return grid


def register_plots(loss, grapher, epoch, args, prefix="train"):
    if args.distributed_rank != 0 or grapher is None:
        return
    for k, v in loss.items():
        if isinstance(v, dict):
            register_plots(v, grapher, epoch, args, prefix=prefix)
        if "mean" in k or "scalar" in k:
            key_name = "-".join(k.split("_")[0:-1])
            value = v.item() if torch.is_tensor(v) else float(v)
            grapher.add_scalar(f"{prefix}_{key_name}", value, epoch)


def register_images(output_map, grapher, args, prefix="train"):
    if args.distributed_rank != 0 or grapher is None:
        return
    for k, v in output_map.items():
        if isinstance(v, dict):
            register_images(v, grapher, args, prefix=prefix)
        if "img" in k or "imgs" in k:
            key_name = "-".join(k.split("_")[0:-1])
            grapher.add_image(f"{prefix}_{key_name}",
                              make_grid(v, normalize=True), global_step=0)


# ---------------------------------------------------------------------------
# the engine loop
# ---------------------------------------------------------------------------

def _sum_scalars(d1: dict, d2: dict) -> dict:
    out = {}
    for k, v in d2.items():
        prev = d1.get(k, 0.0)
        vd = v.detach() if torch.is_tensor(v) else v
        out[k] = prev + vd
    return out


def execute_graph(epoch, model, loader, grapher, args, optimizer=None,
                  prefix="test"):
    start_time = time.time()
    is_eval = "train" not in prefix
    model.eval() if is_eval else model.train()
    assert optimizer is None if is_eval else optimizer is not None
    loss_map, num_samples = {}, 0
    num_minibatches = 0
    is_ddp = isinstance(model, FlatDDP)
    # --half: bf16 autocast by default (no loss scaler needed on MI355X);
    # --half-dtype fp16 adds a dynamic GradScaler for strict parity with
    # the reference's Apex O2 fp16 runs (/root/reference/main.py:614,746)
    use_fp16 = args.half and getattr(args, "half_dtype", "bf16") == "fp16"
    autocast_ctx = (torch.autocast(
        "cuda", dtype=torch.float16 if use_fp16 else torch.bfloat16)
        if args.half and args.cuda else _dummy_context())
    scaler = torch.amp.GradScaler(
        "cuda", enabled=use_fp16 and args.cuda and not is_eval)
    from ..profiling import PhaseTimer
    timer = PhaseTimer(enabled=getattr(args, "perf_stats", False),
                       use_cuda=args.cuda)

    from ..ops.classifier import cross_entropy_topk

    def train_step_body(a1, a2, lab):
        """One full training step (fwd + losses + bwd + opt) — runs eager
        or captured into a hipGraph (``--hip-graph``); returns the five
        scalar tensors the epoch aggregation reads."""
        with autocast_ctx:
            out = model(a1, a2)
            byol_loss = loss_function(
                online_prediction1=out["online_prediction1"].float(),
                online_prediction2=out["online_prediction2"].float(),
                target_projection1=out["target_projection1"].float(),
                target_projection2=out["target_projection2"].float())
            # fused CE + top-1/top-5 in one HIP kernel on GPU
            # (byol_amd/ops/classifier.py); composed oracle elsewhere
            classifier_loss, acc1, acc5 = cross_entropy_topk(
                out["linear_preds"].float().contiguous(),
                torch.cat([lab, lab], 0))
            total = byol_loss + classifier_loss
        optimizer.zero_grad()
        if scaler.is_enabled():
            scaler.scale(total).backward()
            if is_ddp:
                model.finish_grad_sync()
            scaler.unscale_(optimizer)
            if args.clip > 0:
                nn.utils.clip_grad_value_(model.parameters(), args.clip)
            scaler.step(optimizer)
            scaler.update()
        else:
            total.backward()
            if is_ddp:
                model.finish_grad_sync()
            if args.clip > 0:
                nn.utils.clip_grad_value_(model.parameters(), args.clip)
            optimizer.step()
        if args.polyak_ema > 0:
            layers.polyak_ema_parameters(model, args.polyak_ema)
        return total, byol_loss, classifier_loss, acc1, acc5

    # hipGraph path: single-GPU training without polyak (the graph wrapper
    # owns the EMA decay/LR device scalars; DDP+graph is a later
    # experiment).  First GRAPH_WARM minibatches run eager (MIOpen finds,
    # fused-LARS state), then the step is captured once and replayed.
    # (fp16 GradScaler does host-side inf checks each step — not
    # graph-capturable, so fp16 excludes the graph path)
    graph_wanted = (not is_eval and args.cuda
                    and getattr(args, "hip_graph", False)
                    and args.polyak_ema == 0 and not is_ddp
                    and not scaler.is_enabled())
    GRAPH_WARM = 2
    graphed = None

    batches = loader
    if args.cuda:
        # H2D on a dedicated copy stream, overlapped with compute
        # (SURVEY.md K19); pass-through for device-resident batches
        from .prefetch import DevicePrefetcher
        batches = DevicePrefetcher(loader)

    timer.start("data")
    for num_minibatches, (augmentation1, augmentation2, labels) in \
            enumerate(batches):
        if args.cuda and not augmentation1.is_cuda:
            augmentation1 = augmentation1.cuda(non_blocking=True)
            augmentation2 = augmentation2.cuda(non_blocking=True)
            labels = labels.cuda(non_blocking=True)
        timer.stop()

        if not is_eval:
            timer.start("step")
            stepped = False
            if graph_wanted and num_minibatches >= GRAPH_WARM:
                if graphed is None:
                    from .graph_step import GraphedTrainStep
                    static = (augmentation1.clone(), augmentation2.clone(),
                              labels.clone())
                    cand = GraphedTrainStep(model, optimizer,
                                            train_step_body, static,
                                            warmup_steps=0)
                    try:
                        cand.capture()
                        graphed = cand
                    except Exception as exc:  # noqa: BLE001 — eager works
                        graph_wanted = False
                        print(f"hip-graph capture failed ({exc}); "
                              "continuing eager")
                if graphed is not None and \
                        augmentation1.shape == graphed.static_inputs[0].shape:
                    vals = graphed.replay(augmentation1, augmentation2,
                                          labels)
                    stepped = True
            if not stepped:
                vals = train_step_body(augmentation1, augmentation2, labels)
            total, byol_loss, classifier_loss, acc1, acc5 = vals
            timer.stop()
        else:
            timer.start("forward")
            with torch.no_grad():
                with autocast_ctx:
                    if args.polyak_ema > 0:
                        output_dict = layers.get_polyak_prediction(
                            model, pred_fn=functools.partial(
                                model, augmentation1, augmentation2))
                    else:
                        output_dict = model(augmentation1, augmentation2)
                    byol_loss = loss_function(
                        online_prediction1=output_dict[
                            "online_prediction1"].float(),
                        online_prediction2=output_dict[
                            "online_prediction2"].float(),
                        target_projection1=output_dict[
                            "target_projection1"].float(),
                        target_projection2=output_dict[
                            "target_projection2"].float())
                    classifier_loss, acc1, acc5 = cross_entropy_topk(
                        output_dict["linear_preds"].float().contiguous(),
                        labels)
                    total = byol_loss + classifier_loss
            timer.stop()

        with torch.no_grad():
            loss_t = {
                "loss_mean": total,
                "byol_loss_mean": byol_loss,
                "linear_loss_mean": classifier_loss,
                "top1_mean": acc1,
                "top5_mean": acc5,
            }
            # clone: under --hip-graph these are the graph's static output
            # tensors, whose storage is rewritten by the next replay
            detached_t = {k: (v.detach().clone() if torch.is_tensor(v)
                              else v)
                          for k, v in loss_t.items()}
            loss_map = detached_t if not loss_map else \
                _sum_scalars(loss_map, detached_t)
            num_samples += augmentation1.size(0)
            timer.add_samples(augmentation1.size(0))

        if args.debug_step:
            break
        timer.start("data")
    timer.stop()

    loss_map = {k: v / (num_minibatches + 1) for k, v in loss_map.items()}

    to_log = ("{}-{}[Epoch {}][{} samples][{:.2f} sec]:\tLoss: {:.4f}\t"
              "Top-1: {:.4f}\tTop-5: {:.4f}")
    print(to_log.format(
        prefix, args.distributed_rank, epoch, num_samples,
        time.time() - start_time,
        float(loss_map["loss_mean"]),
        float(loss_map["top1_mean"]),
        float(loss_map["top5_mean"])))

    if getattr(args, "perf_stats", False):
        print(timer.epoch_summary())

    register_plots(dict(loss_map), grapher, epoch=epoch, args=args,
                   prefix=prefix)

    num_images_to_post = min(64, augmentation1.shape[0])
    image_size_to_post = min(64, augmentation1.shape[-1])
    image_map = {
        "augmentation1_imgs": F.interpolate(
            augmentation1[0:num_images_to_post].float(),
            size=(image_size_to_post, image_size_to_post)),
        "augmentation2_imgs": F.interpolate(
            augmentation2[0:num_images_to_post].float(),
            size=(image_size_to_post, image_size_to_post)),
    }
    register_images(image_map, grapher, args, prefix=prefix)
    if grapher is not None:
        grapher.save()

    loss_val = float(loss_map["loss_mean"])
    loss_map.clear()
    return loss_val


def train(epoch, model, optimizer, train_loader, grapher, args):
    return execute_graph(epoch, model, train_loader, grapher, args,
                         optimizer, prefix="train")


def test(epoch, model, test_loader, grapher, args):
    return execute_graph(epoch, model, test_loader, grapher, args,
                         prefix="test")


# ---------------------------------------------------------------------------
# epoch driver
# ---------------------------------------------------------------------------

def _rank_consensus_loss(test_loss, args):
    """Mean-reduce the per-rank test loss across the process group so every
    rank feeds the SAME value to ModelSaver and takes the same early-stop
    branch.  Without this, divergent BN running stats (buffers are NOT
    broadcast per step by default here, unlike torch-1.5 DDP) could make
    ranks disagree on stopping and hang the job at the next collective."""
    import torch.distributed as dist
    if args.num_replicas > 1 and dist.is_available() and dist.is_initialized():
        t = torch.tensor([test_loss], dtype=torch.float64,
                         device="cuda" if args.cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return float(t.item()) / dist.get_world_size()
    return test_loss


def run(rank, args):
    from ..parallel import init_multiprocessing_and_cuda
    init_multiprocessing_and_cuda(rank, args)
    loader, model, grapher = build_loader_model_grapher(args)
    print(pprint.PrettyPrinter(indent=4).pformat(vars(args)))
    optimizer, scheduler = build_optimizer(model, args)

    bundle = CheckpointBundle(model, optimizer, scheduler, args)
    saver = ModelSaver(bundle, early_stop=args.early_stop,
                       rank=args.distributed_rank,
                       burn_in_interval=int(0.1 * args.epochs),
                       larger_is_better=False, max_early_stop_steps=10)
    restore_dict = saver.restore()
    init_epoch = restore_dict["epoch"]

    for epoch in range(init_epoch, args.epochs + 1):
        train(epoch, model, optimizer, loader.train_loader, grapher, args)
        test_loss = test(epoch, model, loader.test_loader, grapher, args)
        test_loss = _rank_consensus_loss(test_loss, args)
        loader.set_all_epochs(epoch)

        scheduler.step()
        register_plots(
            {"learning_rate_scalar": optimizer.param_groups[0]["lr"]},
            grapher, epoch, args)

        if saver(test_loss):
            saver.restore()
            test_loss = test(epoch, model, loader.test_loader, grapher, args)
            break

        if epoch == 2 and args.distributed_rank == 0 and grapher is not None:
            from ..utils import get_slurm_id
            config_to_post = dict(vars(args))
            slurm_id = get_slurm_id()
            if slurm_id is not None:
                config_to_post["slurm_job_id"] = slurm_id
            grapher.add_text(
                "config",
                pprint.PrettyPrinter(indent=4).pformat(config_to_post), 0)

    if grapher is not None:
        grapher.close()
