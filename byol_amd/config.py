"""CLI with exact flag parity with the reference
(``/root/reference/main.py:35-119``), plus a few MI355X-native knobs
(bucket size, buffer-broadcast parity, synthetic-data sizing)."""

import argparse

from .models.resnet import arch_names

__all__ = ["build_parser", "parse_args"]


def build_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(description="BYOL MI355X")

    # Task parameters
    parser.add_argument("--task", type=str,
                        default="multi_augment_image_folder",
                        help="task to work on (default: "
                             "multi_augment_image_folder; also: "
                             "synthetic_multi_augment, "
                             "dali_multi_augment_image_folder)")
    parser.add_argument("--batch-size", type=int, default=4096, metavar="N",
                        help="input batch size for training (default: 4096)")
    parser.add_argument("--epochs", type=int, default=3000, metavar="N",
                        help="minimum number of epochs to train "
                             "(default: 3000)")
    parser.add_argument("--download", type=int, default=1,
                        help="download simple datasets (default: 1; no-op "
                             "in this offline environment)")
    parser.add_argument("--image-size-override", type=int, default=224,
                        help="force resizing of images to this size "
                             "(default: 224)")
    parser.add_argument("--data-dir", type=str, default="./.datasets",
                        metavar="DD",
                        help="directory which contains input data")
    parser.add_argument("--log-dir", type=str, default="./runs",
                        help="directory to store logs to (default: ./runs)")
    parser.add_argument("--uid", type=str, default="",
                        help="uid for current session (default: empty-str)")

    # Model related
    parser.add_argument("-a", "--arch", metavar="ARCH", default="resnet50",
                        choices=arch_names(),
                        help="model architecture: "
                             + " | ".join(arch_names())
                             + " (default: resnet50)")
    parser.add_argument("--representation-size", type=int, default=2048,
                        help="size of the representation (default: 2048)")
    parser.add_argument("--projection-size", type=int, default=256,
                        help="output size for projection head "
                             "(default: 256)")
    parser.add_argument("--head-latent-size", type=int, default=4096,
                        help="hidden size for the MLP heads (default: 4096)")
    parser.add_argument("--base-decay", type=float, default=0.996,
                        help="decay for target network (default: 0.996)")
    parser.add_argument("--weight-initialization", type=str, default=None,
                        help="weight init type; None uses default pytorch "
                             "init (default: None)")
    parser.add_argument("--model-dir", type=str, default=".models",
                        help="directory for saved models "
                             "(default: .models)")

    # Regularizer
    parser.add_argument("--color-jitter-strength", type=float, default=1.0,
                        help="scalar weighting for color jitter "
                             "(default: 1.0)")
    parser.add_argument("--weight-decay", type=float, default=1e-6,
                        help="weight decay (default: 1e-6)")
    parser.add_argument("--polyak-ema", type=float, default=0,
                        help="Polyak weight averaging coef (default: 0)")
    parser.add_argument("--convert-to-sync-bn", action="store_true",
                        default=False,
                        help="converts all BNs to SyncBNs (default: False)")

    # Optimization related
    parser.add_argument("--clip", type=float, default=0,
                        help="gradient clipping value (default: 0)")
    parser.add_argument("--lr", type=float, default=0.2, metavar="LR",
                        help="learning rate (default: 0.2)")
    parser.add_argument("--lr-update-schedule", type=str, default="cosine",
                        help="lr schedule fixed/cosine (default: cosine)")
    parser.add_argument("--warmup", type=int, default=10,
                        help="warmup epochs (default: 10)")
    parser.add_argument("--optimizer", type=str, default="lars_momentum",
                        help="optimizer (default: lars_momentum)")
    parser.add_argument("--early-stop", action="store_true", default=False,
                        help="enable early stopping (default: False)")

    # Visdom parameters (parity; falls back to jsonl/tensorboard)
    parser.add_argument("--visdom-url", type=str, default=None,
                        help="visdom URL for graphs (default: None)")
    parser.add_argument("--visdom-port", type=int, default=None,
                        help="visdom port for graphs (default: None)")

    # Device / debug
    parser.add_argument("--num-replicas", type=int, default=8,
                        help="number of compute devices (default: 8)")
    parser.add_argument("--workers-per-replica", type=int, default=2,
                        help="dataloader workers per replica (default: 2)")
    parser.add_argument("--distributed-master", type=str, default=None,
                        help="hostname/IP for distributed master")
    parser.add_argument("--distributed-rank", type=int, default=0,
                        help="rank of this replica (default: 0)")
    parser.add_argument("--distributed-port", type=int, default=29300,
                        help="distributed port (default: 29300)")
    parser.add_argument("--debug-step", action="store_true", default=False,
                        help="single minibatch per execute_graph call")
    parser.add_argument("--seed", type=int, default=None,
                        help="seed for numpy and pytorch (default: None)")
    parser.add_argument("--no-cuda", action="store_true", default=False,
                        help="disables CUDA training")
    parser.add_argument("--half", action="store_true", default=False,
                        help="half precision training (bf16 autocast on "
                             "MI355X; the reference used Apex fp16 O2)")
    parser.add_argument("--half-dtype", type=str, default="bf16",
                        choices=["bf16", "fp16"],
                        help="--half compute dtype: bf16 (MI355X-native, no "
                             "loss scaler needed) or fp16 + dynamic "
                             "GradScaler for strict parity with the "
                             "reference's Apex O2 fp16 experiments")

    # MI355X-native knobs (not in the reference)
    parser.add_argument("--bucket-cap-mb", type=float, default=32.0,
                        help="DDP gradient bucket size in MiB, tuned for "
                             "xGMI (default: 32)")
    parser.add_argument("--broadcast-buffers", action="store_true",
                        default=False,
                        help="parity flag: per-step buffer broadcast like "
                             "torch-1.5 DDP (default off: the EMA mean is "
                             "deterministic+replicated)")
    parser.add_argument("--synthetic-classes", type=int, default=1000,
                        help="synthetic task: number of classes")
    parser.add_argument("--synthetic-train-samples", type=int, default=None,
                        help="synthetic task: train set size")
    parser.add_argument("--synthetic-test-samples", type=int, default=None,
                        help="synthetic task: test set size")
    parser.add_argument("--channels-last", action="store_true", default=False,
                        help="NHWC memory format for conv throughput")
    parser.add_argument("--perf-stats", action="store_true", default=False,
                        help="per-phase HIP-event timing summary per epoch")
    parser.add_argument("--conv-dispatch", type=str, default=None,
                        choices=["auto", "autotune", "force", "off"],
                        help="MFMA conv dispatch: auto = measured per-shape "
                             "tables (default), autotune = runtime "
                             "ours-vs-MIOpen selection per new shape "
                             "(cudnn.benchmark-style; cache via "
                             "BYOL_AUTOTUNE_CACHE), force/off = always/"
                             "never our kernels.  Sets BYOL_MFMA_CONV1X1/"
                             "CONV3X3 for this process")
    parser.add_argument("--hip-graph", action="store_true", default=False,
                        help="capture the training step as one hipGraph and "
                             "replay it (single-GPU, no --polyak-ema; "
                             "removes ~2k kernel-launch boundaries per "
                             "step of host gap time)")
    return parser


def parse_args(argv=None):
    args = build_parser().parse_args(argv)
    if args.synthetic_train_samples is None:
        del args.__dict__["synthetic_train_samples"]
    if getattr(args, "synthetic_test_samples", 1) is None:
        del args.__dict__["synthetic_test_samples"]
    if args.conv_dispatch is not None:
        import os
        mode = {"auto": "auto", "autotune": "autotune", "force": "1",
                "off": "0"}[args.conv_dispatch]
        os.environ["BYOL_MFMA_CONV1X1"] = mode
        # 3x3 default is the (empty) auto table; force/off/autotune map 1:1
        os.environ["BYOL_MFMA_CONV3X3"] = mode
    return args
