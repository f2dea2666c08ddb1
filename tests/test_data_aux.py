"""Folder loading (PIL), GPU-augment parameter sampling, blur oracle,
saver/early-stop semantics, CLI entry."""

import os
import subprocess
import sys

import numpy as np
import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _make_imagefolder(root, classes=("cat", "dog"), per_class=3, size=40):
    from PIL import Image
    for split in ("train", "test"):
        for c in classes:
            d = os.path.join(root, split, c)
            os.makedirs(d, exist_ok=True)
            for i in range(per_class):
                arr = (np.random.RandomState(hash((split, c, i)) % 2**31)
                       .rand(size, size, 3) * 255).astype(np.uint8)
                Image.fromarray(arr).save(os.path.join(d, f"{i}.png"))


def test_image_folder_task(tmp_path):
    from byol_amd.data.loader import get_loader
    from byol_amd.data.transforms import Resize
    _make_imagefolder(tmp_path)
    loader = get_loader(
        train_transform=[Resize((16, 16))],
        test_transform=[Resize((16, 16))],
        task="multi_augment_image_folder", batch_size=2,
        data_dir=str(tmp_path), num_replicas=1, distributed_rank=0,
        workers_per_replica=0, seed=1, image_size_override=16, cuda=False)
    assert loader.output_size == 2
    assert loader.num_train_samples == 6
    a1, a2, labels = next(iter(loader.train_loader))
    assert a1.shape == (2, 3, 16, 16)
    assert a1.min() >= 0 and a1.max() <= 1


def test_sample_params_distributions():
    from byol_amd.data.gpu_augment import sample_params
    rng = np.random.RandomState(0)
    crop, cparam, sigma = sample_params(rng, 500, 256, 256, 224,
                                        jitter_strength=1.0,
                                        dali_mode=False)
    # crops inside bounds
    assert (crop[:, 0] >= 0).all() and (crop[:, 1] >= 0).all()
    assert ((crop[:, 0] + crop[:, 2]) <= 256).all()
    assert ((crop[:, 1] + crop[:, 3]) <= 256).all()
    # flip rate ~0.5, jitter ~0.8, gray ~0.2, blur ~0.5
    assert 0.4 < crop[:, 4].mean() < 0.6
    assert 0.7 < cparam[:, 0].mean() < 0.9
    assert 0.1 < cparam[:, 5].mean() < 0.3
    assert 0.4 < (sigma > 0).mean() < 0.6
    # dali mode: flip 0.2, no blur, saturation range 0.2s
    crop_d, cparam_d, sigma_d = sample_params(rng, 500, 256, 256, 224, 1.0,
                                              dali_mode=True)
    assert 0.12 < crop_d[:, 4].mean() < 0.28
    assert (sigma_d == 0).all()
    assert cparam_d[:, 3].min() >= 0.8 - 1e-6  # 1 - 0.2*s


def test_batched_blur_matches_single():
    from byol_amd.data.gpu_augment import _gaussian_blur_batched
    from byol_amd.data.transforms import GaussianBlur
    torch.manual_seed(0)
    img = torch.rand(3, 3, 20, 20)
    sigma = torch.tensor([1.3, 0.0, 0.6])
    out = _gaussian_blur_batched(img, sigma, kernel_size=5)
    # sigma=0 row is identity
    assert torch.allclose(out[1], img[1], atol=1e-6)
    # nonzero rows match the CPU transform's conv (same kernel math)
    gb = GaussianBlur(kernel_size=5, p=1.0)
    import random
    for i, s in ((0, 1.3), (2, 0.6)):
        random.seed(0)
        gb.sigma = (s, s)
        want = gb(img[i])
        assert torch.allclose(out[i], want, atol=1e-5), \
            (out[i] - want).abs().max()


def test_model_saver_early_stop(tmp_path):
    from byol_amd.engine.saver import CheckpointBundle, ModelSaver
    import torch.nn as nn

    class Args:
        model_dir = str(tmp_path)
        uid = "t"
        arch = "x"
        batch_size = 1
        num_replicas = 1
        optimizer = "o"

    model = nn.Linear(2, 2)
    bundle = CheckpointBundle(model, None, None, Args())
    saver = ModelSaver(bundle, early_stop=True, rank=0, burn_in_interval=2,
                       larger_is_better=False, max_early_stop_steps=3)
    # burn-in: no saving, no stopping
    assert saver(1.0) is False and saver(0.9) is False
    assert not list(tmp_path.glob("*.pt"))
    # improvement saves
    assert saver(0.5) is False
    assert list(tmp_path.glob("*.pt"))
    # 3 non-improvements -> stop
    assert saver(0.6) is False
    assert saver(0.7) is False
    assert saver(0.8) is True


def test_main_cli_end_to_end(tmp_path):
    cmd = [sys.executable, os.path.join(ROOT, "main.py"),
           "--task", "synthetic_multi_augment", "--arch", "resnet18",
           "--representation-size", "512", "--projection-size", "8",
           "--head-latent-size", "16", "--image-size-override", "32",
           "--batch-size", "4", "--epochs", "1", "--warmup", "0",
           "--num-replicas", "1", "--workers-per-replica", "0",
           "--no-cuda", "--debug-step", "--seed", "1",
           "--synthetic-classes", "3", "--synthetic-train-samples", "8",
           "--synthetic-test-samples", "4",
           "--log-dir", str(tmp_path / "runs"),
           "--model-dir", str(tmp_path / "models"),
           "--data-dir", str(tmp_path / "data")]
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=420,
                         cwd=ROOT)
    assert res.returncode == 0, res.stderr[-2000:]
    assert "train-0[Epoch 1]" in res.stdout


def test_cifar10_task(tmp_path):
    import pickle
    from byol_amd.data.loader import get_loader
    from byol_amd.data.transforms import Resize
    d = tmp_path / "cifar-10-batches-py"
    d.mkdir()
    rng = np.random.RandomState(0)
    for name, n in [("data_batch_1", 20), ("test_batch", 10)]:
        payload = {b"data": rng.randint(0, 256, (n, 3072), dtype=np.uint8),
                   b"labels": rng.randint(0, 10, n).tolist()}
        with open(d / name, "wb") as f:
            pickle.dump(payload, f)
    loader = get_loader(
        train_transform=[Resize((16, 16))], test_transform=[Resize((16, 16))],
        task="multi_augment_cifar10", batch_size=4, data_dir=str(tmp_path),
        num_replicas=1, distributed_rank=0, workers_per_replica=0, seed=0,
        image_size_override=16, cuda=False)
    assert loader.output_size == 10
    assert loader.num_train_samples == 20
    a1, a2, labels = next(iter(loader.train_loader))
    assert a1.shape == (4, 3, 16, 16)
    assert a1.min() >= 0 and a1.max() <= 1


def test_mnist_task(tmp_path):
    from byol_amd.data.loader import get_loader
    from byol_amd.data.transforms import Resize
    rng = np.random.RandomState(1)

    def write_idx(prefix, n):
        imgs = rng.randint(0, 256, (n, 28, 28), dtype=np.uint8)
        labs = rng.randint(0, 10, n, dtype=np.uint8)
        with open(tmp_path / f"{prefix}-images-idx3-ubyte", "wb") as f:
            f.write((2051).to_bytes(4, "big") + n.to_bytes(4, "big")
                    + (28).to_bytes(4, "big") + (28).to_bytes(4, "big")
                    + imgs.tobytes())
        with open(tmp_path / f"{prefix}-labels-idx1-ubyte", "wb") as f:
            f.write((2049).to_bytes(4, "big") + n.to_bytes(4, "big")
                    + labs.tobytes())

    write_idx("train", 16)
    write_idx("t10k", 8)
    loader = get_loader(
        train_transform=[Resize((16, 16))], test_transform=[Resize((16, 16))],
        task="multi_augment_mnist", batch_size=4, data_dir=str(tmp_path),
        num_replicas=1, distributed_rank=0, workers_per_replica=0, seed=0,
        image_size_override=16, cuda=False)
    assert loader.output_size == 10
    a1, a2, labels = next(iter(loader.train_loader))
    assert a1.shape == (4, 3, 16, 16)  # mnist expanded to 3 channels


def test_saver_restore_reseeds_best(tmp_path):
    """After resume, the pre-restart best loss must survive: a worse
    post-restart epoch may NOT overwrite the best checkpoint."""
    import types

    import torch.nn as nn

    from byol_amd.engine.saver import CheckpointBundle, ModelSaver

    args = types.SimpleNamespace(model_dir=str(tmp_path), uid="t",
                                 arch="a", batch_size=1, num_replicas=1,
                                 optimizer="sgd")
    model = nn.Linear(2, 2)
    bundle = CheckpointBundle(model, None, None, args)
    saver = ModelSaver(bundle, early_stop=False, rank=0, burn_in_interval=0)
    assert saver(1.0) is False  # epoch 1: best=1.0, checkpoint written
    assert saver(2.0) is False  # worse: stale

    # simulate restart
    saver2 = ModelSaver(bundle, early_stop=False, rank=0,
                        burn_in_interval=0)
    state = saver2.restore()
    assert state["best_loss"] == 1.0
    assert saver2.best == 1.0
    # a worse epoch after resume must NOT count as improved
    before = bundle.path()
    import os
    mtime = os.path.getmtime(before)
    assert saver2(1.5) is False
    assert saver2.best == 1.0
    assert os.path.getmtime(before) == mtime  # best checkpoint untouched
