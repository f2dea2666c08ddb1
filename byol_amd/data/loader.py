"""Data loading with the reference's ``datasets.loader.get_loader`` contract
(call site ``/root/reference/main.py:414-435``):

``get_loader(**{train_transform, test_transform, **vars(args)})`` returns an
object exposing ``input_shape`` (CHW), ``output_size`` (#classes),
``num_train_samples``, ``num_test_samples``, ``num_valid_samples``,
``train_loader``, ``test_loader`` (iterables of ``(aug1, aug2, labels)``),
and ``set_all_epochs(epoch)``.  Train data is sharded DistributedSampler-
style across replicas; test data is NOT (``/root/reference/main.py:422``).

Tasks:
* ``multi_augment_image_folder`` — ImageFolder with ``train/`` and ``test/``
  subdirs (PIL decode, tensor-native transforms in dataloader workers);
* ``dali_multi_augment_image_folder`` — same folder layout, the reference's
  DALI augmentation recipe (the transform list differs; built upstream);
* ``synthetic_multi_augment`` (also: task containing "synthetic") — in-memory
  random images, same two-view pipeline; what the benchmarks and tests use
  (no network, no datasets on the box).
"""

import os
from typing import List, Optional

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler

from .transforms import Compose, TwoViewTransform

__all__ = ["get_loader"]


class _TwoViewDataset(Dataset):
    """Wraps an (image_tensor, label) dataset into two-view samples."""

    def __init__(self, base: Dataset, train_transform, test_transform,
                 train: bool):
        self.base = base
        self.train = train
        self.train_transform = Compose(train_transform)
        self.test_transform = Compose(test_transform)

    def __len__(self):
        return len(self.base)

    def __getitem__(self, idx):
        img, label = self.base[idx]
        if self.train:
            return (self.train_transform(img), self.train_transform(img),
                    label)
        # test: resized image twice (the engine consumes (aug1, aug2, label)
        # triples in eval as well — /root/reference/main.py:579)
        out = self.test_transform(img)
        return out, out, label


class SyntheticImages(Dataset):
    """Deterministic synthetic images in [0, 1] (seeded per index)."""

    def __init__(self, num_samples: int, image_size: int, num_classes: int,
                 base_seed: int = 0, channels: int = 3):
        self.num_samples = num_samples
        self.image_size = image_size
        self.num_classes = num_classes
        self.base_seed = base_seed
        self.channels = channels

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.base_seed * 1000003 + idx)
        img = torch.rand((self.channels, self.image_size, self.image_size),
                         generator=g)
        label = int(torch.randint(self.num_classes, (1,), generator=g))
        return img, label


class FolderImages(Dataset):
    """ImageFolder (class-per-subdir) decoding to float CHW in [0,1]."""

    IMG_EXTS = {".jpg", ".jpeg", ".png", ".bmp", ".ppm", ".webp", ".npy"}

    def __init__(self, root: str):
        self.samples: List = []
        classes = sorted(d for d in os.listdir(root)
                         if os.path.isdir(os.path.join(root, d)))
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        for c in classes:
            cdir = os.path.join(root, c)
            for fn in sorted(os.listdir(cdir)):
                if os.path.splitext(fn)[1].lower() in self.IMG_EXTS:
                    self.samples.append((os.path.join(cdir, fn),
                                         self.class_to_idx[c]))
        if not self.samples:
            raise FileNotFoundError(f"no images under {root}")
        self.num_classes = len(classes)

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        path, label = self.samples[idx]
        if path.endswith(".npy"):
            arr = np.load(path)
            img = torch.from_numpy(arr).float()
            if img.dim() == 3 and img.shape[-1] in (1, 3):
                img = img.permute(2, 0, 1)
            if img.max() > 1.5:
                img = img / 255.0
        else:
            from PIL import Image
            with Image.open(path) as im:
                im = im.convert("RGB")
                img = torch.from_numpy(
                    np.asarray(im, dtype=np.uint8).copy()).permute(2, 0, 1)
                img = img.float().div_(255.0)
        return img, label


class Cifar10Files(Dataset):
    """CIFAR-10 from the standard local archives (python-pickle batches
    ``data_batch_*``/``test_batch`` under ``cifar-10-batches-py``, or the
    binary ``*.bin`` under ``cifar-10-batches-bin``).  The reference's
    datasets submodule handled 'simple datasets like MNIST/CIFAR10'
    (``--download`` flag help, /root/reference/main.py:43-44); there is no
    network here, so files must already be on disk."""

    def __init__(self, root: str, train: bool):
        import glob
        import pickle
        py_dir = os.path.join(root, "cifar-10-batches-py")
        bin_dir = os.path.join(root, "cifar-10-batches-bin")
        imgs, labels = [], []
        if os.path.isdir(py_dir):
            names = ([f"data_batch_{i}" for i in range(1, 6)] if train
                     else ["test_batch"])
            for nm in names:
                path = os.path.join(py_dir, nm)
                if not os.path.isfile(path):
                    continue
                with open(path, "rb") as f:
                    d = pickle.load(f, encoding="bytes")
                data = d.get(b"data", d.get("data"))
                labs = d.get(b"labels", d.get("labels"))
                imgs.append(np.asarray(data, dtype=np.uint8))
                labels.extend(labs)
        elif os.path.isdir(bin_dir):
            names = ([f"data_batch_{i}.bin" for i in range(1, 6)] if train
                     else ["test_batch.bin"])
            for nm in names:
                path = os.path.join(bin_dir, nm)
                if not os.path.isfile(path):
                    continue
                raw = np.fromfile(path, dtype=np.uint8).reshape(-1, 3073)
                labels.extend(raw[:, 0].tolist())
                imgs.append(raw[:, 1:])
        if not imgs:
            raise FileNotFoundError(
                f"no CIFAR-10 batches under {py_dir} or {bin_dir}")
        self.data = np.concatenate(imgs).reshape(-1, 3, 32, 32)
        self.labels = labels
        self.num_classes = 10

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, idx):
        img = torch.from_numpy(self.data[idx].copy()).float().div_(255.0)
        return img, int(self.labels[idx])


class MnistFiles(Dataset):
    """MNIST idx-format files (``train-images-idx3-ubyte`` etc., optionally
    ``.gz``) from ``data_dir`` (or its ``MNIST/raw`` subdir)."""

    def __init__(self, root: str, train: bool):
        import gzip
        prefix = "train" if train else "t10k"
        dirs = [root, os.path.join(root, "MNIST", "raw")]
        img_path = lab_path = None
        for d in dirs:
            for ext in ("", ".gz"):
                ip = os.path.join(d, f"{prefix}-images-idx3-ubyte{ext}")
                lp = os.path.join(d, f"{prefix}-labels-idx1-ubyte{ext}")
                if os.path.isfile(ip) and os.path.isfile(lp):
                    img_path, lab_path = ip, lp
        if img_path is None:
            raise FileNotFoundError(f"no MNIST idx files under {dirs}")

        def read(path):
            op = gzip.open if path.endswith(".gz") else open
            with op(path, "rb") as f:
                return f.read()

        ib = read(img_path)
        n = int.from_bytes(ib[4:8], "big")
        h = int.from_bytes(ib[8:12], "big")
        w = int.from_bytes(ib[12:16], "big")
        self.data = np.frombuffer(ib, dtype=np.uint8,
                                  offset=16).reshape(n, h, w)
        lb = read(lab_path)
        self.labels = np.frombuffer(lb, dtype=np.uint8, offset=8)
        self.num_classes = 10

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, idx):
        img = torch.from_numpy(self.data[idx].copy()).float().div_(255.0)
        img = img.unsqueeze(0).repeat(3, 1, 1)  # 3-channel for the encoder
        return img, int(self.labels[idx])


class _SourceDataset(Dataset):
    """Raw (image, label) pairs resized to a fixed source size — what the
    GPU augmentation pipeline consumes (decode/resize on CPU workers, all
    augmentation on device, like the reference's DALI path)."""

    def __init__(self, base: Dataset, src_size: int):
        from .transforms import Resize
        self.base = base
        self.resize = Resize((src_size, src_size))

    def __len__(self):
        return len(self.base)

    def __getitem__(self, idx):
        img, label = self.base[idx]
        return self.resize(img), label


class _GPUAugmentIterable:
    """Wraps a raw-batch loader; yields (aug1, aug2, labels) on device."""

    def __init__(self, raw_loader, out_size: int, jitter_strength: float,
                 dali_mode: bool, seed: int):
        from .gpu_augment import GPUTwoViewAugment
        self.raw_loader = raw_loader
        self.pipe = GPUTwoViewAugment(out_size,
                                      jitter_strength=jitter_strength,
                                      dali_mode=dali_mode, seed=seed)

    def __len__(self):
        return len(self.raw_loader)

    def __iter__(self):
        for imgs, labels in self.raw_loader:
            imgs = imgs.cuda(non_blocking=True)
            labels = labels.cuda(non_blocking=True)
            aug1, aug2 = self.pipe(imgs)
            yield aug1, aug2, labels


class Loader:
    def __init__(self, train_dataset: Dataset, test_dataset: Dataset,
                 input_shape, output_size: int, batch_size: int,
                 num_replicas: int, rank: int, workers: int, seed: int,
                 pin_memory: bool):
        self.input_shape = tuple(input_shape)
        self.output_size = output_size
        self.num_train_samples = len(train_dataset)
        self.num_test_samples = len(test_dataset)
        self.num_valid_samples = 0
        self.train_sampler: Optional[DistributedSampler] = None
        if num_replicas > 1:
            self.train_sampler = DistributedSampler(
                train_dataset, num_replicas=num_replicas, rank=rank,
                shuffle=True, seed=seed or 0, drop_last=True)
        self.train_loader = DataLoader(
            train_dataset, batch_size=batch_size,
            shuffle=(self.train_sampler is None),
            sampler=self.train_sampler, num_workers=workers,
            pin_memory=pin_memory, drop_last=True,
            persistent_workers=workers > 0)
        self.test_loader = DataLoader(
            test_dataset, batch_size=batch_size, shuffle=False,
            num_workers=workers, pin_memory=pin_memory,
            persistent_workers=workers > 0)

    def set_all_epochs(self, epoch: int):
        if self.train_sampler is not None:
            self.train_sampler.set_epoch(epoch)


def get_loader(train_transform=None, test_transform=None, **kwargs):
    task = kwargs["task"]
    batch_size = kwargs["batch_size"]
    data_dir = kwargs.get("data_dir", "./.datasets")
    num_replicas = kwargs.get("num_replicas", 1)
    rank = kwargs.get("distributed_rank", 0)
    workers = kwargs.get("workers_per_replica", 2)
    seed = kwargs.get("seed", None) or 0
    image_size = kwargs.get("image_size_override", 224)
    cuda = kwargs.get("cuda", False)

    if "synthetic" in task:
        num_classes = kwargs.get("synthetic_classes", 1000)
        num_train = kwargs.get("synthetic_train_samples",
                               max(batch_size * 4, 64))
        num_test = kwargs.get("synthetic_test_samples",
                              max(batch_size * 2, 32))
        train_base = SyntheticImages(num_train, image_size, num_classes,
                                     base_seed=seed)
        test_base = SyntheticImages(num_test, image_size, num_classes,
                                    base_seed=seed + 7)
        output_size = num_classes
    elif "image_folder" in task:
        train_base = FolderImages(os.path.join(data_dir, "train"))
        test_base = FolderImages(os.path.join(data_dir, "test"))
        output_size = train_base.num_classes
    elif "cifar10" in task:
        train_base = Cifar10Files(data_dir, train=True)
        test_base = Cifar10Files(data_dir, train=False)
        output_size = 10
    elif "mnist" in task:
        train_base = MnistFiles(data_dir, train=True)
        test_base = MnistFiles(data_dir, train=False)
        output_size = 10
    else:
        raise ValueError(f"unknown task {task!r}")

    gpu_augment = "dali" in task and cuda
    if gpu_augment:
        # DALI-parity path: CPU workers only decode+resize; crop/flip/
        # jitter/gray run as HIP kernels on device (byol_amd/ops/csrc/
        # augment.hip); blur on the torchvision-recipe tasks only.
        src_size = max(image_size + image_size // 8, image_size + 8)
        train_ds = _SourceDataset(train_base, src_size)
    else:
        train_ds = _TwoViewDataset(train_base, train_transform,
                                   test_transform, train=True)
    test_ds = _TwoViewDataset(test_base, train_transform, test_transform,
                              train=False)
    loader = Loader(train_ds, test_ds,
                    input_shape=(3, image_size, image_size),
                    output_size=output_size, batch_size=batch_size,
                    num_replicas=num_replicas, rank=rank, workers=workers,
                    seed=seed, pin_memory=cuda)
    if gpu_augment:
        loader.train_loader = _GPUAugmentIterable(
            loader.train_loader, image_size,
            jitter_strength=kwargs.get("color_jitter_strength", 1.0),
            dali_mode=True, seed=seed)
    return loader
