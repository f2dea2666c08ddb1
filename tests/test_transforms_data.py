import torch

from byol_amd.data.loader import get_loader
from byol_amd.data.transforms import (ColorJitter, Compose, GaussianBlur,
                                      RandomGrayscale, RandomHorizontalFlip,
                                      RandomResizedCrop, Resize,
                                      build_train_and_test_transforms)


class Args:
    task = "synthetic_multi_augment"
    image_size_override = 32
    color_jitter_strength = 1.0


def test_train_transform_keeps_unit_range_and_shape():
    args = Args()
    train_t, test_t = build_train_and_test_transforms(args)
    pipe = Compose(train_t)
    torch.manual_seed(0)
    for _ in range(8):
        img = torch.rand(3, 48, 40)
        out = pipe(img)
        assert out.shape == (3, 32, 32)
        assert out.min() >= 0.0 and out.max() <= 1.0


def test_dali_variant_has_no_blur():
    class DaliArgs(Args):
        task = "dali_multi_augment_image_folder"
    train_t, _ = build_train_and_test_transforms(DaliArgs())
    assert not any(isinstance(t, GaussianBlur) for t in train_t)


def test_resize_and_crop_shapes():
    img = torch.rand(3, 57, 91)
    assert Resize((32, 32))(img).shape == (3, 32, 32)
    assert RandomResizedCrop((24, 24))(img).shape == (3, 24, 24)


def test_gaussian_blur_preserves_constant_image():
    img = torch.full((3, 16, 16), 0.5)
    out = GaussianBlur(kernel_size=3, p=1.0)(img)
    assert torch.allclose(out, img, atol=1e-6)


def test_grayscale_channels_equal():
    g = RandomGrayscale(p=1.0)(torch.rand(3, 8, 8))
    assert torch.allclose(g[0], g[1]) and torch.allclose(g[1], g[2])


def test_color_jitter_range():
    jit = ColorJitter(0.8, 0.8, 0.8, 0.2)
    torch.manual_seed(1)
    for _ in range(5):
        out = jit(torch.rand(3, 8, 8))
        assert out.min() >= 0.0 and out.max() <= 1.0


def test_loader_contract():
    loader = get_loader(
        train_transform=[Resize((16, 16))], test_transform=[Resize((16, 16))],
        task="synthetic_multi_augment", batch_size=4, data_dir="/tmp",
        num_replicas=1, distributed_rank=0, workers_per_replica=0,
        seed=3, image_size_override=16, cuda=False, synthetic_classes=5,
        synthetic_train_samples=12, synthetic_test_samples=8)
    assert loader.input_shape == (3, 16, 16)
    assert loader.output_size == 5
    assert loader.num_train_samples == 12
    assert loader.num_test_samples == 8
    aug1, aug2, labels = next(iter(loader.train_loader))
    assert aug1.shape == (4, 3, 16, 16)
    assert aug2.shape == (4, 3, 16, 16)
    assert labels.shape == (4,)
    assert labels.dtype == torch.int64
    loader.set_all_epochs(1)  # no-op single replica


def test_two_view_samples_differ():
    from byol_amd.data.transforms import RandomApply
    loader = get_loader(
        train_transform=[RandomResizedCrop((16, 16))],
        test_transform=[Resize((16, 16))],
        task="synthetic_multi_augment", batch_size=2, data_dir="/tmp",
        num_replicas=1, distributed_rank=0, workers_per_replica=0,
        seed=3, image_size_override=16, cuda=False,
        synthetic_train_samples=4, synthetic_test_samples=4)
    aug1, aug2, _ = next(iter(loader.train_loader))
    assert not torch.equal(aug1, aug2)
