"""Self-contained PIL ImageFolder + transforms (data/imagefolder.py):
the real-ImageNet path without torchvision.  Synthetic JPEG/PNG images
are written to a class-directory tree and run through the full train
and val pipelines."""

import os

import numpy as np
import pytest
import torch

PIL = pytest.importorskip("PIL")
from PIL import Image  # noqa: E402

from vit_10b_fsdp_example_amd.data.imagefolder import (  # noqa: E402
    CenterCrop, Compose, ImageFolder, Normalize, RandomHorizontalFlip,
    RandomResizedCrop, Resize, ToTensor,
)


def _write_tree(root, split, n_classes=3, per_class=2, size=(48, 40)):
    rng = np.random.default_rng(0)
    for c in range(n_classes):
        d = os.path.join(root, split, f"class_{c}")
        os.makedirs(d, exist_ok=True)
        for i in range(per_class):
            arr = rng.integers(0, 255, (size[1], size[0], 3), dtype=np.uint8)
            img = Image.fromarray(arr, "RGB")
            img.save(os.path.join(d, f"img_{i}.jpg" if i % 2 else
                                  f"img_{i}.png"))


def test_imagefolder_scan_and_labels(tmp_path):
    _write_tree(str(tmp_path), "train")
    ds = ImageFolder(os.path.join(str(tmp_path), "train"))
    assert len(ds) == 6
    assert ds.class_to_idx == {"class_0": 0, "class_1": 1, "class_2": 2}
    img, target = ds[0]
    assert target == 0 and img.size == (48, 40)
    assert max(t for _, t in (ds[i] for i in range(len(ds)))) == 2


def test_train_transform_pipeline(tmp_path):
    _write_tree(str(tmp_path), "train")
    tf = Compose([
        RandomResizedCrop(16), RandomHorizontalFlip(), ToTensor(),
        Normalize([0.485, 0.456, 0.406], [0.229, 0.224, 0.225]),
    ])
    ds = ImageFolder(os.path.join(str(tmp_path), "train"), tf)
    x, y = ds[3]
    assert x.shape == (3, 16, 16) and x.dtype == torch.float32
    assert 0 <= y < 3
    # normalized: not confined to [0,1]
    assert float(x.min()) < 0 or float(x.max()) > 1


def test_val_transform_matches_reference_geometry():
    # val: Resize(short->(S*256)//224) then CenterCrop(S); for S=16 the
    # resize short side is 18
    img = Image.fromarray(np.zeros((40, 48, 3), dtype=np.uint8), "RGB")
    resized = Resize(18)(img)
    assert min(resized.size) == 18
    assert resized.size == (int(round(18 * 48 / 40)), 18)
    cropped = CenterCrop(16)(resized)
    assert cropped.size == (16, 16)


def test_to_tensor_values_roundtrip():
    arr = np.arange(2 * 3 * 3, dtype=np.uint8).reshape(2, 3, 3)  # H,W,C
    img = Image.fromarray(arr, "RGB")
    t = ToTensor()(img)
    np.testing.assert_allclose(
        t.numpy(), arr.transpose(2, 0, 1).astype(np.float32) / 255.0
    )


def test_random_resized_crop_bounds():
    torch.manual_seed(0)
    rrc = RandomResizedCrop(8)
    img = Image.fromarray(np.zeros((11, 13, 3), dtype=np.uint8), "RGB")
    for _ in range(25):
        out = rrc(img)
        assert out.size == (8, 8)


def test_build_datasets_real_mode(tmp_path):
    _write_tree(str(tmp_path), "train")
    _write_tree(str(tmp_path), "val")
    from vit_10b_fsdp_example_amd.cli import parse_args
    from vit_10b_fsdp_example_amd.data import build_datasets

    cfg = parse_args([
        "--data_dir", str(tmp_path), "--image_size", "16", "--patch_size",
        "4", "--batch_size", "2", "--num_workers", "0",
    ])
    assert not cfg.fake_data
    train_ds, train_loader, sampler, _, val_loader, _ = build_datasets(
        cfg, torch.device("cpu")
    )
    assert len(train_ds) == 6
    sampler.set_epoch(0)
    x, y = next(iter(train_loader))
    assert x.shape == (2, 3, 16, 16)
    assert y.dtype == torch.long
    xv, _ = next(iter(val_loader))
    assert xv.shape == (2, 3, 16, 16)
