"""Dataset construction: real ImageFolder or synthetic ImageNet.

Capability parity with the reference's build_datasets
(run_vit_training.py:30-96) and FakeImageNetDataset (utils.py:46-55):
  * --fake_data swaps in a zero-image dataset with ImageNet-1k lengths
    (train 1,281,167 / val 50,000) — the de-facto integration-test and
    benchmarking mode,
  * real mode: our own PIL-based ImageFolder with the standard
    ImageNet train/val transforms (RandomResizedCrop+flip /
    Resize+CenterCrop, bicubic, ImageNet mean/std),
  * per-rank DistributedSampler with drop_last=True on both splits
    (shuffle only for train), local batch = global batch / world size.

The real-data path is self-contained (PIL + torch, imagefolder.py) —
torchvision is not installed in the target image.
"""

import os

import torch

from .. import dist as xdist
from .loader import DeviceLoader

IMAGENET_TRAIN_LEN = 1281167
IMAGENET_VAL_LEN = 50000
IMAGENET_MEAN = [0.485, 0.456, 0.406]
IMAGENET_STD = [0.229, 0.224, 0.225]


class FakeImageNetDataset(torch.utils.data.Dataset):
    """Synthetic stand-in for ImageNet: a zero image [3, S, S] with label 0
    (reference utils.py:46-55).  Zero tensors keep the host side free so
    the benchmark measures the device/comm path, not JPEG decode."""

    def __init__(self, image_size, length):
        self.image_size = image_size
        self.length = length

    def __getitem__(self, idx):
        return torch.zeros(3, self.image_size, self.image_size), 0

    def __len__(self):
        return self.length


def _imagefolder_datasets(cfg):
    # self-contained PIL+torch ImageFolder/transforms (imagefolder.py):
    # torchvision is not installed in the target image, and the
    # transforms replicate its semantics (reference
    # run_vit_training.py:40-55) directly
    from . import imagefolder as T

    train_transform = T.Compose(
        [
            T.RandomResizedCrop(cfg.image_size),
            T.RandomHorizontalFlip(),
            T.ToTensor(),
            T.Normalize(mean=IMAGENET_MEAN, std=IMAGENET_STD),
        ]
    )
    val_transform = T.Compose(
        [
            T.Resize((cfg.image_size * 256) // 224),
            T.CenterCrop(cfg.image_size),
            T.ToTensor(),
            T.Normalize(mean=IMAGENET_MEAN, std=IMAGENET_STD),
        ]
    )
    train_ds = T.ImageFolder(
        os.path.join(cfg.data_dir, "train"), train_transform
    )
    val_ds = T.ImageFolder(os.path.join(cfg.data_dir, "val"), val_transform)
    return train_ds, val_ds


def build_datasets(cfg, device, compute_dtype=None):
    world_size = xdist.get_world_size()
    rank = xdist.get_rank()

    assert cfg.batch_size % world_size == 0, (
        f"global batch size {cfg.batch_size} must divide by world size {world_size}"
    )
    local_batch_size = cfg.batch_size // world_size

    if cfg.fake_data:
        xdist.master_print("loading fake images")
        # VITFSDP_FAKE_LEN overrides both synthetic split lengths
        # (smoke tests / CI; 0 = the real ImageNet-1k lengths)
        fake_len = int(os.environ.get("VITFSDP_FAKE_LEN", "0"))
        train_dataset = FakeImageNetDataset(
            cfg.image_size, fake_len or IMAGENET_TRAIN_LEN
        )
        val_dataset = FakeImageNetDataset(
            cfg.image_size, fake_len or IMAGENET_VAL_LEN
        )
    else:
        xdist.master_print(f"loading images from directory: {cfg.data_dir}")
        train_dataset, val_dataset = _imagefolder_datasets(cfg)

    def _make(dataset, shuffle):
        sampler = torch.utils.data.distributed.DistributedSampler(
            dataset,
            num_replicas=world_size,
            rank=rank,
            drop_last=True,
            shuffle=shuffle,
        )
        loader = torch.utils.data.DataLoader(
            dataset,
            batch_size=local_batch_size,
            sampler=sampler,
            drop_last=True,
            num_workers=cfg.num_workers,
            pin_memory=torch.cuda.is_available(),
            persistent_workers=cfg.num_workers > 0,
        )
        return sampler, DeviceLoader(loader, device, compute_dtype=compute_dtype)

    train_sampler, train_loader = _make(train_dataset, shuffle=True)
    val_sampler, val_loader = _make(val_dataset, shuffle=False)
    return (
        train_dataset,
        train_loader,
        train_sampler,
        val_dataset,
        val_loader,
        val_sampler,
    )
