"""Data pipeline and dist-helper units: fake dataset semantics, device
loader passthrough, sampler sharding, scalar mesh reduce, async logger."""

import numpy as np
import torch

from vit_10b_fsdp_example_amd import dist as xdist
from vit_10b_fsdp_example_amd.cli import parse_args
from vit_10b_fsdp_example_amd.data import (
    DeviceLoader, FakeImageNetDataset, build_datasets,
)


def test_fake_dataset_shapes():
    ds = FakeImageNetDataset(224, 1281167)
    assert len(ds) == 1281167
    x, y = ds[0]
    assert x.shape == (3, 224, 224) and float(x.abs().sum()) == 0.0
    assert y == 0


def test_build_datasets_lengths_and_batching():
    cfg = parse_args([
        "--fake_data", "--image_size", "16", "--batch_size", "4",
        "--num_workers", "0",
    ])
    xdist.init_distributed()
    train_ds, train_loader, train_sampler, val_ds, val_loader, _ = (
        build_datasets(cfg, torch.device("cpu"))
    )
    assert len(train_ds) == 1281167 and len(val_ds) == 50000
    x, y = next(iter(train_loader))
    assert x.shape == (4, 3, 16, 16) and y.shape == (4,)
    # sampler must re-shuffle per epoch deterministically
    train_sampler.set_epoch(1)
    order1 = list(train_sampler)[:10]
    train_sampler.set_epoch(2)
    order2 = list(train_sampler)[:10]
    train_sampler.set_epoch(1)
    order1b = list(train_sampler)[:10]
    assert order1 == order1b and order1 != order2


def test_device_loader_dtype_cast():
    data = [(torch.randn(2, 3, 8, 8), torch.tensor([0, 1]))]
    loader = DeviceLoader(data, torch.device("cpu"),
                          compute_dtype=torch.bfloat16)
    x, y = next(iter(loader))
    assert x.dtype == torch.bfloat16
    assert y.dtype == torch.long
    assert len(loader) == 1


def test_mesh_reduce_single_process():
    assert xdist.mesh_reduce("tag", 3.5, sum) == 3.5
    assert xdist.mesh_reduce("tag", 2, max) == 2


def test_async_logger_runs_closures():
    out = []
    xdist.add_step_closure(lambda v: out.append(v), args=(42,))
    xdist.drain_step_closures()
    assert out == [42]


def test_memory_info_keys():
    info = xdist.get_memory_info()
    for k in ("bytes_used", "bytes_limit", "allocated", "reserved"):
        assert k in info
