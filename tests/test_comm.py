"""Direct unit tests of the collective primitives (parallel/comm.py) —
both the library algorithms and the one-shot P2P variants, sync and
async handles, at world_size 2 (gloo).  The FSDP parity suite covers
them end-to-end; these localize a failure to the primitive."""

import numpy as np
import torch

from tests.utils_mp import run_multiprocess


def _roundtrip(rank, world_size, ag_algo, rs_algo):
    import torch
    import vit_10b_fsdp_example_amd.parallel.comm as comm_mod
    from vit_10b_fsdp_example_amd import dist as xdist
    from vit_10b_fsdp_example_amd.parallel import CommContext

    comm_mod._AG_ALGO = ag_algo
    comm_mod._RS_ALGO = rs_algo
    CommContext.reset()
    xdist.init_distributed()
    ctx = CommContext.get()
    n = 8

    # all-gather: rank r contributes [r*100 .. r*100+7]
    shard = torch.arange(n, dtype=torch.float32) + 100.0 * rank
    full = torch.empty(world_size * n, dtype=torch.float32)
    ctx.all_gather_into(full, shard, async_op=True).wait()
    ag = full.clone().numpy()

    # reduce-scatter: rank r contributes full vector of (r+1)s;
    # slice k of the sum is sum(r+1) everywhere
    grad = torch.full((world_size * n,), float(rank + 1))
    out = torch.empty(n, dtype=torch.float32)
    ctx.reduce_scatter_into(out, grad, async_op=True).wait()
    rs = out.clone().numpy()

    # scalar all-reduce
    s = torch.tensor([float(rank + 1)])
    ctx.all_reduce_scalar_(s)
    return ag, rs, float(s)


def _check(world_size, ag_algo, rs_algo):
    results = run_multiprocess(
        _roundtrip, world_size=world_size, args=(ag_algo, rs_algo)
    )
    n = 8
    expect_ag = np.concatenate(
        [np.arange(n, dtype=np.float32) + 100.0 * r for r in range(world_size)]
    )
    expect_rs = np.full(n, sum(range(1, world_size + 1)), dtype=np.float32)
    expect_s = float(sum(range(1, world_size + 1)))
    for ag, rs, s in results:
        np.testing.assert_array_equal(ag, expect_ag)
        np.testing.assert_array_equal(rs, expect_rs)
        assert s == expect_s


def test_library_algos_ws2():
    _check(2, "allgather", "reducescatter")


def test_p2p_algos_ws2():
    _check(2, "p2p", "p2p")


def test_p2p_algos_ws4():
    _check(4, "p2p", "p2p")


def test_single_process_short_circuit():
    from vit_10b_fsdp_example_amd.parallel import CommContext

    CommContext.reset()
    ctx = CommContext.get()
    assert ctx.world_size == 1
    n = 4
    shard = torch.arange(n, dtype=torch.float32)
    full = torch.empty(n)
    ctx.all_gather_into(full, shard).wait()
    np.testing.assert_array_equal(full.numpy(), shard.numpy())
    out = torch.empty(n)
    ctx.reduce_scatter_into(out, shard).wait()
    np.testing.assert_array_equal(out.numpy(), shard.numpy())
