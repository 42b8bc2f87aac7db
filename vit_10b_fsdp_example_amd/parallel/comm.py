"""Collective primitives for the FSDP engine.

Design notes (MI355X / RCCL over xGMI):
  * Each MI355X has 7 point-to-point xGMI links (~153 GB/s each); RCCL
    collectives are issued as ONE call per FSDP unit (the unit's whole
    flat parameter is a single bucket, ~630 MB bf16 for the 10B config),
    which is exactly the large-message regime where RCCL's multi-link
    algorithms pay off — never many small per-tensor calls.
  * Parameter all-gathers and gradient reduce-scatters run on two
    *separate* communicators (process groups), so RCCL can overlap the
    backward's re-gather stream with the gradient reduction stream while
    the compute stream keeps running; each group's call order is
    identical on every rank (deterministic module order), which is the
    RCCL deadlock-freedom requirement.
  * gloo (CPU test path) lacks reduce_scatter; it is emulated with
    all_reduce + local slice so the multi-process CPU tests exercise the
    very same FSDP control flow.
"""

import os

import torch
import torch.distributed as dist


class _NoopWork:
    def wait(self):
        return True


class _MultiWork:
    """Aggregate of several async work handles (one-shot P2P gathers)."""

    def __init__(self, works):
        self._works = works

    def wait(self):
        for w in self._works:
            w.wait()
        return True


_NOOP = _NoopWork()

# All-gather algorithm (SURVEY.md §5 "Distributed communication backend"):
#   "allgather" — the RCCL collective (default; RCCL picks its own
#                 multi-link algorithm for large messages)
#   "p2p"       — direct one-shot: every rank sends its shard to all
#                 ws-1 peers and receives each peer's shard straight
#                 into its slice of the full buffer, in one batched
#                 group.  On the xGMI full mesh this drives all 7 links
#                 simultaneously instead of a per-link-bound ring, with
#                 zero staging copies (the recv targets are contiguous
#                 slices of the destination).
# Correctness is backend-independent (gloo parity tests cover p2p);
# which wins on real xGMI is a round-2 A/B (ROADMAP item 1).
_AG_ALGO = os.environ.get("VITFSDP_AG_ALGO", "allgather")

# Reduce-scatter algorithm: "reducescatter" (RCCL collective, default)
# or "p2p" — direct scatter + local reduction: every rank sends slice p
# of its full gradient to peer p, receives ws-1 peer slices, and sums
# them with its own slice at wait() time.
_RS_ALGO = os.environ.get("VITFSDP_RS_ALGO", "reducescatter")


class _DeferredWork:
    """Work handle that runs a local finalizer after the async sends/
    receives complete (one-shot P2P reduce-scatter)."""

    def __init__(self, works, finish):
        self._works = works
        self._finish = finish

    def wait(self):
        for w in self._works:
            w.wait()
        if self._finish is not None:
            self._finish()
            self._finish = None
        return True


def _backend_is_gloo(group):
    if not dist.is_initialized():
        return True
    try:
        return dist.get_backend(group) == "gloo"
    except Exception:
        return False


class CommContext:
    """Holds the process groups used by all FSDP units in the process.

    One gather group (param all-gathers, fwd+bwd) and one reduce group
    (grad reduce-scatters).  On RCCL these are distinct communicators so
    the two directions overlap; on gloo / single process they alias the
    default group.
    """

    _instance = None

    def __init__(self):
        self.world_size = dist.get_world_size() if dist.is_initialized() else 1
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        if dist.is_initialized() and dist.get_backend() == "nccl":
            self.gather_group = dist.new_group(backend="nccl")
            self.reduce_group = dist.new_group(backend="nccl")
        elif dist.is_initialized():
            self.gather_group = dist.group.WORLD
            self.reduce_group = dist.group.WORLD
        else:
            self.gather_group = None
            self.reduce_group = None

    @classmethod
    def get(cls):
        if cls._instance is None:
            cls._instance = cls()
        return cls._instance

    @classmethod
    def reset(cls):
        cls._instance = None

    # -- param all-gather ---------------------------------------------------

    def _all_gather_p2p(self, full, shard, async_op):
        """Direct one-shot all-gather: batched isend/irecv to/from every
        peer, receiving straight into `full`'s per-rank slices."""
        n = shard.numel()
        full.narrow(0, self.rank * n, n).copy_(shard)
        ops = []
        for peer in range(self.world_size):
            if peer == self.rank:
                continue
            ops.append(dist.P2POp(dist.isend, shard, peer,
                                  group=self.gather_group))
            ops.append(dist.P2POp(dist.irecv, full.narrow(0, peer * n, n),
                                  peer, group=self.gather_group))
        works = dist.batch_isend_irecv(ops)
        if async_op:
            return _MultiWork(works)
        for w in works:
            w.wait()
        return _NOOP

    def all_gather_into(self, full, shard, async_op=False):
        """Gather each rank's `shard` into `full` (full.numel == ws * shard.numel).

        Returns a work handle (always has .wait())."""
        if self.world_size == 1:
            full.copy_(shard)
            return _NOOP
        if _AG_ALGO == "p2p":
            return self._all_gather_p2p(full, shard, async_op)
        if _backend_is_gloo(self.gather_group):
            try:
                dist.all_gather_into_tensor(full, shard, group=self.gather_group)
            except (RuntimeError, NotImplementedError):
                chunks = list(full.chunk(self.world_size))
                dist.all_gather(chunks, shard, group=self.gather_group)
            return _NOOP
        work = dist.all_gather_into_tensor(
            full, shard, group=self.gather_group, async_op=async_op
        )
        return work if async_op else _NOOP

    # -- grad reduce-scatter ------------------------------------------------

    def _reduce_scatter_p2p(self, out_shard, full, async_op):
        """Direct scatter + local reduce: slice p of `full` goes to peer
        p; the ws-1 received slices are summed with our own slice when
        the handle is waited.  The caller must keep `full` alive until
        wait() (the FSDP engine frees the grad buffer only after
        finalization)."""
        n = out_shard.numel()
        stage = torch.empty((self.world_size - 1) * n, dtype=full.dtype,
                            device=full.device)
        ops, slot = [], 0
        for peer in range(self.world_size):
            if peer == self.rank:
                continue
            ops.append(dist.P2POp(dist.isend, full.narrow(0, peer * n, n),
                                  peer, group=self.reduce_group))
            ops.append(dist.P2POp(dist.irecv, stage.narrow(0, slot * n, n),
                                  peer, group=self.reduce_group))
            slot += 1
        works = dist.batch_isend_irecv(ops)

        def finish():
            # accumulate in fp32 with ONE final rounding to the comm
            # dtype — a ring reduce-scatter rounds to bf16 at every hop
            # (ws-1 times); the one-shot layout gets better gradient
            # numerics at identical comm volume
            acc = full.narrow(0, self.rank * n, n).to(torch.float32)
            acc += stage.view(self.world_size - 1, n).to(torch.float32).sum(0)
            out_shard.copy_(acc)

        work = _DeferredWork(works, finish)
        if async_op:
            return work
        work.wait()
        return _NOOP

    def reduce_scatter_into(self, out_shard, full, async_op=False):
        """Sum-reduce `full` across ranks, scattering shard `rank` into
        `out_shard`.  Caller divides by world size (mean semantics)."""
        if self.world_size == 1:
            out_shard.copy_(full)
            return _NOOP
        if _RS_ALGO == "p2p":
            return self._reduce_scatter_p2p(out_shard, full, async_op)
        if _backend_is_gloo(self.reduce_group):
            dist.all_reduce(full, group=self.reduce_group)
            n = out_shard.numel()
            out_shard.copy_(full.narrow(0, self.rank * n, n))
            return _NOOP
        work = dist.reduce_scatter_tensor(
            out_shard, full, op=dist.ReduceOp.SUM,
            group=self.reduce_group, async_op=async_op,
        )
        return work if async_op else _NOOP

    # -- scalar all-reduce (grad-norm, C3) ----------------------------------

    def all_reduce_scalar_(self, t):
        if self.world_size > 1:
            dist.all_reduce(t, group=self.reduce_group)
        return t
