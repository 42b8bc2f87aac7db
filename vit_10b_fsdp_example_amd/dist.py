"""Distributed runtime helpers (MI355X-native equivalents of torch_xla's xm.*).

The reference consumes torch_xla.core.xla_model for topology, barriers,
host scalar reduces, rank-0 printing, memory info and deferred step
closures (reference: run_vit_training.py:31,32,205,212,224,289; see
SURVEY.md B8-B15).  Here the same capabilities sit on top of
torch.distributed with the RCCL ("nccl") backend over xGMI for device
collectives and a dedicated gloo group for host-side scalar reduces, so
logging-path collectives never share a communicator with the training
hot path (a background logger thread must not enqueue on the RCCL
communicator that the compute stream is using).
"""

import datetime
import os
import queue
import threading

import torch
import torch.distributed as dist

_STATE = {
    "initialized": False,
    "rank": 0,
    "world_size": 1,
    "local_rank": 0,
    "device": torch.device("cpu"),
    "gloo_group": None,  # host-side scalar reduces / barriers
}

# Collective timeout for the logger side-channel (seconds): bounds how
# long a rank blocks in mesh_reduce when a peer's closure died.
_LOGGER_TIMEOUT_S = int(os.environ.get("VITFSDP_LOGGER_TIMEOUT_S", "300"))


def init_distributed(device_index=None, timeout_minutes=30):
    """Initialise the per-process distributed runtime.

    One process per GPU (reference spawns one worker per device via
    xmp.spawn, run_vit_training.py:364).  Reads RANK / WORLD_SIZE /
    LOCAL_RANK / MASTER_ADDR / MASTER_PORT from the environment the way
    torchrun sets them.  Falls back to a single-process world when no
    env is present so `python run_vit_training.py` works directly.
    """
    if _STATE["initialized"]:
        return get_device()

    if dist.is_available() and dist.is_initialized():
        # An external harness/launcher already created the process group:
        # topology truth is the group, not the env.  The gloo
        # side-channel is still created below so the background logger
        # never shares a communicator with the training hot path.
        rank = dist.get_rank()
        world_size = dist.get_world_size()
        local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    else:
        rank = int(os.environ.get("RANK", "0"))
        world_size = int(os.environ.get("WORLD_SIZE", "1"))
        local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if device_index is not None:
        local_rank = device_index

    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))
        device = torch.device("cuda", local_rank % max(torch.cuda.device_count(), 1))
        backend = "nccl"  # RCCL on ROCm
    else:
        device = torch.device("cpu")
        backend = "gloo"

    # initialize the process group whenever a launcher provided topology
    # env (torchrun sets WORLD_SIZE even for one rank): this keeps the
    # RCCL/gloo communicator-creation path identical between N=1 smoke
    # runs and the real multi-GPU launches
    launched = "WORLD_SIZE" in os.environ
    if (world_size > 1 or launched) and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world_size,
            timeout=datetime.timedelta(minutes=timeout_minutes),
        )
    if dist.is_initialized():
        if dist.get_backend() == "gloo":
            _STATE["gloo_group"] = dist.group.WORLD
        else:
            # Host-side scalar reduces (loss logging, eval counters) go over
            # gloo so the async logger thread never touches the RCCL
            # communicator used by the training step.  Bounded timeout: a
            # closure that fails on one rank leaves its peers' gather
            # erroring out after _LOGGER_TIMEOUT_S instead of deadlocking
            # drain_step_closures() at the epoch boundary.
            _STATE["gloo_group"] = dist.new_group(
                backend="gloo",
                timeout=datetime.timedelta(seconds=_LOGGER_TIMEOUT_S),
            )

    _STATE.update(
        initialized=True,
        rank=rank,
        world_size=world_size,
        local_rank=local_rank,
        device=device,
    )
    return device


def is_distributed():
    return dist.is_available() and dist.is_initialized()


def get_world_size():
    """Reference: xm.xrt_world_size() (run_vit_training.py:31)."""
    return _STATE["world_size"]


def get_rank():
    """Reference: xm.get_ordinal() (run_vit_training.py:32)."""
    return _STATE["rank"]


def get_local_rank():
    """Reference: xm.get_local_ordinal() (run_vit_training.py:220)."""
    return _STATE["local_rank"]


def get_device():
    """Reference: xm.xla_device() (run_vit_training.py:219)."""
    return _STATE["device"]


def master_print(*args, **kwargs):
    """Rank-0-only print (reference: xm.master_print, 17 call sites)."""
    if _STATE["rank"] == 0:
        kwargs.setdefault("flush", True)
        print(*args, **kwargs)


def rendezvous(tag):
    """Named whole-world barrier (reference: xm.rendezvous,
    run_vit_training.py:224,230,241,252).  Uses the gloo group when
    available so it does not interleave with RCCL device collectives."""
    if is_distributed():
        dist.barrier(group=_STATE["gloo_group"])
    return tag


def mesh_reduce(tag, value, reducer):
    """Host-side reduce of a python scalar with an arbitrary reducer
    (reference: xm.mesh_reduce, run_vit_training.py:205,315,316).

    Gathers every rank's value over the gloo group and applies
    ``reducer`` to the list, exactly matching the reference semantics
    (reducer sees all per-rank values, not a pairwise fold).
    """
    if not is_distributed():
        return reducer([value])
    gathered = [None] * _STATE["world_size"]
    dist.all_gather_object(gathered, value, group=_STATE["gloo_group"])
    return reducer(gathered)


def reduce_gradients(optimizer, world_size=None):
    """Plain-DDP gradient all-reduce (mean) for the --run_without_fsdp
    baseline (reference: xm.reduce_gradients, run_vit_training.py:273).

    Bucketed: gradients are flattened into ~64 MiB flat buffers per
    dtype and all-reduced with a single RCCL call per bucket.  xGMI ring
    all-reduce is per-link bound, so fewer/larger messages win.
    """
    if not is_distributed():
        return
    ws = world_size or _STATE["world_size"]
    grads = [
        p.grad
        for group in optimizer.param_groups
        for p in group["params"]
        if p.grad is not None
    ]
    if not grads:
        return
    bucket_bytes = 64 * 1024 * 1024
    buckets = {}
    for g in grads:
        buckets.setdefault(g.dtype, []).append(g)
    for dtype_grads in buckets.values():
        cur, cur_bytes = [], 0
        chunks = []
        for g in dtype_grads:
            cur.append(g)
            cur_bytes += g.numel() * g.element_size()
            if cur_bytes >= bucket_bytes:
                chunks.append(cur)
                cur, cur_bytes = [], 0
        if cur:
            chunks.append(cur)
        for chunk in chunks:
            flat = torch._utils._flatten_dense_tensors(chunk)
            dist.all_reduce(flat)
            flat.div_(ws)
            for g, synced in zip(
                chunk, torch._utils._unflatten_dense_tensors(flat, chunk)
            ):
                g.copy_(synced)


def get_memory_info(device=None):
    """Device memory stats for the log line (reference:
    xm.get_memory_info, run_vit_training.py:212)."""
    if torch.cuda.is_available():
        free, total = torch.cuda.mem_get_info()
        return {
            "bytes_used": total - free,
            "bytes_limit": total,
            "allocated": torch.cuda.memory_allocated(),
            "reserved": torch.cuda.memory_reserved(),
        }
    return {"bytes_used": 0, "bytes_limit": 0, "allocated": 0, "reserved": 0}


class AsyncStepLogger:
    """HIP-event-gated deferred logging (MI355X equivalent of
    xm.add_step_closure, reference run_vit_training.py:289-291).

    The reference defers `loss.item()` to after graph execution so the
    host sync never sits inside the traced step.  Eager HIP has no
    trace/execute split; the equivalent discipline is to never call
    `.item()` on the compute stream mid-step.  Here we record a HIP
    event after the step and hand the closure to a background thread
    that waits on the event before touching tensor values, so the hot
    loop never stalls on device->host copies.
    """

    def __init__(self):
        self._queue = queue.Queue()
        self._thread = None
        self._closed = False
        self.failures = 0

    def _ensure_thread(self):
        if self._thread is None:
            self._thread = threading.Thread(target=self._run, daemon=True)
            self._thread.start()

    def _run(self):
        while True:
            item = self._queue.get()
            if item is None:
                return
            event, fn, args, kwargs = item
            try:
                if event is not None:
                    event.synchronize()
                fn(*args, **kwargs)
            except Exception as exc:  # pragma: no cover - log, don't kill training
                # A failed closure on one rank can leave peers blocked in
                # their logging gather; the gloo side-channel's bounded
                # timeout (_LOGGER_TIMEOUT_S) turns that into an error on
                # their logger threads rather than a drain() deadlock.
                self.failures += 1
                print(
                    f"[async-logger] closure failed ({self.failures} total): "
                    f"{exc!r}",
                    flush=True,
                )
            finally:
                self._queue.task_done()

    def add_step_closure(self, fn, args=(), kwargs=None):
        kwargs = kwargs or {}
        if torch.cuda.is_available():
            event = torch.cuda.Event()
            event.record()
            self._ensure_thread()
            self._queue.put((event, fn, args, kwargs))
        else:
            fn(*args, **kwargs)

    def drain(self):
        """Block until all queued closures have run (epoch end / exit)."""
        if self._thread is not None:
            self._queue.join()

    def close(self):
        if self._thread is not None:
            self.drain()
            self._queue.put(None)
            self._thread.join()
            self._thread = None


_step_logger = AsyncStepLogger()


def add_step_closure(fn, args=(), kwargs=None):
    _step_logger.add_step_closure(fn, args, kwargs)


def drain_step_closures():
    _step_logger.drain()
