"""Multi-process test helper: run a function on N gloo ranks on CPU.

Used by the FSDP parity tests so the distributed control flow (the real
all-gather / reduce-scatter paths) is exercised without GPUs, matching
BASELINE.json config 1 (plumbing on CPU/gloo).
"""

import multiprocessing as mp
import os
import pickle
import tempfile
import traceback


def _worker(rank, world_size, port, fn, args, result_dir):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        result = fn(rank, world_size, *args)
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("ok", result), f)
    except Exception:
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("err", traceback.format_exc()), f)
        raise


def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_multiprocess(fn, world_size=2, args=(), timeout=300):
    """Run fn(rank, world_size, *args) on `world_size` spawned processes.

    Returns the list of per-rank return values (picklable)."""
    ctx = mp.get_context("spawn")
    port = _free_port()
    with tempfile.TemporaryDirectory() as result_dir:
        procs = [
            ctx.Process(
                target=_worker,
                args=(r, world_size, port, fn, args, result_dir),
            )
            for r in range(world_size)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout)
        for r, p in enumerate(procs):
            if p.is_alive():
                p.terminate()
                raise TimeoutError(f"rank {r} timed out")
        results = []
        for r in range(world_size):
            path = os.path.join(result_dir, f"rank{r}.pkl")
            assert os.path.exists(path), f"rank {r} produced no result (exit {procs[r].exitcode})"
            with open(path, "rb") as f:
                status, payload = pickle.load(f)
            if status == "err":
                raise RuntimeError(f"rank {r} failed:\n{payload}")
            results.append(payload)
        return results


class CountMM:
    """TorchDispatchMode counting aten::mm/addmm dispatches (shared by
    the checkpoint-early-stop and wgrad-dispatch tests)."""

    def __init__(self):
        self.n = 0

    def __enter__(self):
        from torch.utils._python_dispatch import TorchDispatchMode

        outer = self

        class _Mode(TorchDispatchMode):
            def __torch_dispatch__(self, func, types, args=(), kwargs=None):
                import torch

                if func._overloadpacket in (torch.ops.aten.mm,
                                            torch.ops.aten.addmm):
                    outer.n += 1
                return func(*args, **(kwargs or {}))

        self._mode = _Mode()
        self._mode.__enter__()
        return self

    def __exit__(self, *exc):
        return self._mode.__exit__(*exc)
