"""Step profiling (the reference has only wall-clock
timing; the new framework adds a --profile flag wrapping a few steps in
torch.profiler with HIP kernel attribution).

Produces, per rank, a chrome trace under {ckpt_dir}/profile/ and a
rank-0 top-kernels table on stdout.  Our kernels keep their C++ names
(fmha_*, ln_*, fused_adamw_*, mt_*, ce_*) so rocprofv3 and this table
attribute them directly.
"""

import os

import torch

from . import dist as xdist


class StepProfiler:
    """Profiles steps [wait, wait+active) of the first epoch."""

    def __init__(self, cfg, enabled):
        self.enabled = enabled and torch.cuda.is_available()
        self._prof = None
        self._done = False
        if not self.enabled:
            return
        self.out_dir = os.path.join(cfg.ckpt_dir, "profile")
        os.makedirs(self.out_dir, exist_ok=True)
        self._prof = torch.profiler.profile(
            activities=[
                torch.profiler.ProfilerActivity.CPU,
                torch.profiler.ProfilerActivity.CUDA,
            ],
            schedule=torch.profiler.schedule(wait=3, warmup=1, active=4,
                                             repeat=1),
            on_trace_ready=self._on_ready,
        )
        self._prof.start()

    def _on_ready(self, prof):
        rank = xdist.get_rank()
        path = os.path.join(self.out_dir, f"trace_rank{rank}.json")
        try:
            prof.export_chrome_trace(path)
        except Exception as exc:  # pragma: no cover
            print(f"[profile] trace export failed: {exc!r}", flush=True)
        xdist.master_print(f"[profile] chrome trace written to {path}")
        xdist.master_print(
            prof.key_averages().table(
                sort_by="self_cuda_time_total", row_limit=20
            )
        )

    def step(self):
        if self._prof is not None and not self._done:
            self._prof.step()

    def stop(self):
        if self._prof is not None and not self._done:
            self._done = True
            try:
                self._prof.stop()
            except Exception:
                pass
