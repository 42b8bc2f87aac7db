"""hipBLASLt/rocBLAS GEMM algorithm selection (PyTorch TunableOp).

The repo ships a pre-tuned solution table for the ViT-10B GEMM shapes on
gfx950 (tuned/tunableop_gfx950.csv, produced offline on an MI355X).
Loading it is read-only: tuning itself stays disabled at runtime, ops
without an entry fall back to the default heuristic, numerics are
unchanged (same library kernels, different algorithm choice).

Disable with VITFSDP_TUNABLEOP=0.
"""

import os

import torch

_DONE = False


def enable_tunableop():
    global _DONE
    if _DONE or not torch.cuda.is_available():
        return
    _DONE = True
    # opt-in only: TunableOp (both live tuning and read-only table load)
    # was observed to crash the ROCm 7.2 / torch 2.10 stack on gfx950
    # (core dump in the tuned bench run, gpurun_out/run3.log), so the
    # default path never touches it.
    if os.environ.get("VITFSDP_TUNABLEOP", "0") != "1":
        return
    csv = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "tuned", "tunableop_gfx950.csv",
    )
    if not os.path.exists(csv):
        return
    try:
        torch.cuda.tunable.enable(True)
        torch.cuda.tunable.tuning_enable(False)  # read-only: never tune live
        torch.cuda.tunable.set_filename(csv, insert_device_ordinal=False)
        torch.cuda.tunable.read_file(csv)
    except Exception as exc:  # pragma: no cover
        print(f"[tuning] TunableOp disabled ({exc!r})", flush=True)
        try:
            torch.cuda.tunable.enable(False)
        except Exception:
            pass
