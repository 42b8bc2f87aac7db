"""hipBLASLt/rocBLAS GEMM algorithm selection (PyTorch TunableOp).

STRICTLY OPT-IN (VITFSDP_TUNABLEOP=1): TunableOp — both live tuning and
read-only table loading — crashes this ROCm 7.2 / torch 2.10 stack on
gfx950 (see enable_tunableop below), so the default path never touches
it.  When opted in, the pre-tuned table for the ViT-10B GEMM shapes
(tuned/tunableop_gfx950.csv, produced offline on an MI355X) is loaded
read-only: ops without an entry fall back to the default heuristic and
numerics are unchanged (same library kernels, different algorithm
choice).  The maintained alternative is the standalone
csrc/tools/hipblaslt_search.cpp enumerator (ROADMAP item 5).
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
