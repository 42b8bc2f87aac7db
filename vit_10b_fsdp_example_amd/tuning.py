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


# ---------------------------------------------------------------------------
# Offline hipBLASLt algorithm table (csrc/tools/hipblaslt_search.cpp ->
# scripts/make_lt_table.py -> tuned/lt_algos_gfx950.json).  Keys are the
# column-major dual of each training GEMM, "opA,opB,m,n,k"; values the
# library algorithm index that beat the heuristic by the gate margin.
# The table is measured data: committing it IS the default flip (the
# dispatch router activates whenever a table is present).
# ---------------------------------------------------------------------------

_LT_TABLE = None


def lt_algo_table():
    """{(opA, opB, m, n, k): algo_index} from tuned/lt_algos_gfx950.json,
    or {} when absent / disabled via VITFSDP_TUNED_GEMM=0."""
    global _LT_TABLE
    if _LT_TABLE is not None:
        return _LT_TABLE
    if os.environ.get("VITFSDP_TUNED_GEMM", "1") == "0":
        _LT_TABLE = {}
        return _LT_TABLE
    path = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "tuned", "lt_algos_gfx950.json",
    )
    table = {}
    if os.path.exists(path):
        import json

        try:
            raw = json.load(open(path))
            for key, val in raw.get("entries", {}).items():
                opa, opb, m, n, k = key.split(",")
                table[(opa, opb, int(m), int(n), int(k))] = int(val["index"])
        except Exception as exc:  # pragma: no cover
            print(f"[tuning] lt algo table unreadable ({exc!r})", flush=True)
            table = {}
    _LT_TABLE = table
    return _LT_TABLE
