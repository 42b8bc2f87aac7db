#!/usr/bin/env python3
"""Convert csrc/tools/hipblaslt_search.cpp CSV output into the tuned
algorithm table the dispatch router loads (tuned/lt_algos_gfx950.json).

Usage:
    python scripts/make_lt_table.py gpurun_out/hipblaslt_search_vit10b.csv \
        [more.csv ...] [--gate 1.03] [-o tuned/lt_algos_gfx950.json]

An entry is emitted only when the best searched algorithm beats the
heuristic pick by the gate factor (default 3%), so committing the table
can only improve on torch's own picks.  Keys are the column-major dual
"opA,opB,m,n,k" (see ops/linear.py _dual_key).
"""

import argparse
import csv
import json
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("csvs", nargs="+")
    ap.add_argument("--gate", type=float, default=1.03)
    ap.add_argument("-o", "--out", default="tuned/lt_algos_gfx950.json")
    args = ap.parse_args()

    # problems[key] = {"heuristic": (idx, ms), "best": (idx, ms), "name"}
    problems = {}
    for path in args.csvs:
        with open(path) as f:
            for row in csv.DictReader(f):
                try:
                    key = ",".join(
                        [row["opA"], row["opB"], row["m"], row["n"], row["k"]]
                    )
                    idx, ms = int(row["algo_index"]), float(row["ms"])
                except (KeyError, ValueError):
                    continue
                rec = problems.setdefault(
                    key, {"heuristic": None, "best": None, "name": row["problem"]}
                )
                if row["note"] == "heuristic":
                    if rec["heuristic"] is None or ms < rec["heuristic"][1]:
                        rec["heuristic"] = (idx, ms)
                elif rec["best"] is None or ms < rec["best"][1]:
                    rec["best"] = (idx, ms)

    entries = {}
    for key, rec in sorted(problems.items()):
        if rec["heuristic"] is None or rec["best"] is None:
            continue
        h_idx, h_ms = rec["heuristic"]
        b_idx, b_ms = rec["best"]
        gain = h_ms / b_ms
        line = (f"{rec['name']:12s} {key:28s} heur {h_idx:7d} {h_ms:8.3f} ms"
                f"  best {b_idx:7d} {b_ms:8.3f} ms  gain {gain:5.3f}")
        if gain >= args.gate and b_idx != h_idx:
            entries[key] = {
                "index": b_idx,
                "name": rec["name"],
                "heuristic_index": h_idx,
                "heuristic_ms": round(h_ms, 4),
                "best_ms": round(b_ms, 4),
                "gain": round(gain, 4),
            }
            print("TUNE", line)
        else:
            print("keep", line)

    out = {
        "comment": (
            "offline hipBLASLt algorithm search winners (gate "
            f"{args.gate:.2f}x vs heuristic); produced by "
            "scripts/make_lt_table.py from csrc/tools/hipblaslt_search.cpp "
            "output on MI355X"
        ),
        "entries": entries,
    }
    with open(args.out, "w") as f:
        json.dump(out, f, indent=2)
    print(f"wrote {args.out}: {len(entries)} tuned entries")
    return 0


if __name__ == "__main__":
    sys.exit(main())
