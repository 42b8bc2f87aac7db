"""Diff two rocprofv3 kernel traces (rocpd SQLite results DBs).

Purpose (ROADMAP.md item 4 and general A/B work): given two
`rocprofv3 --kernel-trace --stats -d DIR -- ...` runs of the *same*
workload with one variable flipped (e.g. VITFSDP_NATIVE_WGRAD on/off),
show per-kernel-name total-time deltas so the cost of the flipped
variable can be attributed to specific kernels instead of guessed at
from the end-to-end number.

Usage:
    python benchmarks/ktrace_diff.py A_results.db B_results.db \
        [--top 30] [--like SUBSTR] [--full-names]

Reads the `top_kernels` view rocprofv3 materialises in every results DB
(columns: name, total_calls, total_duration [us], average, percentage);
falls back to aggregating the raw `kernels` table when the view is
absent (older rocprofv3).  Pure stdlib - runs anywhere, including the
no-GPU container, on DBs copied back through gpurun_out/.
"""

import argparse
import re
import sqlite3


def _load(path):
    """-> {kernel_name: (calls, total_us)}"""
    db = sqlite3.connect(f"file:{path}?mode=ro", uri=True)
    try:
        try:
            rows = db.execute(
                "SELECT name, total_calls, total_duration FROM top_kernels"
            ).fetchall()
        except sqlite3.OperationalError:
            rows = db.execute(
                "SELECT name, COUNT(*), SUM(duration) / 1000.0 FROM kernels "
                "GROUP BY name"
            ).fetchall()
        return {name: (int(calls), float(total)) for name, calls, total in rows}
    finally:
        db.close()


def shorten(name, full=False):
    """Compress Tensile/hipBLASLt kernel names (500+ chars of tuning
    tokens) down to the tile-identifying prefix; leave others alone."""
    if full:
        return name
    m = re.match(r"(Custom_)?(Cijk_[A-Za-z]+_[A-Za-z]+_[A-Z]+)_.*?"
                 r"(MT\d+x\d+x\d+)", name)
    if m:
        return f"{m.group(1) or ''}{m.group(2)}_{m.group(3)}"
    return name if len(name) <= 80 else name[:77] + "..."


def diff(a, b, like=None, full_names=False):
    """-> sorted list of (name, calls_a, us_a, calls_b, us_b, delta_us)."""
    merged = {}
    for src, idx in ((a, 0), (b, 1)):
        for name, (calls, us) in src.items():
            short = shorten(name, full_names)
            if like and like.lower() not in name.lower():
                continue
            row = merged.setdefault(short, [0, 0.0, 0, 0.0])
            row[2 * idx] += calls
            row[2 * idx + 1] += us
    out = [
        (name, ca, ua, cb, ub, ub - ua)
        for name, (ca, ua, cb, ub) in merged.items()
    ]
    out.sort(key=lambda r: -abs(r[5]))
    return out


def main(argv=None):
    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("db_a", help="baseline results DB")
    ap.add_argument("db_b", help="variant results DB")
    ap.add_argument("--top", type=int, default=30)
    ap.add_argument("--like", default=None,
                    help="only kernels whose (full) name contains this")
    ap.add_argument("--full-names", action="store_true")
    args = ap.parse_args(argv)

    a, b = _load(args.db_a), _load(args.db_b)
    rows = diff(a, b, like=args.like, full_names=args.full_names)

    tot_a = sum(us for _, us in a.values())
    tot_b = sum(us for _, us in b.values())
    print(f"A: {args.db_a}  total {tot_a / 1e3:.1f} ms "
          f"({sum(c for c, _ in a.values())} dispatches)")
    print(f"B: {args.db_b}  total {tot_b / 1e3:.1f} ms "
          f"({sum(c for c, _ in b.values())} dispatches)")
    print(f"B - A: {(tot_b - tot_a) / 1e3:+.1f} ms\n")
    hdr = (f"{'kernel':<64} {'callsA':>7} {'msA':>9} "
           f"{'callsB':>7} {'msB':>9} {'Δms':>9}")
    print(hdr)
    print("-" * len(hdr))
    for name, ca, ua, cb, ub, d in rows[: args.top]:
        print(f"{name:<64} {ca:>7} {ua / 1e3:>9.2f} "
              f"{cb:>7} {ub / 1e3:>9.2f} {d / 1e3:>+9.2f}")
    only_a = [n for n, ca, _, cb, _, _ in rows if cb == 0 and ca > 0]
    only_b = [n for n, ca, _, cb, _, _ in rows if ca == 0 and cb > 0]
    if only_a:
        print(f"\nonly in A ({len(only_a)}): " + ", ".join(only_a[:10]))
    if only_b:
        print(f"only in B ({len(only_b)}): " + ", ".join(only_b[:10]))


if __name__ == "__main__":
    main()
