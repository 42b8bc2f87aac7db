"""benchmarks/ktrace_diff.py parses rocpd results DBs and attributes
per-kernel deltas (the tool behind the wgrad root-cause in
profiles/PROFILES.md)."""

import sqlite3

from benchmarks.ktrace_diff import _load, diff, shorten


def _make_db(path, rows):
    db = sqlite3.connect(path)
    db.execute(
        "CREATE TABLE top_kernels "
        "(name TEXT, total_calls INT, total_duration REAL, "
        "average REAL, percentage REAL)"
    )
    db.executemany("INSERT INTO top_kernels VALUES (?,?,?,?,?)", rows)
    db.commit()
    db.close()


TENSILE = (
    "Cijk_Ailk_Bjlk_BBS_BH_Bias_HA_S_SAV_UserArgs_MT256x256x32_"
    "MI16x16x1_SN_LDSB0_AFC1_ISA950_WS64_WG32_8_1"
)


def test_diff_attributes_delta(tmp_path):
    a, b = str(tmp_path / "a.db"), str(tmp_path / "b.db")
    _make_db(a, [("kernA", 10, 1000.0, 100.0, 50.0),
                 (TENSILE, 20, 1000.0, 50.0, 50.0)])
    _make_db(b, [("kernA", 10, 1000.0, 100.0, 40.0),
                 (TENSILE, 10, 500.0, 50.0, 20.0),
                 ("kernB", 5, 1000.0, 200.0, 40.0)])
    rows = diff(_load(a), _load(b))
    by_name = {r[0]: r for r in rows}
    # unchanged kernel: zero delta
    assert by_name["kernA"][5] == 0.0
    # tensile name is shortened to the tile-identifying prefix
    short = shorten(TENSILE)
    assert short == "Cijk_Ailk_Bjlk_BBS_MT256x256x32"
    assert by_name[short][5] == -500.0
    # kernel only in B shows with calls_a == 0
    assert by_name["kernB"][1] == 0 and by_name["kernB"][5] == 1000.0
    # sorted by |delta| descending
    deltas = [abs(r[5]) for r in rows]
    assert deltas == sorted(deltas, reverse=True)


def test_fallback_to_raw_kernels_table(tmp_path):
    p = str(tmp_path / "raw.db")
    db = sqlite3.connect(p)
    db.execute("CREATE TABLE kernels (name TEXT, duration REAL)")
    db.executemany(
        "INSERT INTO kernels VALUES (?,?)",
        [("k1", 2_000_000.0), ("k1", 1_000_000.0), ("k2", 500_000.0)],
    )
    db.commit()
    db.close()
    loaded = _load(p)
    assert loaded["k1"] == (2, 3000.0)  # ns summed -> us
    assert loaded["k2"] == (1, 500.0)


def test_cli_main_prints_table(tmp_path, capsys):
    from benchmarks.ktrace_diff import main

    a, b = str(tmp_path / "a.db"), str(tmp_path / "b.db")
    _make_db(a, [("kern", 4, 400.0, 100.0, 100.0)])
    _make_db(b, [("kern", 4, 800.0, 200.0, 100.0)])
    main([a, b, "--top", "5"])
    out = capsys.readouterr().out
    assert "B - A: +0.4 ms" in out
    assert "kern" in out
