#!/usr/bin/env python3
"""Extract a per-kernel stats table from rocprofv3 output (rocpd sqlite DB
or kernel_stats CSV) into the plain-text format committed under profiles/.

Usage: python scripts/prof_extract.py <dir-or-file> [out.txt]
Introspects the schema (rocprofv3's table/column names move between
versions) and prints: kernel, calls, total_us, avg_us, %, plus
vgpr/sgpr/scratch when the symbol table carries them.
"""

import csv
import glob
import os
import sqlite3
import sys


def find_inputs(path):
    if os.path.isfile(path):
        return [path]
    return (sorted(glob.glob(os.path.join(path, "**", "*.db"), recursive=True))
            or sorted(glob.glob(os.path.join(path, "**", "*kernel_stats*.csv"),
                                recursive=True)))


def cols(con, table):
    return [r[1] for r in con.execute(f"PRAGMA table_info('{table}')")]


def pick(names, *subs):
    for s in subs:
        for n in names:
            if s in n.lower():
                return n
    return None


def extract_db(path):
    con = sqlite3.connect(path)
    tables = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    dispatch = pick(tables, "kernel_dispatch")
    symbol = pick(tables, "kernel_symbol", "kernel_info", "symbol")
    if not dispatch:
        print(f"## {path}: no kernel_dispatch table; tables = {tables}")
        return []
    dc = cols(con, dispatch)
    start = pick(dc, "start")
    end = pick(dc, "end")
    kid = pick(dc, "kernel_id", "symbol_id", "kernel")
    rows = {}
    names = {}
    res = {}
    if symbol:
        sc = cols(con, symbol)
        sid = pick(sc, "id")
        sname = pick(sc, "display_name", "kernel_name", "name")
        svgpr = pick(sc, "arch_vgpr", "vgpr")
        ssgpr = pick(sc, "sgpr")
        sscr = pick(sc, "private_segment", "scratch")
        sagpr = pick(sc, "accum_vgpr", "agpr")
        for r in con.execute(f"SELECT * FROM {symbol}"):
            d = dict(zip(sc, r))
            names[d[sid]] = d.get(sname, "?")
            res[d[sid]] = (d.get(svgpr), d.get(sagpr), d.get(ssgpr), d.get(sscr))
    for r in con.execute(f"SELECT {kid}, {start}, {end} FROM {dispatch}"):
        k, s, e = r
        dur = (e - s) / 1e3  # ns -> us
        c, t = rows.get(k, (0, 0.0))
        rows[k] = (c + 1, t + dur)
    out = []
    for k, (c, t) in rows.items():
        nm = str(names.get(k, k))
        v = res.get(k, (None,) * 4)
        out.append((nm, c, t, v))
    return out


def extract_csv(path):
    out = []
    with open(path) as f:
        for row in csv.DictReader(f):
            nm = row.get("Name") or row.get("KernelName") or "?"
            c = int(row.get("Calls") or row.get("TotalCalls") or 1)
            t = float(row.get("TotalDurationNs", 0)) / 1e3
            out.append((nm, c, t, (None,) * 4))
    return out


def main():
    path = sys.argv[1]
    inputs = find_inputs(path)
    if not inputs:
        print(f"no rocprof outputs under {path}")
        sys.exit(1)
    rows = []
    for p in inputs:
        rows += extract_db(p) if p.endswith(".db") else extract_csv(p)
    agg = {}
    for nm, c, t, v in rows:
        c0, t0, v0 = agg.get(nm, (0, 0.0, v))
        agg[nm] = (c0 + c, t0 + t, v0 if v0[0] is not None else v)
    total = sum(t for _c, t, _v in agg.values()) or 1.0
    lines = [f"{'kernel':<52} {'calls':>6} {'total_us':>11} {'avg_us':>9} "
             f"{'%':>6} {'vgpr':>5} {'agpr':>5} {'sgpr':>5} {'scratch_B':>9}"]
    for nm, (c, t, v) in sorted(agg.items(), key=lambda kv: -kv[1][1]):
        vg, ag, sg, scr = (x if x is not None else "-" for x in v)
        lines.append(f"{nm[:52]:<52} {c:>6} {t:>11.1f} {t / c:>9.2f} "
                     f"{t / total * 100:>5.1f}% {vg:>5} {ag:>5} {sg:>5} {scr:>9}")
    text = "\n".join(lines) + "\n"
    print(text)
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
