#!/usr/bin/env python3
"""Aggregate per-kernel times from a rocprofv3 rocpd SQLite database.

    python benchmarks/extract_rocpd_stats.py <results.db> [out.csv]

Introspects the schema (rocpd table names vary across rocprofv3
versions) and writes kernel,dispatches,total_us,avg_us,pct rows.
"""
import sqlite3
import sys


def main():
    db = sys.argv[1]
    out = sys.argv[2] if len(sys.argv) > 2 else None
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next((t for t in tables if "kernel_dispatch" in t), None)
    if disp is None:
        print("tables:", tables)
        raise SystemExit("no kernel_dispatch table")
    cols = [r[1] for r in cur.execute(f"PRAGMA table_info({disp})")]

    name_expr = None
    if "kernel_name" in cols:
        name_expr = "kernel_name"
        rows = cur.execute(
            f"SELECT {name_expr}, COUNT(*), SUM(end-start) FROM {disp} GROUP BY 1"
        ).fetchall()
    else:
        # names live in a string/symbol table
        sym = next((t for t in tables if "kernel_symbol" in t or "kernel_code" in t), None)
        strtab = next((t for t in tables if t.endswith("_string")), None)
        kid = next((c for c in cols if c in ("kernel_id", "kernel_symbol_id", "code_object_id")), None)
        if sym:
            symcols = [r[1] for r in cur.execute(f"PRAGMA table_info({sym})")]
            nm = next((c for c in symcols if "display_name" in c or "kernel_name" in c or c == "name"), None)
            symid = next((c for c in symcols if c == "id" or c.endswith("_id")), "id")
            q = (f"SELECT s.{nm}, COUNT(*), SUM(d.end-d.start) FROM {disp} d "
                 f"JOIN {sym} s ON d.{kid} = s.{symid} GROUP BY 1")
            try:
                rows = cur.execute(q).fetchall()
            except sqlite3.Error as e:
                print("query failed:", e, "sym cols:", symcols, "disp cols:", cols)
                raise
            # names may be string-table ids
            if rows and isinstance(rows[0][0], int) and strtab:
                id2s = dict(cur.execute(f"SELECT id, string FROM {strtab}"))
                rows = [(id2s.get(r[0], str(r[0])), r[1], r[2]) for r in rows]
        else:
            print("disp cols:", cols, "tables:", tables)
            raise SystemExit("cannot resolve kernel names")

    total = sum(r[2] for r in rows) or 1
    rows.sort(key=lambda r: -r[2])
    lines = ["kernel,dispatches,total_us,avg_us,pct"]
    for name, n, ns in rows[:25]:
        short = str(name).split("(")[0].split("<")[0].strip().split(" ")[-1]
        lines.append(f'"{short}",{n},{ns/1e3:.1f},{ns/1e3/max(n,1):.1f},{100.0*ns/total:.1f}')
    text = "\n".join(lines)
    print(text)
    if out:
        with open(out, "w") as f:
            f.write(text + "\n")


if __name__ == "__main__":
    main()
