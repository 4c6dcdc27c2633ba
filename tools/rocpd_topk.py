"""Summarize a rocprofv3 rocpd SQLite results DB into a top-kernels table.

Usage: python tools/rocpd_topk.py <results.db> [N]

rocprofv3 in this image emits an SQL database (rocpd) instead of CSV
stats; this introspects the schema (tables/columns vary across ROCm
versions) and prints per-kernel total time, call count, mean, and the
total dispatch count — the evidence format kept under profiles/.
"""
from __future__ import annotations

import sqlite3
import sys


def main():
    db, topn = sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 25
    c = sqlite3.connect(db)
    tables = [r[0] for r in c.execute(
        "select name from sqlite_master where type='table'")]

    disp = next((t for t in tables if "kernel_dispatch" in t), None)
    if disp is None:
        print("tables:", tables)
        sys.exit("no kernel_dispatch table")
    cols = [r[1] for r in c.execute(f"pragma table_info({disp})")]

    start = next((x for x in cols if x in ("start", "start_ts", "begin",
                                           "start_timestamp")), None)
    end = next((x for x in cols if x in ("end", "end_ts", "end_timestamp")),
               None)
    kid = next((x for x in cols if "kernel" in x and "id" in x), None)
    if not (start and end and kid):
        print(f"{disp} columns:", cols)
        sys.exit("unrecognized dispatch schema")

    # find the kernel-info table and its name column (usually a string id)
    info = next((t for t in tables if "kernel" in t and
                 ("symbol" in t or "info" in t) and t != disp), None)
    name_expr = None
    if info:
        icols = [r[1] for r in c.execute(f"pragma table_info({info})")]
        ikey = next((x for x in icols if x == "id" or x.endswith("_id")), "id")
        namecol = next((x for x in icols if "name" in x), None)
        if namecol:
            stringt = next((t for t in tables if t.endswith("_string")), None)
            # name column may hold the string directly or a string-table id
            sample = c.execute(
                f"select {namecol} from {info} limit 1").fetchone()
            if sample and isinstance(sample[0], int) and stringt:
                name_expr = (f"(select string from {stringt} "
                             f"where {stringt}.id = i.{namecol})")
            else:
                name_expr = f"i.{namecol}"
            q = (f"select {name_expr} as kname, count(*) as calls, "
                 f"sum(d.{end}-d.{start}) as tot, avg(d.{end}-d.{start}) as av "
                 f"from {disp} d join {info} i on d.{kid} = i.{ikey} "
                 f"group by kname order by tot desc")
    if name_expr is None:
        q = (f"select d.{kid} as kname, count(*) as calls, "
             f"sum(d.{end}-d.{start}) as tot, avg(d.{end}-d.{start}) as av "
             f"from {disp} d group by kname order by tot desc")

    rows = list(c.execute(q))
    total = sum(r[2] for r in rows) or 1
    ncalls = sum(r[1] for r in rows)
    print(f"# {len(rows)} kernels, {ncalls} dispatches, "
          f"{total / 1e6:.1f} ms total GPU time")
    for kname, calls, tot, av in rows[:topn]:
        nm = str(kname)[:95]
        print(f"{100 * tot / total:6.2f}%  {calls:6d}x  {av / 1e6:9.3f}ms  {nm}")


if __name__ == "__main__":
    main()
