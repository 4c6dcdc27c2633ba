"""Aggregate rocprofv3 --pmc counters from a rocpd SQLite results DB.

Usage: python tools/rocpd_pmc.py <results.db>
Prints total per counter (summed over dispatches and SEs/instances).
"""
import sqlite3
import sys


def main():
    c = sqlite3.connect(sys.argv[1])
    tables = [r[0] for r in c.execute(
        "select name from sqlite_master where type='table'")]
    ev = next((t for t in tables if "pmc_event" in t), None)
    info = next((t for t in tables if t.endswith("info_pmc")), None)
    if ev is None:
        sys.exit(f"no pmc_event table in {tables}")
    names = {}
    if info:
        icols = [r[1] for r in c.execute(f"pragma table_info({info})")]
        idc = next((x for x in icols if x == "id" or x.endswith("_id")), "id")
        namec = next((x for x in icols if "name" in x or "symbol" in x), None)
        if namec:
            for i, nm in c.execute(f"select {idc}, {namec} from {info}"):
                names[i] = str(nm)
    rows = list(c.execute(
        f"select pmc_id, count(*), sum(value) from {ev} group by pmc_id"))
    for pid, cnt, tot in rows:
        print(f"{names.get(pid, pid)}: total {tot:.4g} over {cnt} rows")


if __name__ == "__main__":
    main()
