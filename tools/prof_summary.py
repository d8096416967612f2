#!/usr/bin/env python3
"""Summarize rocprofv3 result DBs (gpurun_out/prof/*.db) into profiles/ as
committed, human-readable evidence (kernel stats + PMC counters)."""
import glob
import json
import os
import sqlite3
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def summarize(db):
    con = sqlite3.connect(db)
    tabs = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    out = {"db": os.path.basename(db), "kernels": [], "counters": {}}
    ksym = [t for t in tabs if 'kernel_symbol' in t]
    kdis = [t for t in tabs if 'kernel_dispatch' in t]
    if ksym and kdis:
        for r in con.execute(
            f"""SELECT s.display_name, COUNT(*), AVG(d.end-d.start),
                       MIN(d.end-d.start), MAX(d.end-d.start)
                FROM {kdis[0]} d JOIN {ksym[0]} s ON d.kernel_id = s.id
                GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC"""):
            out["kernels"].append({
                "name": r[0][:80], "dispatches": r[1],
                "avg_ms": round(r[2] / 1e6, 4),
                "min_ms": round(r[3] / 1e6, 4), "max_ms": round(r[4] / 1e6, 4)})
    pev = [t for t in tabs if 'pmc_event' in t]
    pinf = [t for t in tabs if 'info_pmc' in t]
    if pev and pinf:
        for r in con.execute(
            f"""SELECT i.name, COUNT(*), AVG(e.value), SUM(e.value)
                FROM {pev[0]} e JOIN {pinf[0]} i ON e.pmc_id = i.id
                GROUP BY i.name"""):
            out["counters"][r[0]] = {"n": r[1], "avg": r[2], "sum": r[3]}
    return out


def main():
    dbs = sorted(glob.glob(os.path.join(REPO, "gpurun_out", "prof", "*.db")))
    os.makedirs(os.path.join(REPO, "profiles"), exist_ok=True)
    tag = sys.argv[1] if len(sys.argv) > 1 else "r01"
    res = [summarize(db) for db in dbs]
    path = os.path.join(REPO, "profiles", f"rocprof_{tag}.json")
    with open(path, "w") as f:
        json.dump(res, f, indent=1)
    print("wrote", path)
    for r in res:
        print("==", r["db"])
        for k in r["kernels"][:4]:
            print(f"   {k['name'][:60]:60s} n={k['dispatches']} avg={k['avg_ms']}ms")
        for c, v in r["counters"].items():
            print(f"   {c}: avg={v['avg']:.0f}")


if __name__ == "__main__":
    main()
