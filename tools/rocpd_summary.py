#!/usr/bin/env python3
"""Summarize rocprofv3 rocpd SQLite output (kernel stats + PMC counters).

Usage: python tools/rocpd_summary.py <results.db> [--json OUT]

Per-kernel: dispatch count, total/avg duration (ns), grid, VGPR/SGPR; PMC
runs additionally report the summed counter value per kernel per dispatch.
"""
import argparse
import json
import re
import sqlite3
from collections import defaultdict


def load(db):
    con = sqlite3.connect(db)
    cur = con.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    m = re.search(r"rocpd_metadata_(.*)", [t for t in tabs if "metadata" in t][0])
    uuid = m.group(1)

    def T(name):
        return f"{name}_{uuid}"

    ksym = {r[0]: (r[1], r[2], r[3], r[4]) for r in cur.execute(
        f"SELECT id, display_name, sgpr_count, arch_vgpr_count, accum_vgpr_count "
        f"FROM {T('rocpd_info_kernel_symbol')}")}

    stats = defaultdict(lambda: {"count": 0, "total_ns": 0, "grid": None,
                                 "sgpr": 0, "vgpr": 0})
    disp = {}
    for r in cur.execute(
            f"SELECT id, kernel_id, start, end, grid_size_x, workgroup_size_x, "
            f"event_id FROM {T('rocpd_kernel_dispatch')}"):
        did, kid, start, end, gx, wx, event_id = r
        name, sgpr, vgpr, agpr = ksym.get(kid, ("?", 0, 0, 0))
        name = name.split("(")[0]
        s = stats[name]
        s["count"] += 1
        s["total_ns"] += end - start
        s["grid"] = (gx, wx)
        s["sgpr"], s["vgpr"] = sgpr, vgpr
        disp[event_id] = name

    pmc_names = {r[0]: r[1] for r in cur.execute(
        f"SELECT id, name FROM {T('rocpd_info_pmc')}")}
    pmc = defaultdict(lambda: defaultdict(lambda: {"sum": 0.0, "n": 0}))
    for r in cur.execute(
            f"SELECT event_id, pmc_id, value FROM {T('rocpd_pmc_event')}"):
        event_id, pmc_id, value = r
        kname = disp.get(event_id)
        if kname is None:
            continue
        c = pmc[kname][pmc_names.get(pmc_id, str(pmc_id))]
        c["sum"] += value
        c["n"] += 1

    out = []
    for name, s in sorted(stats.items(), key=lambda kv: -kv[1]["total_ns"]):
        row = {
            "kernel": name,
            "dispatches": s["count"],
            "total_us": s["total_ns"] / 1e3,
            "avg_us": s["total_ns"] / 1e3 / s["count"],
            "grid": s["grid"],
            "sgpr": s["sgpr"], "vgpr": s["vgpr"],
        }
        if name in pmc:
            row["pmc"] = {k: {"sum": v["sum"], "per_dispatch": v["sum"] / max(v["n"], 1),
                              "n": v["n"]}
                          for k, v in pmc[name].items()}
        out.append(row)
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--json")
    args = ap.parse_args()
    rows = load(args.db)
    for r in rows:
        line = (f"{r['kernel'][:48]:48s} n={r['dispatches']:4d} "
                f"avg={r['avg_us']:10.1f}us total={r['total_us']:12.1f}us "
                f"vgpr={r['vgpr']:3d} sgpr={r['sgpr']:3d} grid={r['grid']}")
        print(line)
        for k, v in (r.get("pmc") or {}).items():
            print(f"    {k}: per_dispatch={v['per_dispatch']:.3e} (n={v['n']})")
    if args.json:
        with open(args.json, "w") as f:
            json.dump(rows, f, indent=1)


if __name__ == "__main__":
    main()
