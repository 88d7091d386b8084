#!/usr/bin/env python3
"""Aggregate a rocprofv3 rocpd SQLite DB into a small per-kernel summary
(time + PMC counters), so only the summary leaves the GPU box.

Usage: python scripts/aggregate_rocprof.py <results.db> <out.json>
"""
import json
import sqlite3
import sys


def main() -> None:
    db_path, out_path = sys.argv[1], sys.argv[2]
    db = sqlite3.connect(db_path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = next(t for t in tables if t.startswith("rocpd_kernel_dispatch_")
               ).replace("rocpd_kernel_dispatch_", "")

    q = f"""
    SELECT s.display_name, COUNT(*) n, SUM(d.end - d.start)/1e6 total_ms,
           AVG(d.end - d.start)/1e3 avg_us, s.arch_vgpr_count, s.sgpr_count
    FROM rocpd_kernel_dispatch_{sfx} d
    JOIN rocpd_info_kernel_symbol_{sfx} s ON s.id = d.kernel_id
    GROUP BY s.display_name ORDER BY total_ms DESC LIMIT 30
    """
    kernels = [
        {"name": r[0].split("(")[0][:70], "calls": r[1], "total_ms": round(r[2], 2),
         "avg_us": round(r[3], 2), "vgpr": r[4], "sgpr": r[5]}
        for r in db.execute(q)
    ]

    pmc = {}
    if f"rocpd_pmc_event_{sfx}" in tables:
        try:
            qc = f"""
            SELECT s.display_name, i.name, SUM(p.value)
            FROM rocpd_pmc_event_{sfx} p
            JOIN rocpd_info_pmc_{sfx} i ON i.id = p.pmc_id
            JOIN rocpd_kernel_dispatch_{sfx} d ON d.event_id = p.event_id
            JOIN rocpd_info_kernel_symbol_{sfx} s ON s.id = d.kernel_id
            GROUP BY s.display_name, i.name
            """
            for name, counter, value in db.execute(qc):
                pmc.setdefault(name.split("(")[0][:70], {})[counter] = value
        except sqlite3.Error as e:
            pmc = {"error": str(e)}

    with open(out_path, "w") as f:
        json.dump({"kernels": kernels, "pmc": pmc}, f, indent=1)
    print(f"wrote {out_path}: {len(kernels)} kernels, pmc for {len(pmc)}")


if __name__ == "__main__":
    main()
