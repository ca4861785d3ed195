"""Extract a per-kernel stats table from a rocprofv3 rocpd SQLite database.

rocprofv3 on ROCm 7.2 writes results as <pid>_results.db (rocpd format);
this prints the classic --stats style table (and optionally PMC values)
so summaries can be committed under profiles/.

    python tools/rocpd_summary.py gpurun_out/prof/runc/NNN_results.db
"""

from __future__ import annotations

import sqlite3
import sys
from collections import defaultdict


def main(path: str) -> None:
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute("SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [t for t in tables if t.startswith("rocpd_metadata")][0].split("rocpd_metadata_")[1]

    q = f"""
    SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start), AVG(kd.end-kd.start),
           MIN(kd.end-kd.start), MAX(kd.end-kd.start)
    FROM rocpd_kernel_dispatch_{sfx} kd
    JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY SUM(kd.end-kd.start) DESC
    """
    rows = list(db.execute(q))
    total = sum(r[2] for r in rows)
    print(f"{'NAME':70s} {'CALLS':>7s} {'TOTAL_ms':>10s} {'AVG_us':>9s} {'MIN_us':>8s} {'MAX_us':>9s} {'PCT':>6s}")
    for name, cnt, tot, avg, mn, mx in rows:
        print(
            f"{name[:70]:70s} {cnt:7d} {tot/1e6:10.3f} {avg/1e3:9.2f} "
            f"{mn/1e3:8.2f} {mx/1e3:9.2f} {tot/total*100:5.1f}%"
        )

    # PMC values if present, aggregated per kernel+counter
    try:
        q2 = f"""
        SELECT ks.display_name, p.name, AVG(e.value), COUNT(*)
        FROM rocpd_pmc_event_{sfx} e
        JOIN rocpd_info_pmc_{sfx} p ON e.pmc_id = p.id
        JOIN rocpd_kernel_dispatch_{sfx} kd ON e.event_id = kd.event_id
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id = ks.id
        GROUP BY ks.display_name, p.name
        """
        pmc = list(db.execute(q2))
        if pmc:
            print("\nPMC (avg per dispatch):")
            by_kernel: dict[str, list] = defaultdict(list)
            for name, counter, avg, cnt in pmc:
                by_kernel[name].append((counter, avg, cnt))
            for name, counters in by_kernel.items():
                print(f"  {name[:70]}")
                for counter, avg, cnt in sorted(counters):
                    print(f"    {counter:32s} {avg:18.1f}  (n={cnt})")
    except sqlite3.OperationalError:
        pass


if __name__ == "__main__":
    main(sys.argv[1])
