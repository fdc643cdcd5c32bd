"""Aggregate per-kernel statistics from a rocprofv3 rocpd SQLite database.

rocprofv3 (ROCm 7.2) writes `*_results.db` with per-run UUID-suffixed tables
(`rocpd_kernel_dispatch_<uuid>` joined to `rocpd_info_kernel_symbol_<uuid>`).
Prints a markdown table of calls / total ms / avg us per kernel, sorted by
total time — the summary we commit under profiles/.

Usage: python tools/parse_rocpd.py <results.db> [top_n]
"""

import sqlite3
import sys


def kernel_stats(db_path, top_n=30):
    con = sqlite3.connect(db_path)
    tabs = [r[0] for r in con.execute(
        "SELECT name FROM sqlite_master WHERE type='table'"
    )]
    dispatch = [t for t in tabs if t.startswith("rocpd_kernel_dispatch_")]
    rows = []
    for dt in dispatch:
        uuid = dt[len("rocpd_kernel_dispatch_"):]
        sym = f"rocpd_info_kernel_symbol_{uuid}"
        if sym not in tabs:
            continue
        q = f"""
            SELECT s.display_name, COUNT(*), SUM(d.end - d.start),
                   AVG(d.end - d.start)
            FROM {dt} d JOIN {sym} s ON d.kernel_id = s.id
            GROUP BY s.display_name
        """
        rows.extend(con.execute(q).fetchall())
    # merge across uuids (multi-process runs)
    agg = {}
    for name, calls, total, _ in rows:
        c, t = agg.get(name, (0, 0))
        agg[name] = (c + calls, t + (total or 0))
    out = sorted(agg.items(), key=lambda kv: -kv[1][1])[:top_n]
    return [(name, c, t / 1e6, t / 1e3 / max(1, c)) for name, (c, t) in out]


def main():
    db = sys.argv[1]
    top_n = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    stats = kernel_stats(db, top_n)
    total_ms = sum(s[2] for s in stats)
    print("| kernel | calls | total ms | avg us |")
    print("|---|---|---|---|")
    for name, calls, tot_ms, avg_us in stats:
        print(f"| {name[:70]} | {calls} | {tot_ms:.1f} | {avg_us:.1f} |")
    print(f"\nSum of listed: {total_ms:.0f} ms")


if __name__ == "__main__":
    main()
