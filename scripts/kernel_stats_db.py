"""Summarize a rocprofv3 SQLite results db (--kernel-trace --stats output
on ROCm 7.2 writes rocpd_*.db) into the per-kernel table we commit under
profiles/: name, calls, total us, avg us, % of GPU time.

Usage: python scripts/kernel_stats_db.py gpurun_out/prof_r2a/r2a_results.db
"""
import re
import sqlite3
import sys


def summarize(path, top=40):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    disp = next(t for t in tabs if t.startswith("rocpd_kernel_dispatch"))
    sym = next(t for t in tabs if t.startswith("rocpd_info_kernel_symbol"))
    rows = cur.execute(
        f"SELECT s.display_name, COUNT(*), SUM(d.end - d.start) "
        f"FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id "
        f"GROUP BY s.display_name ORDER BY SUM(d.end - d.start) DESC"
    ).fetchall()
    total = sum(r[2] for r in rows) or 1
    out = [f"{'kernel':<72} {'calls':>7} {'total_ms':>10} {'avg_us':>9} {'pct':>6}"]
    for name, calls, ns in rows[:top]:
        short = re.sub(r"\(.*\)", "", name.strip())[:72]
        out.append(f"{short:<72} {calls:>7} {ns/1e6:>10.2f} "
                   f"{ns/1e3/calls:>9.2f} {100.0*ns/total:>5.1f}%")
    out.append(f"{'TOTAL GPU time':<72} {'':>7} {total/1e6:>10.2f}")
    return "\n".join(out)


if __name__ == "__main__":
    print(summarize(sys.argv[1], top=int(sys.argv[2]) if len(sys.argv) > 2 else 40))
