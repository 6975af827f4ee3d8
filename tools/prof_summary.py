"""Summarise a rocprofv3 results.db kernel trace into a text table.

    python tools/prof_summary.py gpurun_out/prof4/prof4_results.db > profiles/...
"""
import sqlite3
import sys


def main(path, top=40):
    db = sqlite3.connect(path)
    cur = db.cursor()
    sfx = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")
        if r[0].startswith("rocpd_kernel_dispatch")][0]
    sfx = sfx.replace("rocpd_kernel_dispatch_", "")
    n, tot = cur.execute(
        f"SELECT COUNT(*), SUM(end-start)/1e6 FROM rocpd_kernel_dispatch_{sfx}"
    ).fetchone()
    print("total kernels: %d, total GPU time: %.1f ms" % (n, tot))
    print("%10s %7s %9s  %s" % ("total_ms", "count", "avg_us", "kernel"))
    q = f"""SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
        AVG(k.end-k.start)/1e3
        FROM rocpd_kernel_dispatch_{sfx} k
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id=ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC LIMIT {top}"""
    for name, cnt, ms, avg in cur.execute(q):
        print("%10.2f %7d %9.1f  %s" % (ms, cnt, avg, name[:110]))


if __name__ == "__main__":
    main(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 40)
