"""Summarize a rocprofv3 rocpd SQLite DB into a small text report.

Runs ON the GPU box right after rocprofv3 so only the summary (not the
40+ MB DB) travels back through gpurun_out.

Usage: python scripts/prof_summarize.py <results.db> <out.txt> [window_ms]
"""
import sqlite3
import sys


def main(db_path, out_path, window_ms=250.0):
    con = sqlite3.connect(db_path)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    u = kd[len("rocpd_kernel_dispatch_"):]
    tmax = cur.execute(f"SELECT MAX(end) FROM {kd}").fetchone()[0]
    w0 = tmax - window_ms * 1e6
    lines = []
    tot, n = cur.execute(
        f"SELECT SUM(end-start)/1e6, COUNT(*) FROM {kd} "
        f"WHERE start > {w0}").fetchone()
    lines.append("steady-state window {:.0f} ms: {:.1f} ms busy, {} "
                 "dispatches".format(window_ms, tot, n))
    q = (f"SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6, "
         f"AVG(kd.end-kd.start)/1e3 "
         f"FROM {kd} kd JOIN rocpd_info_kernel_symbol_{u} ks "
         f"ON kd.kernel_id = ks.id WHERE kd.start > {w0} "
         f"GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 40")
    for name, calls, ms, avg in cur.execute(q).fetchall():
        lines.append("{:9.3f} ms {:6d}x {:9.1f}us  {}".format(
            ms, calls, avg, name[:110]))
    with open(out_path, "w") as f:
        f.write("\n".join(lines) + "\n")
    print("\n".join(lines[:15]))


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2],
         float(sys.argv[3]) if len(sys.argv) > 3 else 250.0)
