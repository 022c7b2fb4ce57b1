#!/usr/bin/env python3
"""Summarize a rocpd kernel-trace DB: top kernels in the last N ms."""
import glob
import sqlite3
import sys


def summarize(db_glob: str, window_ms: float, out_path: str, header: str = ""):
    db = glob.glob(db_glob)[0]
    con = sqlite3.connect(db)
    cur = con.cursor()
    sfx = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'"
    )][0].replace("rocpd_kernel_dispatch_", "")
    t0, t1 = list(cur.execute(
        f"SELECT MIN(start),MAX(end) FROM rocpd_kernel_dispatch_{sfx}"))[0]
    cut = t1 - window_ms * 1e6
    busy, ndisp = list(cur.execute(
        f"SELECT SUM(kd.end-kd.start)/1e6, COUNT(*) FROM "
        f"rocpd_kernel_dispatch_{sfx} kd WHERE kd.start > {cut}"))[0]
    rows = list(cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
               AVG(kd.end-kd.start)/1e3
        FROM rocpd_kernel_dispatch_{sfx} kd
        JOIN rocpd_info_kernel_symbol_{sfx} ks ON kd.kernel_id=ks.id
        WHERE kd.start > {cut} GROUP BY ks.display_name
        ORDER BY 3 DESC LIMIT 20"""))
    lines = [header,
             f"# window last {window_ms:.0f} ms: busy {busy:.1f} ms / "
             f"{ndisp} dispatches; idle {window_ms - busy:.1f} ms",
             "  total_ms  calls   avg_us  kernel"]
    for name, calls, ms, avg in rows:
        lines.append(f"{ms:9.3f} {calls:6d} {avg:8.1f}  {name[:85]}")
    out = "\n".join(lines)
    open(out_path, "w").write(out + "\n")
    print(out)


if __name__ == "__main__":
    summarize(sys.argv[1], float(sys.argv[2]), sys.argv[3],
              sys.argv[4] if len(sys.argv) > 4 else "")
