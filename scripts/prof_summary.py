"""Summarize a rocprofv3 rocpd SQLite DB into a text kernel table
(steady-state window), so only the summary travels back from GPU boxes.

  python scripts/prof_summary.py <dir-with-db> <window_ms> [out.txt]
"""

import glob
import sqlite3
import sys


def summarize(db_dir: str, window_ms: float, out_path: str = "") -> str:
    paths = sorted(glob.glob(db_dir + "/**/*.db", recursive=True))
    if not paths:
        return "no rocpd db under %s" % db_dir
    db = sqlite3.connect(paths[-1])
    cur = db.cursor()
    t = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")
        if "kernel_dispatch" in r[0]][0]
    u = t.replace("rocpd_kernel_dispatch_", "")
    (tmin, tmax), = cur.execute(f"SELECT MIN(start), MAX(end) FROM {t}")
    w0 = tmax - window_ms * 1e6
    q = f"""SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6,
                   AVG(d.end-d.start)/1e3
            FROM {t} d JOIN rocpd_info_kernel_symbol_{u} s
                 ON d.kernel_id = s.id
            WHERE d.start > {w0} GROUP BY s.display_name ORDER BY 3 DESC"""
    rows = list(cur.execute(q))
    total = sum(r[2] for r in rows)
    lines = ["steady-state window %.0f ms, GPU busy %.1f ms (%.0f%%)"
             % (window_ms, total, 100 * total / window_ms)]
    for name, calls, ms, avg in rows[:30]:
        lines.append("%8.2f ms %6d calls %8.1f us  %5.1f%%  %s"
                     % (ms, calls, avg, 100 * ms / total, name[:100]))
    text = "\n".join(lines) + "\n"
    if out_path:
        with open(out_path, "w") as f:
            f.write(text)
    return text


if __name__ == "__main__":
    d = sys.argv[1]
    win = float(sys.argv[2]) if len(sys.argv) > 2 else 200.0
    out = sys.argv[3] if len(sys.argv) > 3 else ""
    print(summarize(d, win, out))
