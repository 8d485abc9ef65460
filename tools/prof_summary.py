#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db: per-kernel total time, calls, % of GPU
time. Usage: python tools/prof_summary.py <results.db> [top_n]"""
import re
import sqlite3
import sys


def summarize(path, top=40):
    db = sqlite3.connect(path)
    tables = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    kd = next(t for t in tables if t.startswith("rocpd_kernel_dispatch"))
    sfx = kd[len("rocpd_kernel_dispatch_"):]
    q = f"""
      SELECT s.string, COUNT(*), SUM(k.end - k.start)
      FROM {kd} k
      JOIN rocpd_info_kernel_symbol_{sfx} ks
           ON ks.id = k.kernel_id AND ks.guid = k.guid
      JOIN rocpd_string_{sfx} s ON s.id = ks.display_name_id AND s.guid = ks.guid
      GROUP BY s.string ORDER BY SUM(k.end - k.start) DESC
    """
    try:
        rows = db.execute(q).fetchall()
    except sqlite3.OperationalError:
        # fallback: kernel_symbol may use different name column
        cols = [r[1] for r in db.execute(
            f"PRAGMA table_info(rocpd_info_kernel_symbol_{sfx})")]
        name_col = next(c for c in cols if "name" in c)
        q = f"""
          SELECT ks.{name_col}, COUNT(*), SUM(k.end - k.start)
          FROM {kd} k
          JOIN rocpd_info_kernel_symbol_{sfx} ks
               ON ks.id = k.kernel_id AND ks.guid = k.guid
          GROUP BY ks.{name_col} ORDER BY SUM(k.end - k.start) DESC
        """
        rows = db.execute(q).fetchall()
    total = sum(r[2] for r in rows) or 1
    out = []
    out.append(f"{'kernel':<80} {'calls':>7} {'total_ms':>10} {'%':>6}")
    for name, calls, ns in rows[:top]:
        short = re.sub(r"<[^>]*>", "<>", str(name))[:80]
        out.append(f"{short:<80} {calls:>7} {ns/1e6:>10.2f} "
                   f"{100.0*ns/total:>6.2f}")
    out.append(f"TOTAL GPU time: {total/1e6:.2f} ms over {len(rows)} kernels")
    return "\n".join(out)


if __name__ == "__main__":
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 40
    print(summarize(sys.argv[1], top))
