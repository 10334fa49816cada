#!/usr/bin/env python3
"""Summarize a rocprofv3 results.db into a small markdown table.
Run ON the GPU box so only the digest returns through gpurun_out."""
import re
import sqlite3
import sys

db_path, out_path = sys.argv[1], sys.argv[2]
steps = sys.argv[3] if len(sys.argv) > 3 else "?"
window_ms = float(sys.argv[4]) if len(sys.argv) > 4 else None
db = sqlite3.connect(db_path)
cur = db.cursor()
tables = [r[0] for r in cur.execute(
    "SELECT name FROM sqlite_master WHERE type='table'").fetchall()]
ks = next(t for t in tables if t.startswith('rocpd_info_kernel_symbol'))
kd = next(t for t in tables if t.startswith('rocpd_kernel_dispatch'))
where = ""
if window_ms is not None:
    # steady-state only: dispatches in the last `window_ms` of the run
    # (excludes MIOpen exhaustive-find warmup kernels)
    max_end = cur.execute(f"SELECT MAX(end) FROM {kd}").fetchone()[0]
    where = f"WHERE d.start >= {max_end - window_ms * 1e6}"
rows = cur.execute(f"""
SELECT s.display_name, COUNT(*), SUM(d.end-d.start)/1e6, AVG(d.end-d.start)/1e3
FROM {kd} d JOIN {ks} s ON d.kernel_id = s.id
{where}
GROUP BY s.display_name ORDER BY SUM(d.end-d.start) DESC""").fetchall()
total = sum(r[2] for r in rows)
with open(out_path, 'w') as f:
    f.write(f"# rocprofv3 kernel summary ({steps} executed steps"
            + (f", last {window_ms:.0f} ms window" if window_ms else "") + ")\n\n")
    f.write(f"Total GPU kernel time: {total:.1f} ms\n\n")
    f.write("| ms | % | calls | avg us | kernel |\n|---:|---:|---:|---:|---|\n")
    for name, calls, ms, avg in rows[:50]:
        n = re.sub(r'[|\n]', ' ', name)[:110]
        f.write(f"| {ms:.2f} | {100*ms/total:.1f} | {calls} | {avg:.1f} | `{n}` |\n")
print(f"wrote {out_path}: {total:.1f} ms total")
