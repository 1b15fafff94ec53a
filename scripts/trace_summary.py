"""Summarize a rocprofv3 rocpd results.db: per-kernel count/total/avg.
Usage: python scripts/trace_summary.py <results.db>"""
import sqlite3
import sys

db = sys.argv[1]
c = sqlite3.connect(db)
tables = [r[0] for r in c.execute(
    "SELECT name FROM sqlite_master WHERE type='table'").fetchall()]
kd = next(t for t in tables if t.startswith('rocpd_kernel_dispatch'))
sym = next(t for t in tables if t.startswith('rocpd_info_kernel_symbol'))
rows = c.execute(f"""
  SELECT s.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
         AVG(kd.end-kd.start)/1e3, MAX(s.arch_vgpr_count),
         MAX(s.group_segment_size)
  FROM {kd} kd JOIN {sym} s ON kd.kernel_id = s.id
  GROUP BY s.display_name ORDER BY 3 DESC LIMIT 20""").fetchall()
print(f"{'total ms':>9} {'n':>6} {'avg us':>8} {'vgpr':>5} {'lds':>7}  kernel")
for n, cnt, tot, avg, vgpr, lds in rows:
    print(f"{tot:9.2f} {cnt:6d} {avg:8.1f} {vgpr:5d} {lds:7d}  {n[:70]}")
print(f"sum: {sum(r[2] for r in rows):.1f} ms")
