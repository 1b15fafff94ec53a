"""Summarize a rocprofv3 --pmc rocpd db: per-kernel wave-state split.
Usage: python scripts/pmc_summary.py <results.db>"""
import collections
import sqlite3
import sys

c = sqlite3.connect(sys.argv[1])
tables = [r[0] for r in c.execute(
    "SELECT name FROM sqlite_master WHERE type='table'").fetchall()]
kd = next(t for t in tables if t.startswith('rocpd_kernel_dispatch'))
sym = next(t for t in tables if t.startswith('rocpd_info_kernel_symbol'))
pmc = next(t for t in tables if t.startswith('rocpd_pmc_event'))
pin = next(t for t in tables if t.startswith('rocpd_info_pmc'))
rows = c.execute(f"""
  SELECT s.display_name, p.name, SUM(pe.value)
  FROM {pmc} pe
  JOIN {kd} kd ON pe.event_id = kd.event_id
  JOIN {sym} s ON kd.kernel_id = s.id
  JOIN {pin} p ON pe.pmc_id = p.id
  GROUP BY s.display_name, p.name""").fetchall()
agg = collections.defaultdict(dict)
for n, cn, v in rows:
    agg[n[:58]][cn] = v
hdr = ('kernel', 'active%', 'memwait%', 'instwait%')
print(f"{hdr[0]:60s} {hdr[1]:>8} {hdr[2]:>9} {hdr[3]:>9}")
for n, d in sorted(agg.items(),
                   key=lambda kv: -kv[1].get('SQ_WAVE_CYCLES', 0))[:12]:
    wc = d.get('SQ_WAVE_CYCLES', 1) or 1
    a = 100 * d.get('SQ_ACTIVE_INST_ANY', 0) / wc
    w = 100 * d.get('SQ_WAIT_ANY', 0) / wc
    i = 100 * d.get('SQ_WAIT_INST_ANY', 0) / wc
    print(f"{n:60s} {a:8.1f} {w:9.1f} {i:9.1f}")
