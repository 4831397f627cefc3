"""Summarize per-kernel PMC sums from a rocprofv3 results.db.

    python benchmarks/pmc_summary.py <db> [name-regex]
"""
import re
import sqlite3
import sys

db = sys.argv[1]
pat = sys.argv[2] if len(sys.argv) > 2 else r"attn_\w+|linear\w+|bn_\w+"
c = sqlite3.connect(db)
sfx = None
for (n,) in c.execute("select name from sqlite_master where type='table'"):
    m = re.match(r"rocpd_pmc_event_(.*)", n)
    if m:
        sfx = m.group(1)
rows = list(c.execute(f"""
  select ks.display_name, i.name, sum(p.value)
  from rocpd_pmc_event_{sfx} p
  join rocpd_info_pmc_{sfx} i on p.pmc_id = i.id
  join rocpd_kernel_dispatch_{sfx} k on p.event_id = k.event_id
  join rocpd_info_kernel_symbol_{sfx} ks on k.kernel_id = ks.id
  group by ks.display_name, i.name"""))
agg = {}
for kn, cn, v in rows:
    m = re.search(pat, kn)
    if not m:
        continue
    agg.setdefault(m.group(0), {})[cn] = v
for kn, d in sorted(agg.items()):
    wc = d.get("SQ_WAVE_CYCLES", 1)
    parts = [kn]
    for cn, v in sorted(d.items()):
        if cn == "SQ_WAVE_CYCLES":
            parts.append(f"WAVE_CYCLES={v:.3e}")
        else:
            parts.append(f"{cn.replace('SQ_', '')}={v / wc * 100:.1f}%")
    print("  ".join(parts))
