"""Summarize a rocprofv3 results.db into a per-kernel text table.

    python benchmarks/kstats.py <results.db> <out.txt>
"""

import re
import sqlite3
import sys


def main(db_path, out_path):
    c = sqlite3.connect(db_path)
    sfx = None
    for (name,) in c.execute("select name from sqlite_master where type='table'"):
        m = re.match(r"rocpd_kernel_dispatch_(.*)", name)
        if m:
            sfx = m.group(1)
            break
    assert sfx, "no kernel dispatch table"
    rows = list(c.execute(f"""
        select ks.display_name, count(*), sum(k.end-k.start) t, avg(k.end-k.start)
        from rocpd_kernel_dispatch_{sfx} k
        join rocpd_info_kernel_symbol_{sfx} ks on k.kernel_id = ks.id
        group by ks.display_name order by t desc limit 40"""))
    alltot = c.execute(
        f"select sum(end-start) from rocpd_kernel_dispatch_{sfx}").fetchone()[0]
    out = [f"# all kernel time: {alltot/1e6:.1f} ms ({db_path})"]
    for n, calls, t, a in rows:
        nm = n.replace("(anonymous namespace)::", "")
        nm = re.sub(r"^void ", "", nm)[:110]
        out.append(f"{t/alltot*100:5.2f}%  {calls:5d}  {a/1e3:9.1f}us  {nm}")
    with open(out_path, "w") as fh:
        fh.write("\n".join(out) + "\n")
    print("\n".join(out[:25]))


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
