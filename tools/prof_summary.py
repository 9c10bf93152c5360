#!/usr/bin/env python3
"""Summarize a rocprofv3 results .db (kernel-trace --stats) into a markdown
table: python tools/prof_summary.py <results.db> <out.md> [steps]"""

import sqlite3
import sys


def main(dbfile, outfile, steps=5):
    db = sqlite3.connect(dbfile)
    cur = db.cursor()
    sfx = [r[0] for r in cur.execute(
        "select name from sqlite_master where type='table' "
        "and name like 'rocpd_kernel_dispatch%'")][0].replace(
            "rocpd_kernel_dispatch_", "")
    rows = cur.execute(f"""
      select ks.display_name, count(*), sum(kd.end-kd.start)/1e6
      from rocpd_kernel_dispatch_{sfx} kd
      join rocpd_info_kernel_symbol_{sfx} ks on ks.id = kd.kernel_id
      group by ks.display_name order by 3 desc
    """).fetchall()
    total = sum(r[2] for r in rows)
    with open(outfile, "w") as fh:
        fh.write(f"# Kernel-time breakdown ({dbfile.split('/')[-1]}, "
                 f"{steps} steps incl. warmup)\n\n")
        fh.write(f"Total GPU kernel time: {total:.1f} ms "
                 f"({total/steps:.2f} ms/step)\n\n")
        fh.write("| ms/step | calls | % | kernel |\n|---|---|---|---|\n")
        for name, cnt, ms in rows[:30]:
            short = name.split("(")[0].replace("void ", "")[:80]
            fh.write(f"| {ms/steps:.3f} | {cnt} | {100*ms/total:.1f} | `{short}` |\n")
    print(f"wrote {outfile}")


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2], int(sys.argv[3]) if len(sys.argv) > 3 else 5)
