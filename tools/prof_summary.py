#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd .db: per-kernel totals + memcpy totals +
wall coverage. Usage: prof_summary.py results.db [out.md]"""
import sqlite3
import sys


def main():
    if len(sys.argv) < 2:
        print(__doc__.strip(), file=sys.stderr)
        sys.exit(2)
    path = sys.argv[1]
    db = sqlite3.connect(path)
    cur = db.cursor()
    out = []
    rows = cur.execute("""
      SELECT ks.display_name, COUNT(*), SUM(kd.end-kd.start)/1e6,
             AVG(kd.end-kd.start)/1e3
      FROM rocpd_kernel_dispatch kd
      JOIN rocpd_info_kernel_symbol ks ON kd.kernel_id=ks.id
      GROUP BY ks.display_name ORDER BY 3 DESC LIMIT 20""").fetchall()
    out.append("| kernel | calls | total ms | avg us |")
    out.append("|---|---|---|---|")
    for name, n, tot, avg in rows:
        out.append(f"| {name[:70]} | {n} | {tot:.2f} | {avg:.1f} |")
    try:
        mc = cur.execute("""
          SELECT s.string, COUNT(*), SUM(m.end-m.start)/1e6
          FROM rocpd_memory_copy m JOIN rocpd_string s ON m.name_id=s.id
          GROUP BY s.string""").fetchall()
        out.append("")
        out.append("| memcpy kind | calls | total ms |")
        out.append("|---|---|---|")
        for name, n, tot in mc:
            out.append(f"| {name} | {n} | {tot:.2f} |")
    except Exception as e:
        out.append(f"(memcpy table: {e})")
    # busy vs wall
    span = cur.execute("SELECT (MAX(end)-MIN(start))/1e6, SUM(end-start)/1e6 "
                       "FROM rocpd_kernel_dispatch").fetchone()
    out.append("")
    out.append(f"kernel-span wall: {span[0]:.1f} ms; busy: {span[1]:.1f} ms "
               f"({100*span[1]/max(span[0],1e-9):.1f}% of span)")
    text = "\n".join(out)
    print(text)
    if len(sys.argv) > 2:
        with open(sys.argv[2], "w") as f:
            f.write(text + "\n")


main()
