#!/usr/bin/env python3
"""Run bench.py under rocprofv3 kernel tracing and emit a markdown kernel
table (the profiles/r02_final_flagship_kernels.md format).

Usage (on a GPU box):
  python tools/profile_bench.py --pipeline full --steps 2 --warmup 1 \
      --out profiles/kernels.md

PMC note: collect counters in a SEPARATE run with --pmc only (rocprofv3
crashes when PMC is combined with trace domains; wrap PMC runs in
`timeout` — replay passes can hang).
"""
import argparse
import glob
import os
import re
import sqlite3
import subprocess
import sys
import tempfile


def label(name):
    m = re.search(r"(\w+_kernel)", name)
    base = m.group(1) if m else name[:40]
    targs = re.search(r"<([^()]*)>", name)
    return base + (("<" + targs.group(1)[:28] + ">") if targs else "")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--pipeline", default="full")
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--out", default=None)
    ap.add_argument("--top", type=int, default=16)
    args = ap.parse_args()

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    work = tempfile.mkdtemp(prefix="profbench_")
    env = dict(os.environ)
    env["TMPDIR"] = work
    prefix = os.path.join(work, "prof")
    cmd = ["rocprofv3", "--kernel-trace", "--stats", "-o", prefix, "--",
           sys.executable, os.path.join(repo, "bench.py"),
           "--pipeline", args.pipeline, "--steps", str(args.steps),
           "--warmup", str(args.warmup)]
    r = subprocess.run(cmd, cwd=work, env=env, capture_output=True,
                      text=True, timeout=1800)
    if r.returncode != 0:
        sys.exit(f"bench under rocprofv3 failed:\n{r.stderr[-2000:]}")

    dbs = glob.glob(prefix + "*results.db")
    if not dbs:
        sys.exit("no rocprofv3 results.db produced")
    db = sqlite3.connect(dbs[0])
    cur = db.cursor()
    sfx = [t[0] for t in cur.execute(
        "select name from sqlite_master where name like "
        "'rocpd_kernel_dispatch%'")][0].replace("rocpd_kernel_dispatch_", "")
    rows = list(cur.execute(f"""
        select s.display_name, count(*), sum(k.end-k.start)/1e6,
               avg(k.end-k.start)/1e3
        from rocpd_kernel_dispatch_{sfx} k
        join rocpd_info_kernel_symbol_{sfx} s on k.kernel_id = s.id
        group by s.display_name order by 3 desc limit {args.top}"""))
    mn, mx, tot = list(cur.execute(
        f"select min(start), max(end), sum(end-start) "
        f"from rocpd_kernel_dispatch_{sfx}"))[0]

    lines = [f"# {args.pipeline} kernel table",
             "",
             f"rocprofv3 --kernel-trace --stats, {args.steps} steps + "
             f"{args.warmup} warmup.",
             f"Wall kernel span {(mx-mn)/1e9:.2f} s, busy-sum "
             f"{tot/1e9:.2f} s.",
             "",
             "| total ms | calls | avg us | kernel |",
             "|---|---|---|---|"]
    for name, n, ms, us in rows:
        lines.append(f"| {ms:.2f} | {n} | {us:.1f} | {label(name)} |")
    text = "\n".join(lines) + "\n"
    if args.out:
        with open(args.out, "w") as f:
            f.write(text)
    print(text)


if __name__ == "__main__":
    main()
