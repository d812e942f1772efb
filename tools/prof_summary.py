"""Summarize a rocprofv3 rocpd results.db into the profiles/ text format.

Usage (on the GPU box):
    python tools/prof_summary.py <results.db> <out.txt> <title> <n_execs> [bench_log]

The db is the `-o <tag>` output of `rocprofv3 --kernel-trace --stats`; the
summary is the `top_kernels` view (name, calls, total us, avg us) sorted by
total time, with the bench JSON line from the run's log quoted in the header.
"""

import json
import re
import sqlite3
import sys


def main() -> None:
    db, out, title, steps = sys.argv[1:5]
    steps = int(steps)
    bench = ""
    if len(sys.argv) > 5:
        log = open(sys.argv[5]).read()
        m = re.search(r"\{.*\}", log)
        if m:
            j = json.loads(m.group(0))
            bench = (f"# bench JSON (this profiled run): "
                     f"ms_per_step={j['ms_per_step']:.3f} "
                     f"value={j['value']:.0f} {j.get('unit', '')}\n")
    c = sqlite3.connect(db)
    rows = c.execute(
        "select name, total_calls, total_duration, average from top_kernels "
        "order by total_duration desc limit 34").fetchall()
    tot = c.execute("select sum(total_duration) from top_kernels").fetchone()[0]
    with open(out, "w") as f:
        f.write(f"# {title}\n")
        f.write(f"# rocprofv3 --kernel-trace --stats, MI355X (gfx950); "
                f"{steps} total fwd+bwd+opt executions profiled\n")
        f.write(bench)
        f.write(f"# total GPU kernel time: {tot / 1e3:.2f} ms -> "
                f"{tot / 1e3 / steps:.2f} ms/execution\n")
        f.write("  total_ms  calls    avg_us  kernel\n")
        for n, cnt, ms, us in rows:
            f.write(f"{ms / 1e3:10.3f} {cnt:6d} {us:9.1f}  {n[:170]}\n")
    print("wrote", out)


if __name__ == "__main__":
    main()
