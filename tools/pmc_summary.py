"""Summarize a rocprofv3 --pmc rocpd db into per-kernel counter rates.

Usage (on the GPU box):
    python tools/pmc_summary.py <results.db> <out.txt> <title>

For each kernel name aggregates over its dispatches:
  - mfma% : SQ_VALU_MFMA_BUSY_CYCLES / (duration * 2.4 GHz * 1024 SIMDs)
  - ldsc% : SQ_LDS_BANK_CONFLICT (stall cycles) on the same basis
  - waves : SQ_WAVE_CYCLES (quad-cycles, per-SE aggregate) -- occupancy proxy
mfma% is approximate (clock estimate) but comparable across kernels of the
same run, which is all the routing decisions need.
"""

import collections
import sqlite3
import sys

GHZ = 2.4
SIMDS = 1024.0  # 256 CU x 4


def main() -> None:
    db, out, title = sys.argv[1:4]
    c = sqlite3.connect(db)
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    dur = collections.defaultdict(float)
    calls = collections.defaultdict(set)
    for name, counter, value, duration, did in c.execute(
            "select kernel_name, counter_name, value, duration, dispatch_id "
            "from counters_collection"):
        agg[name][counter] += value
        calls[name].add(did)
    for name, duration in c.execute(
            "select kernel_name, sum(duration) from counters_collection "
            "where counter_name='SQ_WAVE_CYCLES' group by kernel_name"):
        dur[name] = duration  # ns, one row per dispatch per counter
    rows = []
    for name, d in agg.items():
        ns = dur[name]
        if ns <= 0:
            continue
        denom = ns * GHZ * SIMDS
        rows.append((ns, len(calls[name]),
                     100.0 * d.get("SQ_VALU_MFMA_BUSY_CYCLES", 0.0) / denom,
                     100.0 * d.get("SQ_LDS_BANK_CONFLICT", 0.0) / denom,
                     name))
    rows.sort(reverse=True)
    with open(out, "w") as f:
        f.write(f"# {title}\n")
        f.write("# rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES "
                "SQ_LDS_BANK_CONFLICT --kernel-trace\n")
        f.write("# mfma%/ldsc% normalized by duration*2.4GHz*1024 SIMDs "
                "(approximate, run-relative)\n")
        f.write("  total_ms  calls  mfma%  ldsc%  kernel\n")
        for ns, n, mfma, lds, name in rows[:30]:
            f.write(f"{ns / 1e6:10.3f} {n:6d} {mfma:6.1f} {lds:6.2f}  "
                    f"{name[:150]}\n")
    print("wrote", out)


if __name__ == "__main__":
    main()
