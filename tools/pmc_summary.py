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
    # every counter present in the db becomes a column, normalized by
    # SQ_WAVE_CYCLES (the wave-parked/issue-stall split needs WAIT_ANY /
    # WAIT_INST_ANY relative to wave cycles, not wall duration)
    counters = sorted({cn for d in agg.values() for cn in d} - {"SQ_WAVE_CYCLES"})
    rows = []
    for name, d in agg.items():
        ns = dur[name]
        if ns <= 0:
            continue
        denom = ns * GHZ * SIMDS
        wave = d.get("SQ_WAVE_CYCLES", 0.0) or 1.0
        cols = []
        for cn in counters:
            if cn == "SQ_VALU_MFMA_BUSY_CYCLES":
                cols.append(100.0 * d.get(cn, 0.0) / denom)   # % of wall
            else:
                cols.append(100.0 * d.get(cn, 0.0) / wave)    # % of wave cyc
        rows.append((ns, len(calls[name]), cols, name))
    rows.sort(reverse=True)
    short = {"SQ_VALU_MFMA_BUSY_CYCLES": "mfma%w", "SQ_LDS_BANK_CONFLICT": "ldsc",
             "SQ_WAIT_ANY": "parked", "SQ_WAIT_INST_ANY": "isstal",
             "SQ_ACTIVE_INST_ANY": "active"}
    with open(out, "w") as f:
        f.write(f"# {title}\n")
        f.write("# mfma%w = MFMA busy / wall (2.4GHz x 1024 SIMDs); other "
                "counters as % of SQ_WAVE_CYCLES (quad-cycles)\n")
        hdr = " ".join(f"{short.get(cn, cn[:6]):>7}" for cn in counters)
        f.write(f"  total_ms  calls {hdr}  kernel\n")
        for ns, n, cols, name in rows[:30]:
            cs = " ".join(f"{v:7.1f}" for v in cols)
            f.write(f"{ns / 1e6:10.3f} {n:6d} {cs}  {name[:140]}\n")
    print("wrote", out)


if __name__ == "__main__":
    main()
