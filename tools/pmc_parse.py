"""Aggregate per-kernel PMC counters from a rocprofv3 rocpd results.db.

Usage: python tools/pmc_parse.py gpurun_out/prof_attn4/attn_results.db
Prints one row per kernel: n dispatches, avg us, MFMA%, WAIT%, WAIT_INST%,
LDS CONF% (the derivations used throughout profiles/attn_pmc_r02.md):
  MFMA%  = SQ_VALU_MFMA_BUSY_CYCLES / (4 * SQ_WAVE_CYCLES)
  WAIT%  = SQ_WAIT_ANY / SQ_WAVE_CYCLES
  CONF%  = SQ_LDS_BANK_CONFLICT / SQ_LDS_IDX_ACTIVE
"""
import sqlite3
import sys
from collections import defaultdict


def main(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    cur.execute("SELECT name FROM sqlite_master WHERE type='table' "
                "AND name LIKE 'rocpd_pmc_event%'")
    sfx = cur.fetchone()[0][len("rocpd_pmc_event"):]

    cur.execute(f"SELECT id, name FROM rocpd_info_pmc{sfx}")
    pmc_names = dict(cur.fetchall())
    cur.execute(f"SELECT id, display_name FROM rocpd_info_kernel_symbol{sfx}")
    ksym = dict(cur.fetchall())

    # dispatch event -> kernel, duration
    cur.execute(f"SELECT event_id, kernel_id, end - start "
                f"FROM rocpd_kernel_dispatch{sfx}")
    disp = {e: (k, dur) for e, k, dur in cur.fetchall()}

    agg = defaultdict(lambda: defaultdict(float))
    times = defaultdict(list)
    cur.execute(f"SELECT event_id, pmc_id, value FROM rocpd_pmc_event{sfx}")
    for ev, pmc, val in cur.fetchall():
        if ev not in disp:
            continue
        k, _ = disp[ev]
        agg[k][pmc_names.get(pmc, str(pmc))] += val
    for ev, (k, dur) in disp.items():
        times[k].append(dur)

    rows = []
    for k, counters in agg.items():
        name = ksym.get(k, str(k))
        name = name.replace("void ", "").replace("(anonymous namespace)::", "")
        name = name.split("(")[0]
        ts = times[k]
        wave = counters.get("SQ_WAVE_CYCLES", 0.0)
        if wave <= 0:
            continue
        rows.append((
            sum(ts) / 1e3,  # total us for sorting
            name, len(ts), sum(ts) / len(ts) / 1e3,
            100.0 * counters.get("SQ_VALU_MFMA_BUSY_CYCLES", 0.0) / (4 * wave),
            100.0 * counters.get("SQ_WAIT_ANY", 0.0) / wave,
            100.0 * counters.get("SQ_WAIT_INST_ANY", 0.0) / wave,
            100.0 * counters.get("SQ_LDS_BANK_CONFLICT", 0.0)
            / max(counters.get("SQ_LDS_IDX_ACTIVE", 0.0), 1.0),
        ))
    rows.sort(reverse=True)
    print(f"{'kernel':60s} {'n':>4s} {'avg_us':>8s} {'MFMA%':>6s} "
          f"{'WAIT%':>6s} {'WAITI%':>6s} {'CONF%':>6s}")
    for _, name, n, us, mfma, wait, wi, conf in rows:
        print(f"{name[:60]:60s} {n:4d} {us:8.1f} {mfma:6.1f} "
              f"{wait:6.1f} {wi:6.1f} {conf:6.1f}")


if __name__ == "__main__":
    main(sys.argv[1] if len(sys.argv) > 1 else
         "gpurun_out/prof_attn_final/attn_results.db")
