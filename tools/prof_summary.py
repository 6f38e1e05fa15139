"""Summarize a rocprofv3 rocpd results.db (kernel-trace --stats or --pmc).

Usage: python tools/prof_summary.py <results.db> [trace|pmc]
Prints the per-kernel duration table (trace) or the per-kernel counter
summary with derived MFMA-pipe / wait / L2-hit figures (pmc).
"""
import sqlite3
import sys


def guid_of(c):
    row = c.execute(
        "select name from sqlite_master where type='table' "
        "and name like 'rocpd_kernel_dispatch_%'").fetchone()
    return row[0][len("rocpd_kernel_dispatch_"):]


def kname(nm):
    return nm.split("(")[0].strip()


def trace_summary(db):
    c = sqlite3.connect(db)
    g = guid_of(c)
    q = f"""
      select ks.display_name, count(*), sum(kd.end-kd.start)/1e6,
             avg(kd.end-kd.start)/1e3
      from rocpd_kernel_dispatch_{g} kd
      join rocpd_info_kernel_symbol_{g} ks on kd.kernel_id = ks.id
      group by ks.display_name order by 3 desc
    """
    print(f"{'total_ms':>9} {'n':>5} {'avg_us':>8}  name")
    for name, n, tot, avg in c.execute(q):
        print(f"{tot:9.3f} {n:5d} {avg:8.1f}  {name[:84]}")


def pmc_summary(db):
    c = sqlite3.connect(db)
    g = guid_of(c)
    q = f"""
      select ks.display_name, pi.name, sum(pe.value)
      from rocpd_pmc_event_{g} pe
      join rocpd_info_pmc_{g} pi on pe.pmc_id = pi.id
      join rocpd_kernel_dispatch_{g} kd on pe.event_id = kd.event_id
      join rocpd_info_kernel_symbol_{g} ks on kd.kernel_id = ks.id
      group by ks.display_name, pi.name
    """
    agg = {}
    for name, ctr, val in c.execute(q):
        agg.setdefault(kname(name), {})[ctr] = val
    for name, d in sorted(agg.items(),
                          key=lambda kv: -kv[1].get("SQ_WAVE_CYCLES", 0)):
        wc = d.get("SQ_WAVE_CYCLES", 0)
        if not wc:
            continue
        mfma = d.get("SQ_VALU_MFMA_BUSY_CYCLES", 0)
        wait = d.get("SQ_WAIT_ANY", 0)
        wi = d.get("SQ_WAIT_INST_ANY", 0)
        hit, miss = d.get("TCC_HIT_sum", 0), d.get("TCC_MISS_sum", 0)
        # SQ_WAVE_CYCLES counts in quad-cycles on CDNA; MFMA pipe max is
        # 50% of wave-cycles x4 at 2 waves/SIMD (MI355X_MICROARCH.md)
        print(f"{name[:70]}")
        print(f"  MFMApipe {100*mfma/(wc*4):5.1f}% (max 50% at 2 waves/SIMD)"
              f"  WAIT {100*wait/(wc*4):5.1f}%  WAIT_INST {100*wi/(wc*4):5.1f}%"
              f"  L2hit {100*hit/max(1, hit+miss):5.1f}%")


if __name__ == "__main__":
    db = sys.argv[1]
    mode = sys.argv[2] if len(sys.argv) > 2 else "trace"
    (trace_summary if mode == "trace" else pmc_summary)(db)
