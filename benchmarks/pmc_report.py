"""Summarize a rocprofv3 --pmc rocpd DB: per-kernel counter totals joined
with dispatch times. Usage: python benchmarks/pmc_report.py <db> [<db2> ...]"""
import sqlite3
import sys


def report(path):
    db = sqlite3.connect(path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = [t for t in tabs if t.startswith("rocpd_pmc_event")][0][
        len("rocpd_pmc_event"):]
    q = f"""
    SELECT s.display_name, i.name, SUM(e.value), COUNT(*)
    FROM rocpd_pmc_event{sfx} e
    JOIN rocpd_info_pmc{sfx} i ON e.pmc_id = i.id
    JOIN rocpd_kernel_dispatch{sfx} d ON e.event_id = d.event_id
    JOIN rocpd_info_kernel_symbol{sfx} s ON d.kernel_id = s.id
    GROUP BY s.display_name, i.name
    """
    data = {}
    for kname, cname, val, cnt in cur.execute(q):
        k = kname.replace("(anonymous namespace)::", "").split("(")[0]
        k = k.replace("void ", "").replace("__hip_bfloat16", "bf16")
        data.setdefault(k, {})[cname] = val
    print(f"== {path}")
    keys = sorted({c for v in data.values() for c in v})
    print(f"{'kernel':62s} " + " ".join(f"{k[3:]:>16s}" for k in keys))
    for k, v in sorted(data.items()):
        if "at::native" in k or "elementwise" in k.lower():
            continue
        row = " ".join(f"{v.get(c, 0):16.3e}" for c in keys)
        print(f"{k[:62]:62s} {row}")
    # derived: % of wave cycles
    for k, v in sorted(data.items()):
        wc = v.get("SQ_WAVE_CYCLES")
        if wc:
            act = v.get("SQ_ACTIVE_INST_ANY", 0) / wc
            wa = v.get("SQ_WAIT_ANY", 0) / wc
            wi = v.get("SQ_WAIT_INST_ANY", 0) / wc
            print(f"  {k[:58]:58s} active={act:5.1%} parked={wa:5.1%} "
                  f"issue-stall={wi:5.1%}")
        lds = v.get("SQ_LDS_IDX_ACTIVE")
        if lds is not None and "SQ_VALU_MFMA_BUSY_CYCLES" in v:
            print(f"  {k[:58]:58s} lds_cyc={lds:.3e} "
                  f"conflict={v.get('SQ_LDS_BANK_CONFLICT',0):.3e} "
                  f"mfma_cyc={v['SQ_VALU_MFMA_BUSY_CYCLES']:.3e}")


for p in sys.argv[1:]:
    report(p)
