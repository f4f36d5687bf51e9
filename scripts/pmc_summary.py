#!/usr/bin/env python3
"""Summarize a rocprofv3 --pmc results DB: per-kernel mean counter values.

  rocprofv3 --pmc MfmaUtil VALUBusy SQ_WAIT_ANY SQ_WAIT_INST_ANY \
            SQ_WAVE_CYCLES -d OUT -o run -- python scripts/bench_attention.py
  python scripts/pmc_summary.py OUT/run_results.db
"""
import argparse
import sqlite3
from collections import defaultdict


def main():
    p = argparse.ArgumentParser()
    p.add_argument("db")
    p.add_argument("--like", default="", help="substring filter on kernel name")
    a = p.parse_args()
    cur = sqlite3.connect(a.db).cursor()
    rows = cur.execute(
        "SELECT ks.display_name, pi.name, pe.value "
        "FROM rocpd_pmc_event pe "
        "JOIN rocpd_info_pmc pi ON pe.pmc_id = pi.id "
        "JOIN rocpd_kernel_dispatch kd ON pe.event_id = kd.event_id "
        "JOIN rocpd_info_kernel_symbol ks ON kd.kernel_id = ks.id").fetchall()
    acc = defaultdict(lambda: defaultdict(lambda: [0.0, 0]))
    counters = []
    for kname, cname, val in rows:
        if a.like and a.like not in kname:
            continue
        s = acc[kname.replace("(anonymous namespace)::", "").split("(")[0][:70]][cname]
        s[0] += val
        s[1] += 1
        if cname not in counters:
            counters.append(cname)
    # derive WAIT shares when the SQ trio is present
    print("| kernel | " + " | ".join(counters) + " | waitA% | waitI% |")
    print("|---" * (len(counters) + 3) + "|")
    for kname, cs in sorted(acc.items()):
        vals = [cs[c][0] / max(cs[c][1], 1) for c in counters]
        wc = cs.get("SQ_WAVE_CYCLES", [0, 0])[0] / max(cs.get("SQ_WAVE_CYCLES", [0, 1])[1], 1)
        wa = cs.get("SQ_WAIT_ANY", [0, 0])[0] / max(cs.get("SQ_WAIT_ANY", [0, 1])[1], 1)
        wi = cs.get("SQ_WAIT_INST_ANY", [0, 0])[0] / max(cs.get("SQ_WAIT_INST_ANY", [0, 1])[1], 1)
        shares = f" {100*wa/wc:.0f} | {100*wi/wc:.0f} |" if wc else " - | - |"
        print(f"| `{kname}` | " + " | ".join(f"{v:,.1f}" for v in vals) + " |" + shares)


if __name__ == "__main__":
    main()
