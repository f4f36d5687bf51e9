#!/usr/bin/env python3
"""Summarize a rocprofv3 results DB into a markdown kernel-group table.

Workflow (on a GPU box):
  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --kernel-trace --stats -d OUT -o run -- python bench.py --steps 5
  python scripts/profile_step.py OUT/run_results.db --steps 7 > profile.md
"""
import argparse
import sqlite3


def group_of(name: str) -> str:
    if name.startswith(("Cijk", "Custom_Cijk")):
        return "GEMM (hipBLASLt)"
    for pat, g in [("attn_bwd", "attention bwd (HIP)"), ("preprocess", "attention bwd (HIP)"),
                   ("attn_fwd", "attention fwd (HIP)"), ("attn_decode", "decode attn (HIP)"),
                   ("gemv", "decode GEMV (HIP)"), ("rope", "rope (HIP)"),
                   ("swiglu", "swiglu (HIP)"), ("rmsnorm", "rmsnorm (HIP)"),
                   ("adamw", "fused optimizer (HIP)"), ("sumsq", "fused optimizer (HIP)"),
                   ("lion", "fused optimizer (HIP)"), ("sgd_", "fused optimizer (HIP)"),
                   ("ce_", "cross-entropy (HIP)"), ("fp8", "fp8 quantize (HIP)"),
                   ("cast_e4m3", "fp8 quantize (HIP)"), ("amax", "fp8 quantize (HIP)")]:
        if pat in name:
            return g
    return "torch elementwise/misc"


def main():
    p = argparse.ArgumentParser()
    p.add_argument("db")
    p.add_argument("--steps", type=int, default=1, help="profiled step count (for ms/step)")
    p.add_argument("--top", type=int, default=0, help="also list top-N kernels")
    a = p.parse_args()
    cur = sqlite3.connect(a.db).cursor()
    rows = cur.execute(
        "SELECT name, total_calls, total_duration, average FROM top_kernels "
        "ORDER BY total_duration DESC").fetchall()
    tot = sum(r[2] for r in rows)
    groups = {}
    for name, calls, dur, avg in rows:
        groups[group_of(name)] = groups.get(group_of(name), 0) + dur
    print(f"Total GPU kernel time: {tot/1e6:.3f} s over {a.steps} step(s) ({tot/a.steps/1e3:.1f} ms/step GPU-busy)\n")
    print("| group | ms (whole capture) | share |\n|---|---|---|")
    for g, d in sorted(groups.items(), key=lambda x: -x[1]):
        print(f"| {g} | {d/a.steps/1e3:.1f} | {100*d/tot:.1f}% |")
    if a.top:
        print("\n| kernel | calls | total ms | avg us |\n|---|---|---|---|")
        for name, calls, dur, avg in rows[:a.top]:
            print(f"| `{name[:90]}` | {calls} | {dur/1e3:.1f} | {avg:.0f} |")


if __name__ == "__main__":
    main()
