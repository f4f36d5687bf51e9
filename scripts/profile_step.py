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
    p.add_argument("--dist", action="store_true",
                   help="report RCCL comm time + overlap-with-compute share")
    a = p.parse_args()
    if a.dist:
        dist_overlap(a.db)
        return
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




def dist_overlap(db: str) -> None:
    """--dist mode: RCCL/comm kernel time and its overlap with compute.

    Evidence hook for the DP bucket-overlap design (parallel/ddp.py): run
    bench.py under rocprofv3 --kernel-trace on a multi-GPU box, then this
    reports how much of the RCCL all-reduce time was hidden under backward
    compute (overlapped share ~1.0 = fully hidden).
    """
    cur = sqlite3.connect(db).cursor()
    rows = cur.execute(
        "SELECT kd.start, kd.end, ks.display_name FROM rocpd_kernel_dispatch kd "
        "JOIN rocpd_info_kernel_symbol ks ON kd.kernel_id = ks.id "
        "ORDER BY kd.start").fetchall()
    comm, comp = [], []
    for s, e, name in rows:
        low = name.lower()
        (comm if ("rccl" in low or "nccl" in low or "allreduce" in low or
                  "reducescatter" in low or "allgather" in low or
                  "alltoall" in low or "sendrecv" in low) else comp).append((s, e))
    if not comm:
        print("no RCCL kernels in this capture (single-GPU run?)")
        return
    comp.sort()
    merged = []
    for s, e in comp:
        if merged and s <= merged[-1][1]:
            merged[-1] = (merged[-1][0], max(merged[-1][1], e))
        else:
            merged.append((s, e))
    import bisect
    starts = [m[0] for m in merged]
    tot = sum(e - s for s, e in comm)
    hidden = 0
    for s, e in comm:
        i = bisect.bisect_right(starts, s) - 1
        j = bisect.bisect_right(starts, e)
        for k in range(max(i, 0), j):
            ms, me = merged[k]
            hidden += max(0, min(e, me) - max(s, ms))
    comp_tot = sum(e - s for s, e in merged)
    print(f"comm kernels: {len(comm)}  comm time: {tot/1e6:.2f} ms")
    print(f"compute busy: {comp_tot/1e6:.2f} ms")
    print(f"comm overlapped with compute: {hidden/1e6:.2f} ms ({100.0*hidden/tot:.1f}% hidden)")


if __name__ == "__main__":
    main()
