#!/usr/bin/env python3
"""LDS bank-conflict simulator for gfx950 (dword banks) — the tool that found
the `s_tr` swizzle used by the attention kernels (csrc/attn_common.h).

Models (calibrated against rocprofv3 SQ_LDS_BANK_CONFLICT on MI355X):
  ds_read_b128        : bank class (addr/4) % 64, 16-lane quarter-wave groups
  ds_read_b64_tr_b16  : bank class (addr/4) % 64, 16-lane transpose groups
  ds_write_b128       : bank class (addr/4) % 32 (writes use 32-bank classes)

Run it to re-verify the attention-image constraints for D=64/128:
  A. S-phase b128 row reads conflict-free  <=> s_tr bijective on every
     16-row window,
  B. tr16 [4 row][16 col] gathers conflict-free <=> s_tr>>1 distinct on every
     aligned 4-row window.
"""


def conflicts(addrs_bytes, width_bytes, nbanks, group):
    worst = 1
    for g in range(0, len(addrs_bytes), group):
        banks = {}
        for a in addrs_bytes[g:g + group]:
            for w in range(0, width_bytes, 4):
                banks.setdefault(((a + w) // 4) % nbanks, set()).add(a + w)
        worst = max(worst, max((len(v) for v in banks.values()), default=1))
    return worst  # 1 = conflict-free


def s_tr(D, q):
    if D >= 128:
        return ((q & 7) << 1) | ((q >> 3) & 1)
    return (((q >> 1) & 3) << 1) | ((q >> 3) & 1)


def addr(D, q, d):
    """Byte address of element (q, d) in the tr-compatible row-major image."""
    return 2 * (q * D + (((d & ~7) ^ (s_tr(D, q) << 3)) | (d & 7)))


def check(D):
    # A: S-phase b128 reads (lane l: row l&31, 8 cols at dblk*16 + (l>>5)*8)
    w = 1
    for dblk in range(D // 16):
        a = [addr(D, l & 31, dblk * 16 + (l >> 5) * 8) for l in range(64)]
        w = max(w, conflicts(a, 16, 64, 16))
    print(f"D={D} S-phase b128 reads : worst {w}-way")

    # B: tr16 gathers (group reads window [qb..qb+3][d0..d0+15]; source lane
    #    i reads 4 contiguous cols of row qb + (i>>2))
    w, aligned = 1, True
    for ks in range(2):
        for dc in range(D // 32):
            for rd in range(2):
                for g in range(4):
                    hi, gd = g >> 1, g & 1
                    qb = ks * 16 + hi * 8 + rd * 4
                    d0 = dc * 32 + gd * 16
                    a = [addr(D, qb + (i >> 2), d0 + 4 * (i & 3)) for i in range(16)]
                    aligned &= all(x % 8 == 0 for x in a)
                    w = max(w, conflicts(a, 8, 64, 16))
    print(f"D={D} tr16 gathers       : worst {w}-way, 8B-aligned={aligned}")

    # staging writes (b128, 32-bank classes, 2-row chunk pairs)
    w = 1
    for half in range(2):
        a = [addr(D, (u // (D // 8)) * 2 + half, (u % (D // 8)) * 8)
             for u in range(2 * (D // 8) * 8)]
        w = max(w, conflicts(a, 16, 32, 16))
    print(f"D={D} staging b128 writes: worst {w}-way (2-way is the floor: "
          f"column pairs 64 elements apart alias mod 32 banks)")


if __name__ == "__main__":
    for D in (64, 128):
        check(D)
