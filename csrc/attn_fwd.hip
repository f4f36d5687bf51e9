// Tiled FlashAttention-2 forward (kernel K1/K2).
// Replaces the reference's naive softmax(QK^T)V Python compositions
// (/root/reference/models/attention/flash_attention.py:78-156 — "Simple
// approach without tiling for now" — and simple_attention.py, flex_attention.py).
//
// Structure (gfx950), following the guide's 8-wave attention ladder:
//   - NW waves/block, each wave owns 32 q rows (8 waves = 256 q rows share
//     every K/V tile -> staging traffic amortized 8x),
//   - causal CU load balance: tile_map() rotates the q-tile index across a
//     CU's successive blocks and (head-major layouts) pins each (b,h)'s
//     blocks to one XCD (attn_common.h; fwd 433 -> 569 TF at the 1B shape),
//   - cross-tile software pipeline: PV lags QK^T by one tile in a 3-slot
//     LDS ring, so tile j's exp/psum/pack VALU shares a basic block (and
//     the issue gaps) with tile j-1's PV MFMAs; branchless softmax tail,
//   - KVB-row K/V tiles, ONE barrier per tile,
//   - XOR-swizzled images (guide §6 Guideline 4: row-major D=128 bf16 read
//     column-wise by a lane group is an up-to-16-way bank conflict, and +1
//     padding does NOT fix it): K row-major [KVB][D] with element column
//     col ^ ((row&15)<<3); V transposed [D][KVB] with col ^ ((drow&7)<<3),
//   - T14 async-stage split: tile t+1's global loads ISSUE before tile t's
//     MFMA work, the LDS write lands right after QK^T (vmcnt hidden under
//     the MFMAs; guide §6 Guideline 15),
//   - swapped QK^T (attn_common.h) keeps the online softmax lane-local,
//     exponentials in base-2 with log2(e) folded into the scale,
//   - interior causal tiles take a mask-free fast path (wave-uniform branch),
//   - T5 STATIC form only: one priority raise for the younger wave half
//     (per-cluster setprio flips are scheduling fences and were keeping the
//     softmax VALU out of the MFMA gaps; the sched_group_barrier interleave
//     corrupts numerics and stays compiled-but-disabled — VAR bit 1),
//   - MOD_BLOCKMASK (K2): arbitrary mask_mod via device granule codes +
//     packed keep-bits + per-block live kv ranges; optional additive bias
//     stream for score_mod (ops/attention.py CompiledBlockMask),
//   - causal/sliding-window tiles skipped at block level; GQA reads the
//     shared KV head directly; BSHD layout, strided views accepted.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "attn_common.h"

namespace {

constexpr int QPW = 32;  // q rows per wave
constexpr float LOG2E = 1.4426950408889634f;
constexpr float LN2 = 0.6931471805599453f;

template <int D, int MOD, int NW, int KVB, int VAR = 3>
__global__ __launch_bounds__(NW* WAVE) void attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ o,
    float* __restrict__ lse, const float* __restrict__ slopes,
    int B, int Sq, int Skv, int Hq, int Hkv, float scale, int modarg,
    long q_rs, long k_rs, long v_rs,
    // MOD_BLOCKMASK only (K2: arbitrary mask_mod compiled to device tensors;
    // /root/reference/models/attention/flex_attention.py:356-411):
    //   mb_gran  uint8 [B,H,ceil(Sq/32),ceil(Skv/KVB)] 0=masked 1=bits 2=full
    //   mb_bits  uint8 [B,H,Sq,ceil(Skv/8)] per-element keep bits
    //   mb_range int32 [B,H,ceil(Sq/QPB),2] first/last+1 live kv tile
    //   bias     bf16  [B,H,Sq,Skv] optional additive score bias (score_mod)
    const unsigned char* __restrict__ mb_gran = nullptr,
    const unsigned char* __restrict__ mb_bits = nullptr,
    const int* __restrict__ mb_range = nullptr,
    const __hip_bfloat16* __restrict__ bias = nullptr) {
  constexpr int TPB = NW * WAVE;
  constexpr int QPB = NW * QPW;
  constexpr int DBLK = D / 16;  // QK^T d-slots
  constexpr int DCOL = D / 32;  // PV output column tiles
  constexpr int KT = KVB / 32;  // 32-row k sub-tiles
  // XOR swizzle masks (element units; <<3 = 8-element/16-byte granules)
  constexpr int KSWZ = (D >= 128) ? 15 : 7;   // K image rows (row bytes = 2D)
  constexpr int VROW = (KVB > 64) ? KVB : 64; // V^T image row: col index is the KV row (< KVB)
  constexpr int TILE = KVB * D + D * VROW;    // elements per buffer

  // 3-slot ring: with PV lagging QK^T by one tile (cross-tile pipeline),
  // iteration j reads K(j) from slot j%3, V(j-1) from slot (j-1)%3 and
  // writes tile j+1 into slot (j+1)%3 — all distinct, still ONE barrier
  // per tile (slot j%3's readers both finish before the end-of-(j+1) barrier
  // that precedes its reuse at j+3).
  __shared__ __hip_bfloat16 smem[3 * TILE];

  const TileMap tmap = tile_map();
  const int b = tmap.batch;
  const int hq = tmap.head;
  const int qtile = tmap.tile;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x;
  const int wave = tid / WAVE;
  const int lane = tid % WAVE;
  const int lq = lane & 31;
  const int hi = lane >> 5;

  const int q0w = qtile * QPB + wave * QPW;
  const int qrow = q0w + lq;
  const bool q_valid = qrow < Sq;
  const int q_off = Skv - Sq;
  const int q_pos = qrow + q_off;
  const float scale2 = scale * LOG2E;

  // ---- Q fragments ----
  bf16x8 qf[DBLK];
  {
    const __hip_bfloat16* qp =
        q + ((long)b * Sq + (q_valid ? qrow : 0)) * q_rs + (long)hq * D + hi * 8;
#pragma unroll
    for (int dblk = 0; dblk < DBLK; ++dblk) {
      Bf16x8U u;
      *reinterpret_cast<uint4*>(u.s) =
          q_valid ? *reinterpret_cast<const uint4*>(qp + dblk * 16) : uint4{0, 0, 0, 0};
      qf[dblk] = u.v;
    }
  }

  // ---- kv range ----
  const int blk_qpos_lo = qtile * QPB + q_off;
  const int blk_qpos_hi = blk_qpos_lo + QPB - 1;
  int kv_lo = 0, kv_hi = Skv;
  if constexpr (MOD == MOD_CAUSAL || MOD == MOD_ALIBI) {
    kv_hi = min(Skv, blk_qpos_hi + 1);
  } else if constexpr (MOD == MOD_SLIDING_WINDOW) {
    kv_hi = min(Skv, blk_qpos_hi + 1);
    kv_lo = max(0, blk_qpos_lo - modarg + 1) & ~(KVB - 1);
  } else if constexpr (MOD == MOD_PREFIX_LM) {
    kv_hi = min(Skv, max(blk_qpos_hi + 1, modarg));
  } else if constexpr (MOD == MOD_BLOCKMASK) {
    const int nqpb = (Sq + QPB - 1) / QPB;
    const int* r = mb_range + (((long)b * Hq + hq) * nqpb + qtile) * 2;
    kv_lo = r[0] * KVB;
    kv_hi = min(Skv, r[1] * KVB);
    if (kv_hi <= kv_lo) kv_hi = kv_lo;  // fully-masked row block: loop skips
  }
  const float slope2 = (MOD == MOD_ALIBI) ? slopes[hq] * LOG2E : 0.f;
  // blockmask strides (granule: this wave's 32-row q granule per kv tile)
  const int nkvt = (Skv + KVB - 1) / KVB;
  const int nq32 = (Sq + 31) / 32;
  const long gran_base = (MOD == MOD_BLOCKMASK)
      ? (((long)b * Hq + hq) * nq32 + min(q0w / 32, nq32 - 1)) * (long)nkvt : 0;
  const int kvb8 = (Skv + 7) / 8;
  const long bits_row = (MOD == MOD_BLOCKMASK && q_valid)
      ? (((long)b * Hq + hq) * Sq + qrow) * (long)kvb8 : 0;
  const long bias_row = (MOD == MOD_BLOCKMASK && q_valid && bias)
      ? (((long)b * Hq + hq) * Sq + qrow) * (long)Skv : 0;

  // ---- staging helpers (T14 split: load -> regs early, write -> LDS late) --
  // K: KU4 uint4 chunks per thread (row-major swizzled image, b128 writes).
  // V: VC chunks of (2 rows x 8 d), written as 8 b32 into the transposed
  //    swizzled image (2 k-elements per write).
  constexpr int K_TOT = KVB * (D / 8);            // uint4 per K tile
  constexpr int KU4 = (K_TOT + TPB - 1) / TPB;
  constexpr int V_TOT = (KVB / 2) * (D / 8);      // chunks per V tile
  constexpr int VC = (V_TOT + TPB - 1) / TPB;

  uint4 kreg[KU4], vreg[VC][2];

  auto stage_load = [&](int kv0) {
#pragma unroll
    for (int c = 0; c < KU4; ++c) {
      const int u = tid + c * TPB;
      const int row = u / (D / 8);
      const int d0 = (u % (D / 8)) * 8;
      const bool ok = row < KVB && kv0 + row < Skv;
      kreg[c] = ok ? *reinterpret_cast<const uint4*>(
                         k + ((long)b * Skv + kv0 + row) * k_rs + (long)hkv * D + d0)
                   : uint4{0, 0, 0, 0};
    }
#pragma unroll
    for (int c = 0; c < VC; ++c) {
      const int u = tid + c * TPB;
      const int row = (u / (D / 8)) * 2;
      const int d0 = (u % (D / 8)) * 8;
      const bool ok0 = row < KVB && kv0 + row < Skv;
      const bool ok1 = row + 1 < KVB && kv0 + row + 1 < Skv;
      const long base = ((long)b * Skv + kv0 + row) * v_rs + (long)hkv * D + d0;
      vreg[c][0] = ok0 ? *reinterpret_cast<const uint4*>(v + base) : uint4{0, 0, 0, 0};
      vreg[c][1] = ok1 ? *reinterpret_cast<const uint4*>(v + base + v_rs) : uint4{0, 0, 0, 0};
    }
  };

  auto stage_write = [&](int buf) {
    __hip_bfloat16* k_lds = smem + buf * TILE;
    __hip_bfloat16* vt_lds = k_lds + KVB * D;
#pragma unroll
    for (int c = 0; c < KU4; ++c) {
      const int u = tid + c * TPB;
      const int row = u / (D / 8);
      const int d0 = (u % (D / 8)) * 8;
      if (row < KVB)
        *reinterpret_cast<uint4*>(k_lds + row * D + (d0 ^ ((row & KSWZ) << 3))) = kreg[c];
    }
#pragma unroll
    for (int c = 0; c < VC; ++c) {
      const int u = tid + c * TPB;
      const int row = (u / (D / 8)) * 2;
      const int d0 = (u % (D / 8)) * 8;
      if (row < KVB) {
        Bf16x8U v0, v1;
        *reinterpret_cast<uint4*>(v0.s) = vreg[c][0];
        *reinterpret_cast<uint4*>(v1.s) = vreg[c][1];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int drow = d0 + j;
          const uint pair = (uint)v0.s[j] | ((uint)v1.s[j] << 16);
          *reinterpret_cast<uint*>(vt_lds + drow * VROW + (row ^ swzt(drow))) = pair;
        }
      }
    }
  };

  float m = -INFINITY, l = 0.f;  // m in base-2 domain
  float o_acc[DCOL][16];
#pragma unroll
  for (int dc = 0; dc < DCOL; ++dc)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[dc][r] = 0.f;

  // T5 static form: the second-dispatched half (waves 4-7) loses VALU
  // arbitration on every segment; one static priority raise for it (and no
  // per-cluster flips: s_setprio is a scheduling fence that was keeping the
  // exp/pack VALU OUT of the PV MFMA issue gaps -- seen in the .s).
  if ((VAR & 1) && NW == 8 && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  // prologue: stage tile 0 into slot 0
  stage_load(kv_lo);
  stage_write(0);
  __syncthreads();

  // Cross-tile software pipeline: PV lags QK^T by one tile, so tile j's
  // exp2/psum/pack VALU is register-independent of tile j-1's 16 PV MFMAs
  // and the scheduler interleaves them (today's serial order exposes the
  // whole softmax between the two MFMA clusters). T13 ordering is kept:
  // the o/l rescale for tile j's max applies AFTER PV(j-1) completes, so
  // P_{j-1} enters o at its own scale exactly once.
  bf16x8 pa[KT][2];  // A-fragments of the PREVIOUS tile's P (tile -1: zeros)
#pragma unroll
  for (int kt = 0; kt < KT; ++kt) {
    pa[kt][0] = bf16x8{};
    pa[kt][1] = bf16x8{};
  }
  // pv_slot starts at slot 0 (= tile 0's finite data): pa==0 keeps the
  // dummy PV contribution exactly 0 without reading uninitialized LDS.
  int buf = 0, pv_slot = 0, nslot = 1;
  for (int kv0 = kv_lo; kv0 < kv_hi; kv0 += KVB) {
    const bool has_next = kv0 + KVB < kv_hi;
    if (has_next) stage_load(kv0 + KVB);  // loads in flight under the MFMAs

    const __hip_bfloat16* k_lds = smem + buf * TILE;
    const __hip_bfloat16* vt_prev = smem + pv_slot * TILE + KVB * D;

    // ---- S^T = mfma(K, Q) per 32-row k sub-tile ----
    f32x16 st[KT];
#pragma unroll
    for (int kt = 0; kt < KT; ++kt) st[kt] = f32x16{};
#pragma unroll
    for (int kt = 0; kt < KT; ++kt) {
      const int krow = kt * 32 + lq;
#pragma unroll
      for (int dblk = 0; dblk < DBLK; ++dblk) {
        Bf16x8U kf;
        *reinterpret_cast<uint4*>(kf.s) = *reinterpret_cast<const uint4*>(
            k_lds + krow * D + ((dblk * 16 + hi * 8) ^ ((krow & KSWZ) << 3)));
        st[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf.v, qf[dblk], st[kt], 0, 0, 0);
      }
    }

    // ---- tile max (tree reduce, raw units on the full fast path) ----
    // interior tiles (wave-uniform): every (q,k) pair of this wave is kept
    bool full = kv0 + KVB <= Skv;
    if constexpr (MOD == MOD_CAUSAL) {
      full = full && (kv0 + KVB - 1 <= q0w + q_off);
    } else if constexpr (MOD == MOD_SLIDING_WINDOW) {
      full = full && (kv0 + KVB - 1 <= q0w + q_off) &&
             ((q0w + QPW - 1 + q_off) - kv0 < modarg);
    } else if constexpr (MOD == MOD_PREFIX_LM) {
      full = full && ((kv0 + KVB - 1 <= q0w + q_off) || (kv0 + KVB <= modarg));
    } else if constexpr (MOD == MOD_ALIBI) {
      full = false;  // slope term needs per-element positions anyway
    } else if constexpr (MOD == MOD_BLOCKMASK) {
      // per-wave granule: full only if every lane's granule says 2 AND no
      // bias (bias must be added per element)
      const unsigned char g = mb_gran[gran_base + (kv0 / KVB)];
      full = full && (g == 2) && (bias == nullptr) && q_valid;
      full = __all(full);
    }

    float tmax;
    if (full) {
      // exp2+fma fold: scale2 is folded into the exp argument below, so the
      // 32-element scale pass disappears; the max runs on raw scores
      // (scale2 > 0 keeps it monotone) as a depth-5 tree, not a 31-op chain.
      float tm[8];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        tm[i] = fmaxf(fmaxf(st[0][i], st[0][i + 8]),
                      KT > 1 ? fmaxf(st[1][i], st[1][i + 8]) : -INFINITY);
#pragma unroll
      for (int i = 0; i < 4; ++i) tm[i] = fmaxf(tm[i], tm[i + 4]);
      tmax = fmaxf(fmaxf(tm[0], tm[1]), fmaxf(tm[2], tm[3])) * scale2;
    } else {
      // masked path: materialize base-2 scores in place of st
      tmax = -INFINITY;
#pragma unroll
      for (int kt = 0; kt < KT; ++kt)
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
          const int k_pos = kv0 + kt * 32 + acc_row(reg, hi);
          bool keep;
          if constexpr (MOD == MOD_BLOCKMASK) {
            keep = q_valid && k_pos < Skv;
            if (keep)
              keep = (mb_bits[bits_row + (k_pos >> 3)] >> (k_pos & 7)) & 1;
          } else {
            keep = q_valid && attn_keep<MOD>(q_pos, k_pos, Skv, modarg);
          }
          float s = st[kt][reg] * scale2;
          if constexpr (MOD == MOD_ALIBI) s += slope2 * (k_pos - q_pos);
          if constexpr (MOD == MOD_BLOCKMASK) {
            if (bias != nullptr && keep)
              s += __bfloat162float(bias[bias_row + k_pos]) * LOG2E;
          }
          st[kt][reg] = keep ? s : -INFINITY;
          tmax = fmaxf(tmax, st[kt][reg]);
        }
    }
    tmax = cross32_max(tmax);
    // defer-max (guide T13): while no lane's tile max exceeds the running
    // max by more than THR (base-2), keep the old max and SKIP the O/l
    // rescale entirely — P is then bounded by 2^THR, which the fp32
    // accumulators tolerate (~3x max-abs error vs THR=0). The decision is
    // wave-uniform; the o_acc rescale it gates runs after PV(j-1) below.
    const float m_old = m;
    const bool defer = __all(tmax - m <= 8.0f);
    if (!defer) m = fmaxf(m, tmax);
    const float mc = fmaxf(m, -1e30f);
    // branchless forms so [PV MFMAs + exp + psum + pack] is ONE basic block
    // (a branch splits the scheduling region and exiles the VALU from the
    // MFMA issue gaps): masked-path st is pre-scaled, so sc2 folds to 1.
    const float sc2 = full ? scale2 : 1.0f;
    // alpha == 1 exactly when defer (mc == m_old); exp2(-inf)=0 at tile 0.
    const float alpha = __builtin_amdgcn_exp2f(m_old - mc);

    // ---- PV(j-1): accumulates P_{j-1}*V at scale m_{j-1} (o_acc still at
    // that scale). The exp/psum/pack block below is independent of o_acc
    // and interleaves into these MFMAs' issue gaps. ----
#pragma unroll
    for (int dc = 0; dc < DCOL; ++dc) {
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = o_acc[dc][r];
#pragma unroll
      for (int kt = 0; kt < KT; ++kt)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const int drow = dc * 32 + lq;
          Bf16x8U vfr;
          *reinterpret_cast<uint4*>(vfr.s) = *reinterpret_cast<const uint4*>(
              vt_prev + drow * VROW + ((kt * 32 + ks * 16 + hi * 8) ^ swzt(drow)));
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[kt][ks], vfr.v, acc, 0, 0, 0);
        }
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[dc][r] = acc[r];
    }

    // ---- exp (base-2) + row sum + pack for the NEXT iteration's PV ----
    // branchless (sc2 / alpha above) — same basic block as the PV MFMAs
    float p[KT][16];
    float psum = 0.f;
#pragma unroll
    for (int kt = 0; kt < KT; ++kt)
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        p[kt][reg] = __builtin_amdgcn_exp2f(fmaf(st[kt][reg], sc2, -mc));
        psum += p[kt][reg];
      }
    psum = cross32_sum(psum);
    l = l * alpha + psum;  // alpha == 1 on the defer path
#pragma unroll
    for (int kt = 0; kt < KT; ++kt) acc_to_afrag(p[kt], pa[kt][0], pa[kt][1]);

    // sm-split (guide T19 + ladder rung 2b): the scheduler otherwise emits
    // the whole exp/psum/pack VALU block AFTER the PV MFMAs (checked in the
    // .s); these directives interleave ~24 VALU per 2 MFMAs so the VALU
    // issues inside the matrix-pipe gaps.
    if constexpr ((VAR & 2) != 0) {
#pragma unroll
      for (int g = 0; g < 8; ++g) {
        __builtin_amdgcn_sched_group_barrier(0x100, 4, 0);  // DS_READ (V frags)
        __builtin_amdgcn_sched_group_barrier(0x008, 2, 0);  // MFMA
        __builtin_amdgcn_sched_group_barrier(0x002, 24, 0); // VALU slice
      }
    }

    // o_acc rescale: strictly after PV(j-1), before PV(j) (T13 hazard);
    // skipped on the defer path (alpha == 1 there, the l-update above is
    // already folded in)
    if (!defer) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const float ar = __shfl(alpha, acc_row(reg, hi));
#pragma unroll
        for (int dc = 0; dc < DCOL; ++dc) o_acc[dc][reg] *= ar;
      }
    }

    // write tile t+1 LAST: the vmcnt wait on stage_load's global loads has
    // the whole iteration's MFMA/softmax work to hide under (T14).
    if (has_next) stage_write(nslot);

    __syncthreads();
    pv_slot = buf;
    buf = nslot;
    nslot = (nslot == 2) ? 0 : nslot + 1;
  }

  // ---- drain: PV of the last tile ----
  {
    const __hip_bfloat16* vt_prev = smem + pv_slot * TILE + KVB * D;
#pragma unroll
    for (int dc = 0; dc < DCOL; ++dc) {
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = o_acc[dc][r];
#pragma unroll
      for (int kt = 0; kt < KT; ++kt)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          const int drow = dc * 32 + lq;
          Bf16x8U vfr;
          *reinterpret_cast<uint4*>(vfr.s) = *reinterpret_cast<const uint4*>(
              vt_prev + drow * VROW + ((kt * 32 + ks * 16 + hi * 8) ^ swzt(drow)));
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[kt][ks], vfr.v, acc, 0, 0, 0);
        }
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[dc][r] = acc[r];
    }
  }

  // ---- epilogue ----
  if (q_valid && hi == 0)
    lse[((long)b * Hq + hq) * Sq + qrow] = (m + __builtin_amdgcn_logf(l)) * LN2;

  const int dl = lane & 31;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int r = acc_row(reg, hi);
    const int q_r = q0w + r;
    if (q_r >= Sq) continue;
    const float linv = 1.f / __shfl(l, r);
    __hip_bfloat16* orow = o + (((long)b * Sq + q_r) * Hq + hq) * D + dl;
#pragma unroll
    for (int dc = 0; dc < DCOL; ++dc)
      orow[dc * 32] = __float2bfloat16(o_acc[dc][reg] * linv);
  }
}

int fwd_qpb() {  // q rows per block (needed by the host for grid sizing)
  static int nw = []() {
    const char* e = getenv("MCDP_ATTN_NW");
    return e ? atoi(e) : 8;
  }();
  return nw * QPW;
}

template <int D, int MOD>
void launch_fwd_cfg(dim3 grid, dim3 block, hipStream_t stream,
                    const __hip_bfloat16* q, const __hip_bfloat16* k, const __hip_bfloat16* v,
                    __hip_bfloat16* o, float* lse, const float* slopes,
                    int B, int Sq, int Skv, int Hq, int Hkv, float scale, int modarg,
                    long q_rs, long k_rs, long v_rs) {
  const char* e = getenv("MCDP_ATTN_KVB");
  // D=64 defaults to 128-row tiles: twice the MFMAs per barrier at the same
  // occupancy (234 VGPR, no spill) — fwd 312 -> 363 TF at the 124M shape.
  int kvb = e ? atoi(e) : (D == 64 ? 128 : 64);
  if (kvb == 128 && D != 64) kvb = 64;  // KVB=128 is D=64-only (VGPR budget)
  const int nw = fwd_qpb() / QPW;
  static const int var = []() {
    const char* e = getenv("MCDP_ATTN_FWD_VAR");
    // default 1 = static-prio only. bit1 (sched_group_barrier interleave)
    // CORRUPTS numerics (NaN via a poisoned cross-lane max; bisect
    // gpurun_out/r2_call14.log: VAR 0/1 exact, VAR 2/3 NaN on every shape)
    // — kept compiled for future debugging, not for use.
    return e ? atoi(e) : 1;
  }();
#define LAUNCH(NW_, KVB_)                                                              \
  do {                                                                                 \
    if (var == 0)                                                                      \
      attn_fwd_kernel<D, MOD, NW_, KVB_, 0><<<grid, block, 0, stream>>>(               \
          q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs); \
    else if (var == 1)                                                                 \
      attn_fwd_kernel<D, MOD, NW_, KVB_, 1><<<grid, block, 0, stream>>>(               \
          q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs); \
    else if (var == 2)                                                                 \
      attn_fwd_kernel<D, MOD, NW_, KVB_, 2><<<grid, block, 0, stream>>>(               \
          q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs); \
    else                                                                               \
      attn_fwd_kernel<D, MOD, NW_, KVB_, 3><<<grid, block, 0, stream>>>(               \
          q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs); \
  } while (0)
  if (nw == 8 && kvb == 64) LAUNCH(8, 64);
  else if (nw == 8 && kvb == 32) LAUNCH(8, 32);
  else if (nw == 4 && kvb == 64) LAUNCH(4, 64);
  else if (nw == 4 && kvb == 32) LAUNCH(4, 32);
  else if (nw == 16 && kvb == 64) LAUNCH(16, 64);
  else if (nw == 8 && kvb == 128) {
    if constexpr (D == 64) LAUNCH(8, 128);
    else TORCH_CHECK(false, "attn_fwd: KVB=128 is D=64-only (LDS/VGPR budget)");
  }
  else TORCH_CHECK(false, "attn_fwd: unsupported NW/KVB ", nw, "/", kvb);
#undef LAUNCH
}

template <int D>
void launch_fwd(int mod, dim3 grid, dim3 block, hipStream_t stream,
                const __hip_bfloat16* q, const __hip_bfloat16* k, const __hip_bfloat16* v,
                __hip_bfloat16* o, float* lse, const float* slopes,
                int B, int Sq, int Skv, int Hq, int Hkv, float scale, int modarg,
                long q_rs, long k_rs, long v_rs) {
  switch (mod) {
    case MOD_NONE:
      launch_fwd_cfg<D, MOD_NONE>(grid, block, stream, q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    case MOD_CAUSAL:
      launch_fwd_cfg<D, MOD_CAUSAL>(grid, block, stream, q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    case MOD_SLIDING_WINDOW:
      launch_fwd_cfg<D, MOD_SLIDING_WINDOW>(grid, block, stream, q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    case MOD_PREFIX_LM:
      launch_fwd_cfg<D, MOD_PREFIX_LM>(grid, block, stream, q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    case MOD_ALIBI:
      launch_fwd_cfg<D, MOD_ALIBI>(grid, block, stream, q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    default:
      TORCH_CHECK(false, "attn_fwd: unknown mod ", mod);
  }
}

}  // namespace

static long bshd_row_stride(at::Tensor& t) {
  // accept [B,S,H,D] views whose (h,d) inner block is contiguous (e.g. slices
  // of the fused QKV projection); otherwise materialize.
  const int S = t.size(1), H = t.size(2), D = t.size(3);
  if (!(t.stride(3) == 1 && t.stride(2) == D && t.stride(0) == (long)S * t.stride(1)))
    t = t.contiguous();
  return t.stride(1);
}

std::vector<at::Tensor> attn_fwd_blockmask(at::Tensor q, at::Tensor k, at::Tensor v,
                                           double scale, at::Tensor gran, at::Tensor bits,
                                           at::Tensor range, at::Tensor bias) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16, "bf16 only");
  const long q_rs = bshd_row_stride(q), k_rs = bshd_row_stride(k), v_rs = bshd_row_stride(v);
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1), Hkv = k.size(2);
  TORCH_CHECK(Hq % Hkv == 0 && (D == 64 || D == 128));
  TORCH_CHECK(gran.scalar_type() == at::kByte && bits.scalar_type() == at::kByte &&
              range.scalar_type() == at::kInt, "blockmask tensor dtypes");
  constexpr int QPB = 8 * 32;
  TORCH_CHECK(range.size(2) == (Sq + QPB - 1) / QPB, "range shape mismatch");
  auto o = at::empty({B, Sq, Hq, D}, q.options());
  auto lse = at::empty({B, Hq, Sq}, q.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(cdiv(Sq, QPB), Hq, B), block(512);
  const bool has_bias = bias.numel() > 0;
  auto* bp = has_bias ? reinterpret_cast<const __hip_bfloat16*>(bias.data_ptr()) : nullptr;
  auto launch1 = [&](auto dtag) {
    constexpr int DD = decltype(dtag)::value;
    attn_fwd_kernel<DD, MOD_BLOCKMASK, 8, 64, 1><<<grid, block, 0, stream>>>(
        reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
        reinterpret_cast<__hip_bfloat16*>(o.data_ptr()), lse.data_ptr<float>(),
        nullptr, B, Sq, Skv, Hq, Hkv, (float)scale, 0, q_rs, k_rs, v_rs,
        gran.data_ptr<unsigned char>(), bits.data_ptr<unsigned char>(),
        range.data_ptr<int>(), bp);
  };
  if (D == 64) launch1(std::integral_constant<int, 64>{});
  else launch1(std::integral_constant<int, 128>{});
  return {o, lse};
}

std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v, double scale,
                                 long mod, long modarg, at::Tensor slopes) {
  TORCH_CHECK(q.is_cuda());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attn_fwd: bf16 only");
  const long q_rs = bshd_row_stride(q), k_rs = bshd_row_stride(k), v_rs = bshd_row_stride(v);
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1), Hkv = k.size(2);
  TORCH_CHECK(Hq % Hkv == 0, "GQA requires Hq % Hkv == 0");
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd: head_dim must be 64 or 128, got ", D);
  auto o = at::empty({B, Sq, Hq, D}, q.options());
  auto lse = at::empty({B, Hq, Sq}, q.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  const int qpb = fwd_qpb();
  dim3 grid(cdiv(Sq, qpb), Hq, B);
  dim3 block(qpb / QPW * WAVE);
  const float* sl = slopes.numel() > 0 ? slopes.data_ptr<float>() : nullptr;
  auto* qp = reinterpret_cast<const __hip_bfloat16*>(q.data_ptr());
  auto* kp = reinterpret_cast<const __hip_bfloat16*>(k.data_ptr());
  auto* vp = reinterpret_cast<const __hip_bfloat16*>(v.data_ptr());
  auto* op = reinterpret_cast<__hip_bfloat16*>(o.data_ptr());
  if (D == 64)
    launch_fwd<64>((int)mod, grid, block, stream, qp, kp, vp, op, lse.data_ptr<float>(), sl, B, Sq, Skv, Hq, Hkv, (float)scale, (int)modarg, q_rs, k_rs, v_rs);
  else
    launch_fwd<128>((int)mod, grid, block, stream, qp, kp, vp, op, lse.data_ptr<float>(), sl, B, Sq, Skv, Hq, Hkv, (float)scale, (int)modarg, q_rs, k_rs, v_rs);
  return {o, lse};
}
