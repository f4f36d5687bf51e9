// Tiled FlashAttention-2 backward (kernel K1 bwd).
//
// Deterministic three-kernel design (no atomics):
//   preprocess: Drow[b,h,s] = sum_d dO*O  (vectorized 16B loads, 4 rows/wave)
//   dQ kernel : block per q-tile; recomputes P from (Q,K,lse); accumulates
//               dQ = (P∘(dP−Drow))·scale @ K in registers
//   dK kernel : block per kv-tile; loops the GQA group's q heads; accumulates
//               dK = dS^T @ Q in registers
//   dV kernel : block per kv-tile; accumulates dV = P^T @ dO in registers
// Splitting dK/dV keeps each kernel under the 256-VGPR budget at D=128
// (merged, the two 32x128 fp32 accumulators alone are 128 VGPRs).
//
// Same gfx950 structure as the forward (attn_fwd.hip): 8 waves/block sharing
// every staged tile, DOUBLE-BUFFERED tiles with one barrier per iteration,
// T14 register-prefetch of the next tile under the MFMA clusters, base-2
// exponentials with log2(e) folded into the scale, mask-free fast path on
// interior causal tiles, s_setprio around MFMAs, MFMA layout +
// acc_to_afrag transform from attn_common.h.
//
// The bwd-specific layout trick: ALL images are row-major with the s_tr
// swizzle (attn_common.h), conflict-free for BOTH read shapes — b128 row
// reads AND the k-strided B-fragment gathers, which use gfx950's
// ds_read_b64_tr_b16 hardware transpose-read (tr16_frag). No transposed
// LDS copies are staged anywhere in the backward.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "attn_common.h"

namespace {

// kv tile per dQ iteration: 64 rows staged per barrier at D=128 (two 32-row
// sub-tiles -> 48 MFMAs/barrier), 128 rows at D=64 (four sub-tiles, 96
// MFMAs/barrier at 166 VGPR) — the dQ kernel overrides this per D.
constexpr int KVB_FILE = 64;
constexpr float LOG2E = 1.4426950408889634f;

// waves per block (8 or 16): 16-wave blocks halve staging traffic and
// barrier frequency per unit of MFMA work at the same 4 waves/SIMD
// (1 block/CU instead of 2). Env MCDP_ATTN_BWD_NW selects at launch.
inline int bwd_nw() {
  static int nw = []() {
    const char* e = getenv("MCDP_ATTN_BWD_NW");
    // NW=4 measured 2.4x SLOWER at D=128 (call7: 139 vs 331 TF) once the
    // launch bug that faked its earlier win was fixed — default stays 8.
    return e ? atoi(e) : 8;
  }();
  return nw;
}

// ---------------- preprocess: Drow = rowsum(dO * O) ----------------
// 4 rows per wave: 16 lanes x 8 elements cover a D=128 row in one uint4 load.
__global__ void bwd_preprocess_kernel(const __hip_bfloat16* __restrict__ dout,
                                      const __hip_bfloat16* __restrict__ o,
                                      float* __restrict__ drow,
                                      long rows, int D, long do_rs, int Hq) {
  const int lpr = D / 8;                  // lanes needed per row (16 at D=128)
  const int rpw = WAVE / lpr;             // rows per wave (4 at D=128)
  const int lane = threadIdx.x % WAVE;
  const long row = (blockIdx.x * (long)(256 / WAVE) + threadIdx.x / WAVE) * rpw
                   + lane / lpr;
  if (row >= rows) return;
  const long bs = row / Hq;
  const int h = (int)(row % Hq);
  const int d0 = (lane % lpr) * 8;
  Bf16x8U du, ou;
  *reinterpret_cast<uint4*>(du.s) =
      *reinterpret_cast<const uint4*>(dout + bs * do_rs + (long)h * D + d0);
  *reinterpret_cast<uint4*>(ou.s) =
      *reinterpret_cast<const uint4*>(o + row * D + d0);
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) acc += to_f32(du.h[j]) * to_f32(ou.h[j]);
  // segmented reduce within the lpr-lane group
  for (int off = lpr / 2; off > 0; off >>= 1) acc += __shfl_xor(acc, off);
  if (lane % lpr == 0) drow[row] = acc;
}

// q-range of kv-tile blocks per mod (in q-row space)
template <int MOD>
__device__ __forceinline__ void q_range_for_kv(int kv0, int kpb, int q_off, int Sq,
                                               int modarg, int& q_lo, int& q_hi) {
  q_lo = 0;
  q_hi = Sq;
  if constexpr (MOD == MOD_CAUSAL || MOD == MOD_ALIBI) {
    q_lo = max(0, kv0 - q_off) & ~31;
  } else if constexpr (MOD == MOD_SLIDING_WINDOW) {
    q_lo = max(0, kv0 - q_off) & ~31;
    q_hi = min(Sq, kv0 + kpb - 1 + modarg - q_off + 1);
  } else if constexpr (MOD == MOD_PREFIX_LM) {
    if (kv0 >= modarg) q_lo = max(0, kv0 - q_off) & ~31;
  }
}

// ---------------- dQ kernel (block per q-tile of QPB rows) ----------------
template <int D, int MOD, int NW>
__global__ __launch_bounds__(NW* WAVE) void attn_bwd_dq_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    __hip_bfloat16* __restrict__ dq, const float* __restrict__ slopes,
    int B, int Sq, int Skv, int Hq, int Hkv, float scale, int modarg,
    long q_rs, long k_rs, long v_rs, long do_rs, long dq_rs) {
  constexpr int TPB = NW * WAVE;
  constexpr int QPB = 32 * NW;
  constexpr int DBLK = D / 16;
  constexpr int DCOL = D / 32;
  // D=64: 128-row kv tiles double the MFMAs per barrier at halved per-tile
  // register cost (mirrors the fwd D=64 KVB bump)
  constexpr int KVB = (D == 64) ? 128 : KVB_FILE;
  // K rm + V rm only: the dQ-accumulate B-fragments (K, k-strided) are read
  // straight from the row-major K image with ds_read_b64_tr_b16 (tr16_frag),
  // so no transposed K image is staged.
  constexpr int TILE = KVB * D * 2;

  __shared__ __align__(16) __hip_bfloat16 smem[2 * TILE];

  const TileMap tmap = tile_map();
  const int b = tmap.batch, hq = tmap.head, qtile = tmap.tile;
  const int hkv = hq / (Hq / Hkv);
  const int tid = threadIdx.x, wave = tid / WAVE, lane = tid % WAVE;
  const int lq = lane & 31, hi = lane >> 5;
  const int q0w = qtile * QPB + wave * 32;
  const int qrow = q0w + lq;
  const bool q_valid = qrow < Sq;
  const int q_off = Skv - Sq;
  const int q_pos = qrow + q_off;
  const float scale2 = scale * LOG2E;
  const float slope2 = (MOD == MOD_ALIBI) ? slopes[hq] * LOG2E : 0.f;

  // Q and dO fragments (B-operand layout: lane holds row q=lq, 8 d values)
  bf16x8 qf[DBLK], dof[DBLK];
  {
    const long row0 = (long)b * Sq + (q_valid ? qrow : 0);
    const long hd = (long)hq * D + hi * 8;
#pragma unroll
    for (int dblk = 0; dblk < DBLK; ++dblk) {
      Bf16x8U uq, ud;
      *reinterpret_cast<uint4*>(uq.s) =
          q_valid ? *reinterpret_cast<const uint4*>(q + row0 * q_rs + hd + dblk * 16) : uint4{0, 0, 0, 0};
      *reinterpret_cast<uint4*>(ud.s) =
          q_valid ? *reinterpret_cast<const uint4*>(dout + row0 * do_rs + hd + dblk * 16) : uint4{0, 0, 0, 0};
      qf[dblk] = uq.v;
      dof[dblk] = ud.v;
    }
  }
  const float Lq2 = (q_valid ? lse[((long)b * Hq + hq) * Sq + qrow] : INFINITY) * LOG2E;
  const float Dq = q_valid ? drow[((long)b * Sq + qrow) * Hq + hq] : 0.f;

  // kv range (same as forward)
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
  }

  // T14 staging: chunks of (2 kv rows x 8 d); K and V both write row-major.
  constexpr int CH_TOT = (KVB / 2) * (D / 8);  // chunks per tile
  constexpr int NCH = (CH_TOT + TPB - 1) / TPB;
  uint4 kreg[NCH][2], vreg[NCH][2];

  auto stage_load = [&](int kv0) {
#pragma unroll
    for (int c = 0; c < NCH; ++c) {
      const int u = tid + c * TPB;
      const int row = (u / (D / 8)) * 2;
      const int d0 = (u % (D / 8)) * 8;
      if (row >= KVB) continue;
      const bool ok0 = kv0 + row < Skv;
      const bool ok1 = kv0 + row + 1 < Skv;
      const long kb = ((long)b * Skv + kv0 + row) * k_rs + (long)hkv * D + d0;
      const long vb = ((long)b * Skv + kv0 + row) * v_rs + (long)hkv * D + d0;
      kreg[c][0] = ok0 ? *reinterpret_cast<const uint4*>(k + kb) : uint4{0, 0, 0, 0};
      kreg[c][1] = ok1 ? *reinterpret_cast<const uint4*>(k + kb + k_rs) : uint4{0, 0, 0, 0};
      vreg[c][0] = ok0 ? *reinterpret_cast<const uint4*>(v + vb) : uint4{0, 0, 0, 0};
      vreg[c][1] = ok1 ? *reinterpret_cast<const uint4*>(v + vb + v_rs) : uint4{0, 0, 0, 0};
    }
  };
  auto stage_write = [&](int bufsel) {
    __hip_bfloat16* k_lds = smem + bufsel * TILE;
    __hip_bfloat16* v_lds = k_lds + KVB * D;
#pragma unroll
    for (int c = 0; c < NCH; ++c) {
      const int u = tid + c * TPB;
      const int row = (u / (D / 8)) * 2;
      const int d0 = (u % (D / 8)) * 8;
      if (row >= KVB) continue;
      *reinterpret_cast<uint4*>(k_lds + rm_swz<D>(row, d0)) = kreg[c][0];
      *reinterpret_cast<uint4*>(k_lds + rm_swz<D>(row + 1, d0)) = kreg[c][1];
      *reinterpret_cast<uint4*>(v_lds + rm_swz<D>(row, d0)) = vreg[c][0];
      *reinterpret_cast<uint4*>(v_lds + rm_swz<D>(row + 1, d0)) = vreg[c][1];
    }
  };

  float dq_acc[DCOL][16];
#pragma unroll
  for (int dc = 0; dc < DCOL; ++dc)
#pragma unroll
    for (int r = 0; r < 16; ++r) dq_acc[dc][r] = 0.f;

  stage_load(kv_lo);
  stage_write(0);
  __syncthreads();

  int buf = 0;
  for (int kv0 = kv_lo; kv0 < kv_hi; kv0 += KVB) {
    const bool has_next = kv0 + KVB < kv_hi;
    if (has_next) stage_load(kv0 + KVB);

    // 32-row sub-tiles of the staged kv tile (s_tr only uses low row bits,
    // so each sub-image at rows 32h.. is the same layout at a +32h*D offset)
#pragma unroll
    for (int hf = 0; hf < KVB / 32; ++hf) {
      const int kvh = kv0 + hf * 32;
      if (kvh >= kv_hi) break;  // block-uniform (kv_hi is block-level)
      const __hip_bfloat16* k_lds = smem + buf * TILE + hf * 32 * D;
      const __hip_bfloat16* v_lds = k_lds + KVB * D;

      // S^T = mfma(K, Q); dP^T = mfma(V, dO) — both (r=k_local, c=q_local)
      f32x16 st = {}, dpt = {};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dblk = 0; dblk < DBLK; ++dblk) {
        const int off = rm_swz<D>(lq, dblk * 16 + hi * 8);
        Bf16x8U kf, vf;
        *reinterpret_cast<uint4*>(kf.s) = *reinterpret_cast<const uint4*>(k_lds + off);
        *reinterpret_cast<uint4*>(vf.s) = *reinterpret_cast<const uint4*>(v_lds + off);
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf.v, qf[dblk], st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf.v, dof[dblk], dpt, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      // interior half-tiles: every k row kept for every q row of this wave
      // (wave-uniform branch)
      bool full = kvh + 32 <= Skv;
      if constexpr (MOD == MOD_CAUSAL) {
        full = full && (kvh + 31 <= q0w + q_off);
      } else if constexpr (MOD == MOD_SLIDING_WINDOW) {
        full = full && (kvh + 31 <= q0w + q_off) &&
               ((q0w + 31 + q_off) - kvh < modarg);
      } else if constexpr (MOD == MOD_PREFIX_LM) {
        full = full && ((kvh + 31 <= q0w + q_off) || (kvh + 32 <= modarg));
      } else if constexpr (MOD == MOD_ALIBI) {
        full = false;
      }

      float ds[16];
      if (full) {
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
          const float p = __builtin_amdgcn_exp2f(st[reg] * scale2 - Lq2);
          ds[reg] = p * (dpt[reg] - Dq) * scale;
        }
      } else {
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
          const int k_pos = kvh + acc_row(reg, hi);
          const bool keep = q_valid && attn_keep<MOD>(q_pos, k_pos, Skv, modarg);
          float s2 = st[reg] * scale2;
          if constexpr (MOD == MOD_ALIBI) s2 += slope2 * (k_pos - q_pos);
          const float p = keep ? __builtin_amdgcn_exp2f(s2 - Lq2) : 0.f;
          ds[reg] = p * (dpt[reg] - Dq) * scale;
        }
      }

      bf16x8 da0, da1;
      acc_to_afrag(ds, da0, da1);  // -> dS[32q x 16k] A-fragments
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dc = 0; dc < DCOL; ++dc) {
        f32x16 acc;
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[r] = dq_acc[dc][r];
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          // B = K[16k x 32d]: tr_read from the row-major K image
          bf16x8 kb = tr16_frag<D>(k_lds, ks * 16 + hi * 8, dc * 32, lane);
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ks == 0 ? da0 : da1, kb, acc, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) dq_acc[dc][r] = acc[r];
      }
      __builtin_amdgcn_s_setprio(0);
    }

    // write tile t+1 LAST: the whole iteration hides the global-load flight
    // (buf^1 was last read before the previous barrier)
    if (has_next) stage_write(buf ^ 1);

    __syncthreads();
    buf ^= 1;
  }

  // store dq: element (r=q_local, c=d_local)
  const int dl = lane & 31;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int r = acc_row(reg, hi);
    const int q_r = q0w + r;
    if (q_r >= Sq) continue;
    __hip_bfloat16* dqr = dq + ((long)b * Sq + q_r) * dq_rs + (long)hq * D + dl;
#pragma unroll
    for (int dc = 0; dc < DCOL; ++dc) dqr[dc * 32] = __float2bfloat16(dq_acc[dc][reg]);
  }
}

// ---------------- dK / dV kernels (block per kv-tile, loop GQA group) -------
// WANT_DK: true -> dK (needs dP: V frags + dO rm); false -> dV (needs dO^T).
template <int D, int MOD, bool WANT_DK, int NW>
__global__ __launch_bounds__(NW* WAVE) void attn_bwd_dkv_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, const __hip_bfloat16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    __hip_bfloat16* __restrict__ dkv_out, const float* __restrict__ slopes,
    int B, int Sq, int Skv, int Hq, int Hkv, float scale, int modarg,
    long q_rs, long k_rs, long v_rs, long do_rs, long dkv_rs) {
  constexpr int TPB = NW * WAVE;
  constexpr int KPB = 32 * NW;
  constexpr int DBLK = D / 16;
  constexpr int DCOL = D / 32;
  // q-tile images: Q rm + dO rm, TWO (GQA head, q tile) iterations staged per
  // barrier (sub-tiles s=0,1 at +s*32*D inside each 64-row tensor region).
  // The accumulate B-fragments (Q for dK, dO for dV — k-strided) are
  // tr16_frag reads from the row-major images, so no transposed copy is
  // staged.
  constexpr int TILE = 2 * 64 * D;

  __shared__ __align__(16) __hip_bfloat16 smem[2 * TILE];
  __shared__ float stats_lds[2][2][64];  // [buf][L|D][s*32 + qrow]

  const TileMap tmap = tile_map();
  const int b = tmap.batch, hkv = tmap.head, kvtile = tmap.tile;
  const int group = Hq / Hkv;
  const int tid = threadIdx.x, wave = tid / WAVE, lane = tid % WAVE;
  const int lk = lane & 31, hi = lane >> 5;
  const int kv0w = kvtile * KPB + wave * 32;
  const int krow = kv0w + lk;
  const bool k_valid = krow < Skv;
  const int q_off = Skv - Sq;
  const float scale2 = scale * LOG2E;

  // K (and V for dK) fragments: lane holds row k=lk, 8 d values
  bf16x8 kf[DBLK], vf[WANT_DK ? DBLK : 1];
  {
    const long kr0 = (long)b * Skv + (k_valid ? krow : 0);
    const long khd = (long)hkv * D + hi * 8;
#pragma unroll
    for (int dblk = 0; dblk < DBLK; ++dblk) {
      Bf16x8U ku;
      *reinterpret_cast<uint4*>(ku.s) =
          k_valid ? *reinterpret_cast<const uint4*>(k + kr0 * k_rs + khd + dblk * 16) : uint4{0, 0, 0, 0};
      kf[dblk] = ku.v;
      if constexpr (WANT_DK) {
        Bf16x8U vu;
        *reinterpret_cast<uint4*>(vu.s) =
            k_valid ? *reinterpret_cast<const uint4*>(v + kr0 * v_rs + khd + dblk * 16) : uint4{0, 0, 0, 0};
        vf[dblk] = vu.v;
      }
    }
  }

  float acc_out[DCOL][16];
#pragma unroll
  for (int dc = 0; dc < DCOL; ++dc)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc_out[dc][r] = 0.f;

  int q_lo, q_hi;
  q_range_for_kv<MOD>(kvtile * KPB, KPB, q_off, Sq, modarg, q_lo, q_hi);

  // sub-tiles per barrier: dV runs two (32 MFMAs/barrier, 220-238 VGPR);
  // dK holds V fragments too and spills at NP=2, so it stays at one.
  // dK at NP=2 re-checked in round 2 (post-NU staging): 256 VGPR with 7
  // spills + 32 B scratch AND occupancy halves (2 blocks/CU -> 1) — stays 1.
  constexpr int NP = WANT_DK ? ((D == 64) ? 2 : 1) : 2;  // dK fits NP=2 at D=64

  // T14 staging of two 32-row q tiles: threads 0..255 own Q chunks
  // (2 rows x 8 d), threads 256..511 own dO chunks; each thread carries one
  // chunk pair per sub-tile.
  constexpr int CH_TOT = (32 / 2) * (D / 8);  // 256 chunks per tensor-tile
  // chunk units per thread: 1 at NW=8 (Q half / dO half split across the
  // 512 threads), 2 at NW=4 — each unit decides Q-vs-dO by its global index
  constexpr int NU = (2 * CH_TOT + TPB - 1) / TPB;
  uint4 sreg[NU][2];  // ONE sub-tile's chunk pairs at a time (split-half
                      // staging: load s -> compute s-1 -> write s)

  auto stage_load = [&](int hq_, int q0) {
#pragma unroll
    for (int cu = 0; cu < NU; ++cu) {
      const int u0 = tid + cu * TPB;
      if (u0 >= 2 * CH_TOT) continue;
      const bool isq = u0 < CH_TOT;
      const int u = isq ? u0 : u0 - CH_TOT;
      const int row = (u / (D / 8)) * 2;
      const int d0 = (u % (D / 8)) * 8;
      const __hip_bfloat16* src = isq ? q : dout;
      const long rs_ = isq ? q_rs : do_rs;
      const bool ok0 = q0 + row < Sq;
      const bool ok1 = q0 + row + 1 < Sq;
      const long base = ((long)b * Sq + q0 + row) * rs_ + (long)hq_ * D + d0;
      sreg[cu][0] = ok0 ? *reinterpret_cast<const uint4*>(src + base) : uint4{0, 0, 0, 0};
      sreg[cu][1] = ok1 ? *reinterpret_cast<const uint4*>(src + base + rs_) : uint4{0, 0, 0, 0};
    }
  };
  auto stage_write = [&](int bufsel, int s) {
    __hip_bfloat16* q_lds = smem + bufsel * TILE;
    __hip_bfloat16* do_lds = q_lds + 64 * D;
#pragma unroll
    for (int cu = 0; cu < NU; ++cu) {
      const int u0 = tid + cu * TPB;
      if (u0 >= 2 * CH_TOT) continue;
      const bool isq = u0 < CH_TOT;
      const int u = isq ? u0 : u0 - CH_TOT;
      const int row = (u / (D / 8)) * 2;
      const int d0 = (u % (D / 8)) * 8;
      __hip_bfloat16* dst = (isq ? q_lds : do_lds) + s * 32 * D;
      *reinterpret_cast<uint4*>(dst + rm_swz<D>(row, d0)) = sreg[cu][0];
      *reinterpret_cast<uint4*>(dst + rm_swz<D>(row + 1, d0)) = sreg[cu][1];
    }
  };

  // iterate (GQA head, q tile) pairs with a flat prefetch pipeline
  const int ntiles = (q_hi - q_lo + 31) / 32;
  const int total_iters = group * ntiles;
  if (total_iters == 0) goto store;

  {
    auto iter_to = [&](int it, int& hq_, int& q0) {
      hq_ = hkv * group + it / ntiles;
      q0 = q_lo + (it % ntiles) * 32;
    };
    auto load_stats = [&](int bufsel, int s, int hq_, int q0) {
      if (tid < 32) {
        const int qr = q0 + tid;
        stats_lds[bufsel][0][s * 32 + tid] =
            ((qr < Sq) ? lse[((long)b * Hq + hq_) * Sq + qr] : INFINITY) * LOG2E;
        stats_lds[bufsel][1][s * 32 + tid] =
            (qr < Sq) ? drow[((long)b * Sq + qr) * Hq + hq_] : 0.f;
      }
    };
    int hq0_, q00;
    iter_to(0, hq0_, q00);
    stage_load(hq0_, q00);
    stage_write(0, 0);
    load_stats(0, 0, hq0_, q00);
    if (NP == 2 && total_iters > 1) {
      int hq1_, q01;
      iter_to(1, hq1_, q01);
      stage_load(hq1_, q01);
      stage_write(0, 1);
      load_stats(0, 1, hq1_, q01);
    }
    __syncthreads();

    int buf = 0;
    for (int it0 = 0; it0 < total_iters; it0 += NP) {
      const int npair = min(NP, total_iters - it0);
      const bool has_next = it0 + NP < total_iters;
      const int nleft = total_iters - it0 - NP;  // sub-tiles in the next pair

      // split-half staging: next pair's half h loads during THIS pair's
      // half h compute and is written to buf^1 right after it (buf^1 was
      // last read before the previous barrier, so the write is race-free)
      if (has_next) {
        int nhq, nq0;
        iter_to(it0 + NP, nhq, nq0);
        stage_load(nhq, nq0);
      }

#pragma unroll
      for (int s = 0; s < NP; ++s) {
        if (s >= npair) break;  // block-uniform
        int hq_, q0;
        iter_to(it0 + s, hq_, q0);
        const float slope2 = (MOD == MOD_ALIBI) ? slopes[hq_] * LOG2E : 0.f;
        const __hip_bfloat16* q_lds = smem + buf * TILE + s * 32 * D;
        const __hip_bfloat16* do_lds = smem + buf * TILE + 64 * D + s * 32 * D;

        // S = mfma(Q, K^T): A=Q rm frags from LDS, B = register kf.
        // element (r=q_local, c=k_local)
        f32x16 s_acc = {}, dp_acc = {};
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int dblk = 0; dblk < DBLK; ++dblk) {
          const int off = rm_swz<D>(lk, dblk * 16 + hi * 8);
          Bf16x8U qa;
          *reinterpret_cast<uint4*>(qa.s) = *reinterpret_cast<const uint4*>(q_lds + off);
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa.v, kf[dblk], s_acc, 0, 0, 0);
          if constexpr (WANT_DK) {
            Bf16x8U da;
            *reinterpret_cast<uint4*>(da.s) = *reinterpret_cast<const uint4*>(do_lds + off);
            dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da.v, vf[dblk], dp_acc, 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);

        // interior (q tile entirely below this wave's k rows): mask-free
        bool full = q0 + 31 < Sq;
        if constexpr (MOD == MOD_CAUSAL) {
          full = full && (q0 + q_off >= kv0w + 31);
        } else if constexpr (MOD == MOD_SLIDING_WINDOW) {
          full = full && (q0 + q_off >= kv0w + 31) &&
                 ((q0 + 31 + q_off) - kv0w < modarg);
        } else if constexpr (MOD == MOD_PREFIX_LM) {
          // second clause wave-uniform: whole wave's k rows inside the prefix
          full = full && ((q0 + q_off >= kv0w + 31) || (kv0w + 31 < modarg));
        } else {
          full = false;  // MOD_NONE boundary Sq checks + ALIBI slope
        }

        float pv[16];
        if (full) {
#pragma unroll
          for (int reg = 0; reg < 16; ++reg) {
            const int qr = s * 32 + acc_row(reg, hi);
            const float p = __builtin_amdgcn_exp2f(s_acc[reg] * scale2 - stats_lds[buf][0][qr]);
            if constexpr (WANT_DK) {
              pv[reg] = p * (dp_acc[reg] - stats_lds[buf][1][qr]) * scale;  // dS
            } else {
              pv[reg] = p;
            }
          }
        } else {
#pragma unroll
          for (int reg = 0; reg < 16; ++reg) {
            const int qr = acc_row(reg, hi);
            const int q_r = q0 + qr;
            const int q_pos = q_r + q_off;
            const int k_pos = krow;
            const bool keep = k_valid && q_r < Sq && attn_keep<MOD>(q_pos, k_pos, Skv, modarg);
            float s2 = s_acc[reg] * scale2;
            if constexpr (MOD == MOD_ALIBI) s2 += slope2 * (k_pos - q_pos);
            const float p =
                keep ? __builtin_amdgcn_exp2f(s2 - stats_lds[buf][0][s * 32 + qr]) : 0.f;
            if constexpr (WANT_DK) {
              pv[reg] = p * (dp_acc[reg] - stats_lds[buf][1][s * 32 + qr]) * scale;  // dS
            } else {
              pv[reg] = p;
            }
          }
        }

        // transform: acc holds M[r=q][c=k]; A-frags of M^T = dS^T (dK) / P^T (dV)
        bf16x8 a0, a1;
        acc_to_afrag(pv, a0, a1);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int dc = 0; dc < DCOL; ++dc) {
          f32x16 acc;
#pragma unroll
          for (int r = 0; r < 16; ++r) acc[r] = acc_out[dc][r];
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            // B = X[16q x 32d] (X = Q for dK, dO for dV): tr_read from rm image
            bf16x8 xb = tr16_frag<D>(WANT_DK ? q_lds : do_lds, ks * 16 + hi * 8,
                                     dc * 32, lane);
            acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ks == 0 ? a0 : a1, xb, acc, 0, 0, 0);
          }
#pragma unroll
          for (int r = 0; r < 16; ++r) acc_out[dc][r] = acc[r];
        }
        __builtin_amdgcn_s_setprio(0);

        // split-half staging epilogue: write the half that loaded during
        // this half's compute, then start the next half's load
        if (has_next) {
          int nhq, nq0;
          if (s == 0) {
            iter_to(it0 + NP, nhq, nq0);
            stage_write(buf ^ 1, 0);
            load_stats(buf ^ 1, 0, nhq, nq0);
            if (NP == 2 && nleft > 1) {
              iter_to(it0 + NP + 1, nhq, nq0);
              stage_load(nhq, nq0);
            }
          } else if (nleft > 1) {
            iter_to(it0 + NP + 1, nhq, nq0);
            stage_write(buf ^ 1, 1);
            load_stats(buf ^ 1, 1, nhq, nq0);
          }
        }
      }

      __syncthreads();
      buf ^= 1;
    }
  }

store:
  // store: element (r=k_local, c=d_local)
  {
    const int dl = lane & 31;
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int r = acc_row(reg, hi);
      const int k_r = kv0w + r;
      if (k_r >= Skv) continue;
      __hip_bfloat16* out = dkv_out + ((long)b * Skv + k_r) * dkv_rs + (long)hkv * D + dl;
#pragma unroll
      for (int dc = 0; dc < DCOL; ++dc) out[dc * 32] = __float2bfloat16(acc_out[dc][reg]);
    }
  }
}

template <int D, int MOD, int NW>
void launch_bwd_all(dim3 gq, dim3 gkv, dim3 block, hipStream_t stream,
                    const __hip_bfloat16* q, const __hip_bfloat16* k, const __hip_bfloat16* v,
                    const __hip_bfloat16* dout, const float* lse, const float* drow,
                    __hip_bfloat16* dq, __hip_bfloat16* dk, __hip_bfloat16* dv,
                    const float* slopes, int B, int Sq, int Skv, int Hq, int Hkv,
                    float scale, int modarg, long q_rs, long k_rs, long v_rs, long do_rs,
                    long dq_rs, long dk_rs, long dv_rs) {
  attn_bwd_dq_kernel<D, MOD, NW><<<gq, block, 0, stream>>>(
      q, k, v, dout, lse, drow, dq, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs, do_rs, dq_rs);
  attn_bwd_dkv_kernel<D, MOD, true, NW><<<gkv, block, 0, stream>>>(
      q, k, v, dout, lse, drow, dk, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs, do_rs, dk_rs);
  attn_bwd_dkv_kernel<D, MOD, false, NW><<<gkv, block, 0, stream>>>(
      q, k, v, dout, lse, drow, dv, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs, do_rs, dv_rs);
}

template <int D, int MOD>
void launch_all_nw(dim3 gq, dim3 gkv, dim3 block, hipStream_t stream,
                   const __hip_bfloat16* q, const __hip_bfloat16* k, const __hip_bfloat16* v,
                   const __hip_bfloat16* dout, const float* lse, const float* drow,
                   __hip_bfloat16* dq, __hip_bfloat16* dk, __hip_bfloat16* dv,
                   const float* slopes, int B, int Sq, int Skv, int Hq, int Hkv,
                   float scale, int modarg, long q_rs, long k_rs, long v_rs, long do_rs,
                   long dq_rs, long dk_rs, long dv_rs) {
  if (bwd_nw() == 16)
    launch_bwd_all<D, MOD, 16>(gq, gkv, block, stream, q, k, v, dout, lse, drow, dq, dk, dv,
                               slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs,
                               do_rs, dq_rs, dk_rs, dv_rs);
  else if (bwd_nw() == 4)
    launch_bwd_all<D, MOD, 4>(gq, gkv, block, stream, q, k, v, dout, lse, drow, dq, dk, dv,
                              slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs,
                              do_rs, dq_rs, dk_rs, dv_rs);
  else
    launch_bwd_all<D, MOD, 8>(gq, gkv, block, stream, q, k, v, dout, lse, drow, dq, dk, dv,
                              slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs,
                              do_rs, dq_rs, dk_rs, dv_rs);
}

template <int D>
void launch_bwd_mod(int mod, dim3 gq, dim3 gkv, dim3 block, hipStream_t stream,
                    const __hip_bfloat16* q, const __hip_bfloat16* k, const __hip_bfloat16* v,
                    const __hip_bfloat16* dout, const float* lse, const float* drow,
                    __hip_bfloat16* dq, __hip_bfloat16* dk, __hip_bfloat16* dv,
                    const float* slopes, int B, int Sq, int Skv, int Hq, int Hkv,
                    float scale, int modarg, long q_rs, long k_rs, long v_rs, long do_rs,
                    long dq_rs, long dk_rs, long dv_rs) {
  switch (mod) {
    case MOD_NONE:
      launch_all_nw<D, MOD_NONE>(gq, gkv, block, stream, q, k, v, dout, lse, drow, dq, dk, dv, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs, do_rs, dq_rs, dk_rs, dv_rs);
      break;
    case MOD_CAUSAL:
      launch_all_nw<D, MOD_CAUSAL>(gq, gkv, block, stream, q, k, v, dout, lse, drow, dq, dk, dv, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs, do_rs, dq_rs, dk_rs, dv_rs);
      break;
    case MOD_SLIDING_WINDOW:
      launch_all_nw<D, MOD_SLIDING_WINDOW>(gq, gkv, block, stream, q, k, v, dout, lse, drow, dq, dk, dv, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs, do_rs, dq_rs, dk_rs, dv_rs);
      break;
    case MOD_PREFIX_LM:
      launch_all_nw<D, MOD_PREFIX_LM>(gq, gkv, block, stream, q, k, v, dout, lse, drow, dq, dk, dv, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs, do_rs, dq_rs, dk_rs, dv_rs);
      break;
    case MOD_ALIBI:
      launch_all_nw<D, MOD_ALIBI>(gq, gkv, block, stream, q, k, v, dout, lse, drow, dq, dk, dv, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs, do_rs, dq_rs, dk_rs, dv_rs);
      break;
    default:
      TORCH_CHECK(false, "attn_bwd: unknown mod ", mod);
  }
}

}  // namespace

std::vector<at::Tensor> attn_bwd_out(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o,
                                     at::Tensor dout, at::Tensor lse, double scale, long mod,
                                     long modarg, at::Tensor slopes,
                                     at::Tensor dq, at::Tensor dk, at::Tensor dv) {
  TORCH_CHECK(q.is_cuda());
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attn_bwd: bf16 only");
  auto rs = [](at::Tensor& t) {
    const int S = t.size(1), H = t.size(2), D = t.size(3);
    if (!(t.stride(3) == 1 && t.stride(2) == D && t.stride(0) == (long)S * t.stride(1)))
      t = t.contiguous();
    return t.stride(1);
  };
  const long q_rs = rs(q), k_rs = rs(k), v_rs = rs(v), do_rs = rs(dout);
  o = o.contiguous();
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1), Hkv = k.size(2);
  TORCH_CHECK(D == 64 || D == 128, "attn_bwd: head_dim must be 64 or 128");
  // outputs may be pre-allocated row-strided views (slices of a fused dQKV
  // grad buffer) — element (b,s,h,d) at (b*S+s)*stride(1) + h*D + d
  auto check_out = [&](at::Tensor& t, int S_, int H_) {
    if (t.numel() == 0) {
      t = at::empty({B, S_, H_, D}, q.options());
    } else {
      TORCH_CHECK(t.size(0) == B && t.size(1) == S_ && t.size(2) == H_ && t.size(3) == D &&
                      t.stride(3) == 1 && t.stride(2) == D &&
                      t.stride(0) == (long)S_ * t.stride(1),
                  "attn_bwd: bad out view");
    }
    return t.stride(1);
  };
  const long dq_rs = check_out(dq, Sq, Hq);
  const long dk_rs = check_out(dk, Skv, Hkv);
  const long dv_rs = check_out(dv, Skv, Hkv);
  auto drow = at::empty({(long)B * Sq * Hq}, q.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();

  {  // preprocess: 4 rows per wave at D=128 (vectorized 16 B loads)
    const long rows = (long)B * Sq * Hq;
    const int rpw = WAVE / (D / 8);
    const long grid = cdiv(rows, (long)(256 / WAVE) * rpw);
    bwd_preprocess_kernel<<<grid, 256, 0, stream>>>(
        reinterpret_cast<const __hip_bfloat16*>(dout.data_ptr()),
        reinterpret_cast<const __hip_bfloat16*>(o.data_ptr()),
        drow.data_ptr<float>(), rows, D, do_rs, Hq);
  }

  const int qpb = 32 * bwd_nw();
  dim3 gq(cdiv(Sq, qpb), Hq, B);
  dim3 gkv(cdiv(Skv, qpb), Hkv, B);
  dim3 block(bwd_nw() * WAVE);
  const float* sl = slopes.numel() > 0 ? slopes.data_ptr<float>() : nullptr;
  auto* qp = reinterpret_cast<const __hip_bfloat16*>(q.data_ptr());
  auto* kp = reinterpret_cast<const __hip_bfloat16*>(k.data_ptr());
  auto* vp = reinterpret_cast<const __hip_bfloat16*>(v.data_ptr());
  auto* dop = reinterpret_cast<const __hip_bfloat16*>(dout.data_ptr());
  auto* dqp = reinterpret_cast<__hip_bfloat16*>(dq.data_ptr());
  auto* dkp = reinterpret_cast<__hip_bfloat16*>(dk.data_ptr());
  auto* dvp = reinterpret_cast<__hip_bfloat16*>(dv.data_ptr());
  if (D == 64)
    launch_bwd_mod<64>((int)mod, gq, gkv, block, stream, qp, kp, vp, dop,
                       lse.data_ptr<float>(), drow.data_ptr<float>(), dqp, dkp, dvp, sl,
                       B, Sq, Skv, Hq, Hkv, (float)scale, (int)modarg, q_rs, k_rs, v_rs, do_rs,
                       dq_rs, dk_rs, dv_rs);
  else
    launch_bwd_mod<128>((int)mod, gq, gkv, block, stream, qp, kp, vp, dop,
                        lse.data_ptr<float>(), drow.data_ptr<float>(), dqp, dkp, dvp, sl,
                        B, Sq, Skv, Hq, Hkv, (float)scale, (int)modarg, q_rs, k_rs, v_rs, do_rs,
                        dq_rs, dk_rs, dv_rs);
  return {dq, dk, dv};
}

std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o,
                                 at::Tensor dout, at::Tensor lse, double scale, long mod,
                                 long modarg, at::Tensor slopes) {
  auto none = at::empty({0}, q.options());
  return attn_bwd_out(q, k, v, o, dout, lse, scale, mod, modarg, slopes,
                      none, none.clone(), none.clone());
}
