// Tiled FlashAttention-2 forward (kernel K1/K2).
// Replaces the reference's naive softmax(QK^T)V Python compositions
// (/root/reference/models/attention/flash_attention.py:78-156 — "Simple
// approach without tiling for now" — and simple_attention.py, flex_attention.py).
//
// Design (gfx950): 4 waves/block, each wave owns 32 q rows (128/block);
// KV tiles of 32 staged cooperatively in LDS (K row-major for A-fragments,
// V transposed for B-fragments); swapped QK^T (see attn_common.h) keeps the
// online softmax lane-local; GQA reads the shared KV head directly (no
// repeat); causal/sliding-window tiles are skipped at block level.
// BSHD layout: q [B,Sq,Hq,D], k/v [B,Skv,Hkv,D], o [B,Sq,Hq,D], lse [B,Hq,Sq].
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "attn_common.h"

namespace {

constexpr int KVB = 32;   // kv tile
constexpr int QPW = 32;   // q rows per wave
constexpr int NW = 4;     // waves per block
constexpr int QPB = QPW * NW;  // q rows per block

template <int D, int MOD>
__global__ __launch_bounds__(NW * WAVE) void attn_fwd_kernel(
    const __hip_bfloat16* __restrict__ q, const __hip_bfloat16* __restrict__ k,
    const __hip_bfloat16* __restrict__ v, __hip_bfloat16* __restrict__ o,
    float* __restrict__ lse, const float* __restrict__ slopes,
    int B, int Sq, int Skv, int Hq, int Hkv, float scale, int modarg,
    long q_rs, long k_rs, long v_rs) {
  constexpr int DBLK = D / 16;   // QK^T k-slots (over d)
  constexpr int DCOL = D / 32;   // PV output column tiles
  constexpr int KPAD = 8;        // elements of row padding (16B) against bank conflicts
  constexpr int KSTR = D + KPAD;
  constexpr int VSTR = KVB + 8;

  __shared__ __hip_bfloat16 smem[KVB * KSTR + D * VSTR];
  __hip_bfloat16* k_lds = smem;            // [KVB][KSTR] row-major
  __hip_bfloat16* vt_lds = smem + KVB * KSTR;  // [D][VSTR] transposed V

  const int b = blockIdx.z;
  const int hq = blockIdx.y;
  const int qtile = blockIdx.x;
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

  // ---- Q fragments (B-operand: lane holds Q[q=lq][16*dblk + 8*hi + j]) ----
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

  // ---- kv range for this block ----
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
  const float slope = (MOD == MOD_ALIBI) ? slopes[hq] : 0.f;

  float m = -INFINITY, l = 0.f;
  float o_acc[DCOL][16];
#pragma unroll
  for (int dc = 0; dc < DCOL; ++dc)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[dc][r] = 0.f;

  for (int kv0 = kv_lo; kv0 < kv_hi; kv0 += KVB) {
    // ---- cooperative staging: K row-major, V transposed ----
    {
      constexpr int U4ROW = D / 8;                      // uint4 per row
      constexpr int TOT = KVB * U4ROW;                  // uint4 per tile
      for (int u = tid; u < TOT; u += NW * WAVE) {
        const int row = u / U4ROW;
        const int d0 = (u % U4ROW) * 8;
        const bool valid = kv0 + row < Skv;
        const long r0 = (long)b * Skv + (valid ? kv0 + row : 0);
        const long hd = (long)hkv * D + d0;
        Bf16x8U kv_u;
        *reinterpret_cast<uint4*>(kv_u.s) =
            valid ? *reinterpret_cast<const uint4*>(k + r0 * k_rs + hd) : uint4{0, 0, 0, 0};
        *reinterpret_cast<uint4*>(k_lds + row * KSTR + d0) = *reinterpret_cast<uint4*>(kv_u.s);
        Bf16x8U vv;
        *reinterpret_cast<uint4*>(vv.s) =
            valid ? *reinterpret_cast<const uint4*>(v + r0 * v_rs + hd) : uint4{0, 0, 0, 0};
#pragma unroll
        for (int j = 0; j < 8; ++j) vt_lds[(d0 + j) * VSTR + row] = vv.h[j];
      }
    }
    __syncthreads();

    // ---- S^T = mfma(K, Q): element (r=k_local, c=q_local) ----
    f32x16 st = {};
#pragma unroll
    for (int dblk = 0; dblk < DBLK; ++dblk) {
      Bf16x8U kf;
      *reinterpret_cast<uint4*>(kf.s) =
          *reinterpret_cast<const uint4*>(k_lds + lq * KSTR + dblk * 16 + hi * 8);
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf.v, qf[dblk], st, 0, 0, 0);
    }

    // ---- scale + mask (+alibi) ----
    float p[16];
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int k_pos = kv0 + acc_row(reg, hi);
      const bool keep = q_valid && attn_keep<MOD>(q_pos, k_pos, Skv, modarg);
      float s = st[reg] * scale;
      if constexpr (MOD == MOD_ALIBI) s += slope * (k_pos - q_pos);
      p[reg] = keep ? s : -INFINITY;
    }

    // ---- online softmax (lane-local in q) ----
    float tmax = p[0];
#pragma unroll
    for (int reg = 1; reg < 16; ++reg) tmax = fmaxf(tmax, p[reg]);
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32));
    const float m_new = fmaxf(m, tmax);
    const float mc = fmaxf(m_new, -1e30f);  // clamp only inside exponentials
    const float alpha = __expf(m - mc);      // m=-inf -> 0 on the first live tile
    float psum = 0.f;
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      p[reg] = __expf(p[reg] - mc);  // -inf -> 0
      psum += p[reg];
    }
    psum += __shfl_xor(psum, 32);
    l = l * alpha + psum;
    m = m_new;

    // rescale o_acc rows (alpha gathered per accumulator row)
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const float ar = __shfl(alpha, acc_row(reg, hi));
#pragma unroll
      for (int dc = 0; dc < DCOL; ++dc) o_acc[dc][reg] *= ar;
    }

    // ---- P -> A fragments, PV ----
    bf16x8 pa0, pa1;
    acc_to_afrag(p, pa0, pa1);
#pragma unroll
    for (int dc = 0; dc < DCOL; ++dc) {
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = o_acc[dc][r];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        Bf16x8U vf;
        *reinterpret_cast<uint4*>(vf.s) = *reinterpret_cast<const uint4*>(
            vt_lds + (dc * 32 + lq) * VSTR + ks * 16 + hi * 8);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ks == 0 ? pa0 : pa1, vf.v, acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[dc][r] = acc[r];
    }
    __syncthreads();
  }

  // ---- epilogue ----
  if (q_valid && hi == 0)
    lse[((long)b * Hq + hq) * Sq + qrow] = m + __logf(l);

  // o_acc element (r=q_local, c=d_local): lane&31 = d_local here.
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

template <int D>
void launch_fwd(int mod, dim3 grid, dim3 block, hipStream_t stream,
                const __hip_bfloat16* q, const __hip_bfloat16* k, const __hip_bfloat16* v,
                __hip_bfloat16* o, float* lse, const float* slopes,
                int B, int Sq, int Skv, int Hq, int Hkv, float scale, int modarg,
                long q_rs, long k_rs, long v_rs) {
  switch (mod) {
    case MOD_NONE:
      attn_fwd_kernel<D, MOD_NONE><<<grid, block, 0, stream>>>(q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    case MOD_CAUSAL:
      attn_fwd_kernel<D, MOD_CAUSAL><<<grid, block, 0, stream>>>(q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    case MOD_SLIDING_WINDOW:
      attn_fwd_kernel<D, MOD_SLIDING_WINDOW><<<grid, block, 0, stream>>>(q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    case MOD_PREFIX_LM:
      attn_fwd_kernel<D, MOD_PREFIX_LM><<<grid, block, 0, stream>>>(q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
      break;
    case MOD_ALIBI:
      attn_fwd_kernel<D, MOD_ALIBI><<<grid, block, 0, stream>>>(q, k, v, o, lse, slopes, B, Sq, Skv, Hq, Hkv, scale, modarg, q_rs, k_rs, v_rs);
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
  dim3 grid(cdiv(Sq, QPB), Hq, B);
  dim3 block(NW * WAVE);
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
