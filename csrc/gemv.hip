// bf16 GEMV for the decode path: y[b,n] = dot(W[n,:], x[b,:]).
// M=1 projections (qkv/wo/gate_up/down/lm_head at batch 1) are pure weight
// streaming — hipBLASLt's M=1 kernels measured ~0.87 TB/s on the captured
// decode graph; this kernel streams W coalesced (uint4 lanes) with x staged
// in LDS and targets the ~6 TB/s HBM bound.
//
// W: [N, K] row-major (torch nn.Linear weight layout), x: [B, K], y: [B, N].
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// nontemporal builtin needs a native vector type (uint4 is a class)
typedef unsigned int u32x4_nt __attribute__((ext_vector_type(4)));
__device__ __forceinline__ uint4 nt_load_u4(const void* p) {
  u32x4_nt t = __builtin_nontemporal_load(reinterpret_cast<const u32x4_nt*>(p));
  return *reinterpret_cast<const uint4*>(&t);
}

// block = 256 (4 waves). Each wave owns one output row per pass; lanes cover
// 512 elements per pass (64 lanes x 8). Rows grid-strided.
__global__ __launch_bounds__(256) void gemv_bf16_kernel(
    const __hip_bfloat16* __restrict__ W, const __hip_bfloat16* __restrict__ x,
    __hip_bfloat16* __restrict__ y, int N, int K, int B) {
  extern __shared__ __hip_bfloat16 x_lds[];
  const int b = blockIdx.y;
  const int tid = threadIdx.x;
  // stage x[b,:] (vectorized)
  {
    const uint4* xv = reinterpret_cast<const uint4*>(x + (long)b * K);
    uint4* xl = reinterpret_cast<uint4*>(x_lds);
    for (int i = tid; i < K / 8; i += 256) xl[i] = xv[i];
    for (int i = (K / 8) * 8 + tid; i < K; i += 256) x_lds[i] = x[(long)b * K + i];
  }
  __syncthreads();

  const int wid = tid / WAVE, lane = tid % WAVE;
  for (int row = blockIdx.x * 4 + wid; row < N; row += gridDim.x * 4) {
    const __hip_bfloat16* wr = W + (long)row * K;
    float acc = 0.f;
    int k = lane * 8;
    // deep-unrolled NON-TEMPORAL weight stream (guide: decode weights are
    // read once per token per CU — keep >=4 loads in flight, nt policy;
    // a single in-flight uint4 leaves the loop latency-bound at ~25% BW)
    for (; k + 8 * WAVE * 3 + 8 <= K; k += WAVE * 8 * 4) {
      U4 wv[4], xv[4];
#pragma unroll
      for (int u = 0; u < 4; ++u)
        wv[u].u = nt_load_u4(wr + k + u * WAVE * 8);
#pragma unroll
      for (int u = 0; u < 4; ++u)
        xv[u].u = *reinterpret_cast<const uint4*>(x_lds + k + u * WAVE * 8);
#pragma unroll
      for (int u = 0; u < 4; ++u)
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc += bf16_bits_to_f32(wv[u].s[j]) * bf16_bits_to_f32(xv[u].s[j]);
    }
    for (; k + 8 <= K; k += WAVE * 8) {
      U4 wv, xv;
      wv.u = nt_load_u4(wr + k);
      xv.u = *reinterpret_cast<const uint4*>(x_lds + k);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc += bf16_bits_to_f32(wv.s[j]) * bf16_bits_to_f32(xv.s[j]);
    }
    for (; k < K; ++k) acc += to_f32(wr[k]) * to_f32(x_lds[k]);
    acc = wave_reduce_sum(acc);
    if (lane == 0) from_f32(&y[(long)b * N + row], acc);
  }
}

// ---- fused decode GEMV ----
// Staging modes fold the op PRECEDING the projection into the x->LDS copy
// (each block redoes the tiny reduction — x is a few KB, the W stream is the
// cost): MODE_RMSNORM stages rmsnorm(x)*nw, MODE_SWIGLU stages
// silu(gate)*up from a [2I] input. The optional residual epilogue folds the
// op FOLLOWING it: y = W@x_staged + res. Cuts ~5 kernels per decode layer.
#define GMODE_PLAIN 0
#define GMODE_RMSNORM 1
#define GMODE_SWIGLU 2

template <int MODE, int BB>
__global__ __launch_bounds__(256) void gemv_ex_kernel(
    const __hip_bfloat16* __restrict__ W, const __hip_bfloat16* __restrict__ x,
    const __hip_bfloat16* __restrict__ nw, const __hip_bfloat16* __restrict__ res,
    __hip_bfloat16* __restrict__ y, int N, int K, int Kin, float eps, int B) {
  // BB = compile-time batch: all BB x-vectors staged in LDS, W streamed ONCE
  // with BB accumulators per output row (the whole point of batching — a
  // per-row grid.y launch re-reads W B times).
  extern __shared__ __hip_bfloat16 x_lds[];  // [BB][K]
  __shared__ float scratch[256 / WAVE];
  const int tid = threadIdx.x;

#pragma unroll
  for (int b = 0; b < BB; ++b) {
    if (b >= B) break;
    const __hip_bfloat16* xb = x + (long)b * Kin;
    __hip_bfloat16* xl = x_lds + (long)b * K;
    if constexpr (MODE == GMODE_RMSNORM) {
      float ss = 0.f;
      for (int i = tid; i < K / 8; i += 256) {
        U4 u; u.u = reinterpret_cast<const uint4*>(xb)[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) { float f = bf16_bits_to_f32(u.s[j]); ss += f * f; }
      }
      ss = block_reduce_sum<256>(ss, scratch);
      const float r = rsqrtf(ss / K + eps);
      for (int i = tid; i < K / 8; i += 256) {
        U4 u, w8, o;
        u.u = reinterpret_cast<const uint4*>(xb)[i];
        w8.u = reinterpret_cast<const uint4*>(nw)[i];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o.s[j] = f32_to_bf16_bits(bf16_bits_to_f32(u.s[j]) * r * bf16_bits_to_f32(w8.s[j]));
        reinterpret_cast<uint4*>(xl)[i] = o.u;
      }
    } else if constexpr (MODE == GMODE_SWIGLU) {
      for (int i = tid; i < K / 8; i += 256) {
        U4 g, u, o;
        g.u = reinterpret_cast<const uint4*>(xb)[i];
        u.u = reinterpret_cast<const uint4*>(xb + K)[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float gf = bf16_bits_to_f32(g.s[j]);
          const float uf = bf16_bits_to_f32(u.s[j]);
          o.s[j] = f32_to_bf16_bits(gf / (1.f + __expf(-gf)) * uf);
        }
        reinterpret_cast<uint4*>(xl)[i] = o.u;
      }
    } else {
      for (int i = tid; i < K / 8; i += 256)
        reinterpret_cast<uint4*>(xl)[i] = reinterpret_cast<const uint4*>(xb)[i];
    }
  }
  __syncthreads();

  const int wid = tid / WAVE, lane = tid % WAVE;
  for (int row = blockIdx.x * 4 + wid; row < N; row += gridDim.x * 4) {
    const __hip_bfloat16* wr = W + (long)row * K;
    float acc[BB];
#pragma unroll
    for (int b = 0; b < BB; ++b) acc[b] = 0.f;
    int k = lane * 8;
    constexpr int UNR2 = (BB > 2) ? 2 : 4;  // keep several nt loads in flight
    for (; k + 8 * WAVE * (UNR2 - 1) + 8 <= K; k += WAVE * 8 * UNR2) {
      U4 wv[UNR2];
#pragma unroll
      for (int u = 0; u < UNR2; ++u)
        wv[u].u = nt_load_u4(wr + k + u * WAVE * 8);
#pragma unroll
      for (int u = 0; u < UNR2; ++u)
#pragma unroll
        for (int b = 0; b < BB; ++b) {
          U4 xv;
          xv.u = *reinterpret_cast<const uint4*>(x_lds + (long)b * K + k + u * WAVE * 8);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[b] += bf16_bits_to_f32(wv[u].s[j]) * bf16_bits_to_f32(xv.s[j]);
        }
    }
    for (; k + 8 <= K; k += WAVE * 8) {
      U4 wv;
      wv.u = nt_load_u4(wr + k);
#pragma unroll
      for (int b = 0; b < BB; ++b) {
        U4 xv;
        xv.u = *reinterpret_cast<const uint4*>(x_lds + (long)b * K + k);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          acc[b] += bf16_bits_to_f32(wv.s[j]) * bf16_bits_to_f32(xv.s[j]);
      }
    }
#pragma unroll
    for (int b = 0; b < BB; ++b) {
      if (b >= B) break;
      float a = wave_reduce_sum(acc[b]);
      if (lane == 0) {
        if (res != nullptr) a += to_f32(res[(long)b * N + row]);
        from_f32(&y[(long)b * N + row], a);
      }
    }
  }
}

}  // namespace

at::Tensor gemv_ex(at::Tensor x, at::Tensor W, long mode, at::Tensor nw, double eps,
                   at::Tensor res) {
  // x: [B, Kin]; W: [N, K]; mode: 0 plain (Kin==K), 1 rmsnorm (Kin==K,
  // nw [K]), 2 swiglu (Kin==2K). res: [B, N] or empty. B <= 8.
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && W.is_contiguous());
  const int K = W.size(1), N = W.size(0);
  auto xc = x.contiguous().view({-1, x.size(-1)});
  const int Kin = xc.size(1);
  TORCH_CHECK(K % 8 == 0, "gemv_ex: K must be a multiple of 8");
  TORCH_CHECK((mode == 2 && Kin == 2 * K) || (mode != 2 && Kin == K), "gemv_ex: bad Kin");
  const int B = xc.size(0);
  TORCH_CHECK(B <= 8, "gemv_ex: B must be <= 8");
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto y = at::empty({B, N}, x.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const int grid_x = std::min(cdiv(N, 4), 2048);
  auto* Wp = reinterpret_cast<const __hip_bfloat16*>(W.data_ptr());
  auto* np = mode == 1 ? reinterpret_cast<const __hip_bfloat16*>(nw.contiguous().data_ptr())
                       : nullptr;
  auto rc = res.numel() > 0 ? res.contiguous() : res;
  // BB caps at 4: beyond that the inner loop turns LDS/VALU-bound (measured
  // at B=8); rows 5..8 run as two <=4-row chunks (W streamed once per chunk)
  for (int b0 = 0; b0 < B; b0 += 4) {
    const int Bc = std::min(4, B - b0);
    const int BB = Bc <= 1 ? 1 : (Bc <= 2 ? 2 : 4);
    const size_t lds = (size_t)BB * K * sizeof(__hip_bfloat16);
    TORCH_CHECK(lds <= 160 * 1024, "gemv_ex: K too large for LDS staging");
    auto* xp = reinterpret_cast<const __hip_bfloat16*>(xc.data_ptr()) + (long)b0 * Kin;
    auto* rp = rc.numel() > 0
                   ? reinterpret_cast<const __hip_bfloat16*>(rc.data_ptr()) + (long)b0 * N
                   : nullptr;
    auto* yp = reinterpret_cast<__hip_bfloat16*>(y.data_ptr()) + (long)b0 * N;
    const int Bv = Bc;
#define GLAUNCH(M_, B_) \
  gemv_ex_kernel<M_, B_><<<grid_x, 256, lds, stream>>>(Wp, xp, np, rp, yp, N, K, Kin, (float)eps, Bv)
#define GMODE_SWITCH(B_)                                   \
  switch (mode) {                                          \
    case 0: GLAUNCH(GMODE_PLAIN, B_); break;               \
    case 1: GLAUNCH(GMODE_RMSNORM, B_); break;             \
    case 2: GLAUNCH(GMODE_SWIGLU, B_); break;              \
    default: TORCH_CHECK(false, "gemv_ex: bad mode");      \
  }
    switch (BB) {
      case 1: GMODE_SWITCH(1); break;
      case 2: GMODE_SWITCH(2); break;
      case 4: GMODE_SWITCH(4); break;
    }
  }
#undef GMODE_SWITCH
#undef GLAUNCH
  return y.view(sizes);
}

at::Tensor gemv_bf16(at::Tensor x, at::Tensor W) {
  // x: [B, K] (or any shape collapsing to [B, K]); W: [N, K] row-major.
  TORCH_CHECK(x.is_cuda() && W.is_cuda());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && W.scalar_type() == at::kBFloat16);
  TORCH_CHECK(W.dim() == 2 && W.is_contiguous(), "gemv: W must be [N,K] contiguous");
  const int K = W.size(1), N = W.size(0);
  auto xc = x.contiguous().view({-1, K});
  const int B = xc.size(0);
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto y = at::empty({B, N}, x.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const int grid_x = std::min(cdiv(N, 4), 2048);
  const size_t lds = (size_t)K * sizeof(__hip_bfloat16);
  TORCH_CHECK(lds <= 160 * 1024, "gemv: K too large for LDS staging");
  gemv_bf16_kernel<<<dim3(grid_x, B), 256, lds, stream>>>(
      reinterpret_cast<const __hip_bfloat16*>(W.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(xc.data_ptr()),
      reinterpret_cast<__hip_bfloat16*>(y.data_ptr()), N, K, B);
  return y.view(sizes);
}
