// Fused cross-entropy over the vocab with pad masking (kernel K6).
// Replaces /root/reference/core/training.py:1222-1234. One workgroup per row;
// two passes (max, then sum-exp) over the bf16 logits with fp32 accumulation —
// the fp32 logits are never materialized. Backward writes
// dlogits = (softmax - onehot) * scale in one streaming pass.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

template <typename T, int BLOCK>
__global__ void ce_fwd_kernel(const T* __restrict__ logits, const long* __restrict__ targets,
                              float* __restrict__ loss_sum, long* __restrict__ ntok,
                              float* __restrict__ lse_out, long rows, int V, long ignore_index) {
  __shared__ float scratch[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * (long)V;
    // pass 1: max
    float m = -INFINITY;
    if constexpr (sizeof(T) == 2) {
      const int VV = V / 8;
      const uint4* lv = reinterpret_cast<const uint4*>(lr);
      for (int i = threadIdx.x; i < VV; i += BLOCK) {
        U4 u; u.u = lv[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) m = fmaxf(m, bf16_bits_to_f32(u.s[j]));
      }
      for (int i = VV * 8 + threadIdx.x; i < V; i += BLOCK) m = fmaxf(m, to_f32(lr[i]));
    } else {
      for (int i = threadIdx.x; i < V; i += BLOCK) m = fmaxf(m, to_f32(lr[i]));
    }
    m = block_reduce_max<BLOCK>(m, scratch);
    // pass 2: sum exp
    float s = 0.f;
    if constexpr (sizeof(T) == 2) {
      const int VV = V / 8;
      const uint4* lv = reinterpret_cast<const uint4*>(lr);
      for (int i = threadIdx.x; i < VV; i += BLOCK) {
        U4 u; u.u = lv[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) s += __expf(bf16_bits_to_f32(u.s[j]) - m);
      }
      for (int i = VV * 8 + threadIdx.x; i < V; i += BLOCK) s += __expf(to_f32(lr[i]) - m);
    } else {
      for (int i = threadIdx.x; i < V; i += BLOCK) s += __expf(to_f32(lr[i]) - m);
    }
    s = block_reduce_sum<BLOCK>(s, scratch);
    const float lse = m + __logf(s);
    if (threadIdx.x == 0) {
      lse_out[row] = lse;
      const long t = targets[row];
      if (t != ignore_index) {
        atomicAdd(loss_sum, lse - to_f32(lr[t]));
        atomicAdd(reinterpret_cast<unsigned long long*>(ntok), 1ull);
      }
    }
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits, const long* __restrict__ targets,
                              const float* __restrict__ lse, const float* __restrict__ scale,
                              T* __restrict__ dlogits, long rows, int V, long ignore_index) {
  const float sc = *scale;
  const int VV = V / 8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < rows * (long)VV;
       idx += gridDim.x * (long)blockDim.x) {
    const long row = idx / VV;
    const int col = (int)(idx % VV) * 8;
    const long t = targets[row];
    const float l = lse[row];
    const T* lr = logits + row * (long)V + col;
    T* dr = dlogits + row * (long)V + col;
    if (t == ignore_index) {
#pragma unroll
      for (int j = 0; j < 8; ++j) from_f32(&dr[j], 0.f);
      continue;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = __expf(to_f32(lr[j]) - l);
      if (col + j == (int)t) p -= 1.f;
      from_f32(&dr[j], p * sc);
    }
  }
}

// ---- vocab-parallel CE (TP lm-head shards; parallel/tp.py) ----
// Pass 1: per-row local max + the target logit if it lives in this shard.
template <int BLOCK>
__global__ void ce_vp_stats_kernel(const __hip_bfloat16* __restrict__ logits,
                                   const long* __restrict__ targets,
                                   float* __restrict__ m_out, float* __restrict__ tgt_out,
                                   long rows, int V, long v0, long ignore_index) {
  __shared__ float scratch[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const __hip_bfloat16* lr = logits + row * (long)V;
    float m = -INFINITY;
    const int VV = V / 8;
    const uint4* lv = reinterpret_cast<const uint4*>(lr);
    for (int i = threadIdx.x; i < VV; i += BLOCK) {
      U4 u; u.u = lv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) m = fmaxf(m, bf16_bits_to_f32(u.s[j]));
    }
    for (int i = VV * 8 + threadIdx.x; i < V; i += BLOCK) m = fmaxf(m, to_f32(lr[i]));
    m = block_reduce_max<BLOCK>(m, scratch);
    if (threadIdx.x == 0) {
      m_out[row] = m;
      const long t = targets[row];
      const long tl = t - v0;
      tgt_out[row] = (t != ignore_index && tl >= 0 && tl < V) ? to_f32(lr[tl]) : 0.f;
    }
  }
}

// Pass 2 (after the group max all-reduce): sum exp(l - m_group).
template <int BLOCK>
__global__ void ce_vp_sumexp_kernel(const __hip_bfloat16* __restrict__ logits,
                                    const float* __restrict__ m_group,
                                    float* __restrict__ se_out, long rows, int V) {
  __shared__ float scratch[BLOCK / WAVE];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const __hip_bfloat16* lr = logits + row * (long)V;
    const float m = m_group[row];
    float s = 0.f;
    const int VV = V / 8;
    const uint4* lv = reinterpret_cast<const uint4*>(lr);
    for (int i = threadIdx.x; i < VV; i += BLOCK) {
      U4 u; u.u = lv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) s += __expf(bf16_bits_to_f32(u.s[j]) - m);
    }
    for (int i = VV * 8 + threadIdx.x; i < V; i += BLOCK) s += __expf(to_f32(lr[i]) - m);
    s = block_reduce_sum<BLOCK>(s, scratch);
    if (threadIdx.x == 0) se_out[row] = s;
  }
}

// Backward: dlogits = (exp(l - lse_group) - onehot(t - v0)) * (*scale), rows
// with ignored targets get 0 (scale = upstream_grad / ntok, device scalar).
__global__ void ce_vp_bwd_kernel(const __hip_bfloat16* __restrict__ logits,
                                 const long* __restrict__ targets,
                                 const float* __restrict__ lse,
                                 const float* __restrict__ scale,
                                 __hip_bfloat16* __restrict__ dlogits,
                                 long rows, int V, long v0, long ignore_index) {
  const float sc = *scale;
  const int VV = V / 8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < rows * (long)VV;
       idx += gridDim.x * (long)blockDim.x) {
    const long row = idx / VV;
    const int col = (int)(idx % VV) * 8;
    const long t = targets[row];
    const float l = lse[row];
    const __hip_bfloat16* lr = logits + row * (long)V + col;
    __hip_bfloat16* dr = dlogits + row * (long)V + col;
    if (t == ignore_index) {
#pragma unroll
      for (int j = 0; j < 8; ++j) from_f32(&dr[j], 0.f);
      continue;
    }
    const long tl = t - v0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = __expf(to_f32(lr[j]) - l);
      if (col + j == (int)tl) p -= 1.f;
      from_f32(&dr[j], p * sc);
    }
  }
}

}  // namespace

std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor targets, long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 2);
  TORCH_CHECK(targets.scalar_type() == at::kLong);
  const long rows = logits.size(0);
  const int V = logits.size(1);
  auto loss_sum = at::zeros({}, logits.options().dtype(at::kFloat));
  auto ntok = at::zeros({}, logits.options().dtype(at::kLong));
  auto lse = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  const long grid = std::min<long>(rows, 4096);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, logits.scalar_type(), "ce_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    if constexpr (std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>) {
      ce_fwd_kernel<T, BLOCK><<<grid, BLOCK, 0, stream>>>(
          reinterpret_cast<const T*>(logits.data_ptr()), targets.data_ptr<long>(),
          loss_sum.data_ptr<float>(), ntok.data_ptr<long>(), lse.data_ptr<float>(),
          rows, V, ignore_index);
    } else {
      TORCH_CHECK(false, "ce: unsupported dtype");
    }
  });
  return {loss_sum, ntok, lse};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse, at::Tensor scale,
                  long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous());
  const long rows = logits.size(0);
  const int V = logits.size(1);
  TORCH_CHECK(V % 8 == 0, "ce_bwd: vocab must be a multiple of 8");
  auto dlogits = at::empty_like(logits);
  auto stream = at::cuda::getCurrentHIPStream();
  auto scale_f = scale.to(at::kFloat);
  const int block = 256;
  const long grid = std::min<long>(cdiv(rows * (V / 8), block), 4096);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, logits.scalar_type(), "ce_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    if constexpr (std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>) {
      ce_bwd_kernel<T><<<grid, block, 0, stream>>>(
          reinterpret_cast<const T*>(logits.data_ptr()), targets.data_ptr<long>(),
          lse.data_ptr<float>(), scale_f.data_ptr<float>(),
          reinterpret_cast<T*>(dlogits.data_ptr()), rows, V, ignore_index);
    } else {
      TORCH_CHECK(false, "ce: unsupported dtype");
    }
  });
  return dlogits;
}

std::vector<at::Tensor> ce_vp_stats(at::Tensor logits, at::Tensor targets, long v0,
                                    long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 2 &&
              logits.scalar_type() == at::kBFloat16);
  const long rows = logits.size(0);
  const int V = logits.size(1);
  TORCH_CHECK(V % 8 == 0, "ce_vp: vocab shard must be a multiple of 8");
  auto m = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto tgt = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  const int grid = (int)std::min<long>(rows, 2048);
  ce_vp_stats_kernel<256><<<grid, 256, 0, stream>>>(
      reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
      targets.data_ptr<long>(), m.data_ptr<float>(), tgt.data_ptr<float>(),
      rows, V, v0, ignore_index);
  return {m, tgt};
}

at::Tensor ce_vp_sumexp(at::Tensor logits, at::Tensor m_group) {
  const long rows = logits.size(0);
  const int V = logits.size(1);
  auto se = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  const int grid = (int)std::min<long>(rows, 2048);
  ce_vp_sumexp_kernel<256><<<grid, 256, 0, stream>>>(
      reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
      m_group.data_ptr<float>(), se.data_ptr<float>(), rows, V);
  return se;
}

at::Tensor ce_vp_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse,
                     at::Tensor scale, long v0, long ignore_index) {
  const long rows = logits.size(0);
  const int V = logits.size(1);
  auto dl = at::empty_like(logits);
  auto stream = at::cuda::getCurrentHIPStream();
  const long work = rows * (V / 8);
  const int grid = (int)std::min<long>((work + 255) / 256, 4096);
  ce_vp_bwd_kernel<<<grid, 256, 0, stream>>>(
      reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
      targets.data_ptr<long>(), lse.data_ptr<float>(), scale.data_ptr<float>(),
      reinterpret_cast<__hip_bfloat16*>(dl.data_ptr()), rows, V, v0, ignore_index);
  return dl;
}
