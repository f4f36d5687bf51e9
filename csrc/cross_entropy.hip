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
