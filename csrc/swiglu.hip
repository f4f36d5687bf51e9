// Fused SwiGLU epilogue (kernel K5): out = silu(gate) * up
// on the [.., 2I] output of the fused gate+up GEMM. HBM-bound: bf16x8 loads.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// amax_partial (nullable): per-block max|out| (plain store, reduced later by
// fp8_quantize_pre — no hot atomics; ops/fp8.py).
template <typename T>
__global__ void swiglu_fwd_kernel(const T* __restrict__ gu, T* __restrict__ out,
                                  long rows, int I, float* __restrict__ amax_partial) {
  __shared__ float scratch[256 / WAVE];
  float am = 0.f;
  const int IV = I / 8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < rows * (long)IV;
       idx += gridDim.x * (long)blockDim.x) {
    const long row = idx / IV;
    const int col = (int)(idx % IV) * 8;
    const T* g = gu + row * (long)(2 * I) + col;
    const T* u = g + I;
    T* o = out + row * (long)I + col;
    if constexpr (sizeof(T) == 2) {
      U4 gv, uv, ov;
      gv.u = *reinterpret_cast<const uint4*>(g);
      uv.u = *reinterpret_cast<const uint4*>(u);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float gf = bf16_bits_to_f32(gv.s[j]);
        const float uf = bf16_bits_to_f32(uv.s[j]);
        const float of = gf / (1.f + __expf(-gf)) * uf;
        ov.s[j] = f32_to_bf16_bits(of);
        am = fmaxf(am, fabsf(of));
      }
      *reinterpret_cast<uint4*>(o) = ov.u;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float gf = to_f32(g[j]);
        const float of = gf / (1.f + __expf(-gf)) * to_f32(u[j]);
        from_f32(&o[j], of);
        am = fmaxf(am, fabsf(of));
      }
    }
  }
  if (amax_partial) {
    am = block_reduce_max<256>(am, scratch);
    if (threadIdx.x == 0) amax_partial[blockIdx.x] = am;
  }
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T* __restrict__ gu, const T* __restrict__ dy,
                                  T* __restrict__ dgu, long rows, int I) {
  const int IV = I / 8;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < rows * (long)IV;
       idx += gridDim.x * (long)blockDim.x) {
    const long row = idx / IV;
    const int col = (int)(idx % IV) * 8;
    const T* g = gu + row * (long)(2 * I) + col;
    const T* u = g + I;
    const T* d = dy + row * (long)I + col;
    T* dg = dgu + row * (long)(2 * I) + col;
    T* du = dg + I;
    if constexpr (sizeof(T) == 2) {
      U4 gv, uv, dv, dgv, duv;
      gv.u = *reinterpret_cast<const uint4*>(g);
      uv.u = *reinterpret_cast<const uint4*>(u);
      dv.u = *reinterpret_cast<const uint4*>(d);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float gf = bf16_bits_to_f32(gv.s[j]);
        const float uf = bf16_bits_to_f32(uv.s[j]);
        const float df = bf16_bits_to_f32(dv.s[j]);
        const float sg = 1.f / (1.f + __expf(-gf));
        dgv.s[j] = f32_to_bf16_bits(df * uf * (sg * (1.f + gf * (1.f - sg))));
        duv.s[j] = f32_to_bf16_bits(df * gf * sg);
      }
      *reinterpret_cast<uint4*>(dg) = dgv.u;
      *reinterpret_cast<uint4*>(du) = duv.u;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float gf = to_f32(g[j]);
        const float uf = to_f32(u[j]);
        const float df = to_f32(d[j]);
        const float sg = 1.f / (1.f + __expf(-gf));
        const float silu = gf * sg;
        from_f32(&dg[j], df * uf * (sg * (1.f + gf * (1.f - sg))));
        from_f32(&du[j], df * silu);
      }
    }
  }
}

}  // namespace

std::vector<at::Tensor> swiglu_fwd_amax(at::Tensor gu, bool with_amax) {
  TORCH_CHECK(gu.is_cuda() && gu.is_contiguous());
  const int I2 = gu.size(-1);
  TORCH_CHECK(I2 % 16 == 0, "swiglu: intermediate size must be a multiple of 8");
  const int I = I2 / 2;
  const long rows = gu.numel() / I2;
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto out = at::empty(sizes, gu.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const long total = rows * (I / 8);
  const int block = 256;
  const long grid = std::min<long>(cdiv(total, block), 2048);
  auto amax = with_amax ? at::empty({grid}, gu.options().dtype(at::kFloat))
                        : at::empty({0}, gu.options().dtype(at::kFloat));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, gu.scalar_type(), "swiglu_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    if constexpr (std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>) {
      swiglu_fwd_kernel<T><<<grid, block, 0, stream>>>(
          reinterpret_cast<const T*>(gu.data_ptr()), reinterpret_cast<T*>(out.data_ptr()), rows, I,
          with_amax ? amax.data_ptr<float>() : nullptr);
    } else {
      TORCH_CHECK(false, "swiglu: unsupported dtype");
    }
  });
  return {out, amax};
}

at::Tensor swiglu_fwd(at::Tensor gu) {
  return swiglu_fwd_amax(gu, false)[0];
}

at::Tensor swiglu_bwd(at::Tensor gu, at::Tensor dy) {
  TORCH_CHECK(gu.is_cuda() && gu.is_contiguous() && dy.is_contiguous());
  const int I2 = gu.size(-1);
  const int I = I2 / 2;
  const long rows = gu.numel() / I2;
  auto dgu = at::empty_like(gu);
  auto stream = at::cuda::getCurrentHIPStream();
  const long total = rows * (I / 8);
  const int block = 256;
  const long grid = std::min<long>(cdiv(total, block), 2048);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, gu.scalar_type(), "swiglu_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    if constexpr (std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>) {
      swiglu_bwd_kernel<T><<<grid, block, 0, stream>>>(
          reinterpret_cast<const T*>(gu.data_ptr()), reinterpret_cast<const T*>(dy.data_ptr()),
          reinterpret_cast<T*>(dgu.data_ptr()), rows, I);
    } else {
      TORCH_CHECK(false, "swiglu: unsupported dtype");
    }
  });
  return dgu;
}
