// RoPE apply (kernel K4). BSHD layout [B,S,H,D].
// Host-precomputed cos/sin tables [Smax, D/2] (on-device trig would turn this
// memory-bound op VALU-bound — guide Appendix B). conj=true applies the
// inverse rotation (backward).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// neox style: pair (d, d+D/2). Each thread handles 4 consecutive d within one
// (b,s,h): loads 8B from each half, fully coalesced.
template <typename T, bool TRAD>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ y,
                            const float* __restrict__ cost, const float* __restrict__ sint,
                            long total_groups, int S, int H, int D, int offset, float sgn,
                            long x_row_stride, long y_row_stride) {
  // x and y may be strided views (e.g. the q slice of the fused QKV output /
  // of the fused dQKV grad buffer): element (b,s,h,d) sits at
  // (b*S+s)*row_stride + h*D + d.
  const int half = D / 2;
  const int groups_per_row = half / 4;  // 4 pairs per thread
  for (long g = blockIdx.x * (long)blockDim.x + threadIdx.x; g < total_groups;
       g += gridDim.x * (long)blockDim.x) {
    const int gi = (int)(g % groups_per_row);
    const long row = g / groups_per_row;        // (b*S + s)*H + h
    const int s = (int)((row / H) % S);
    const int h = (int)(row % H);
    const long xrow_off = (row / H) * x_row_stride + (long)h * D;
    const long yrow_off = (row / H) * y_row_stride + (long)h * D;
    const int d0 = gi * 4;
    const float* crow = cost + (long)(s + offset) * half + d0;
    const float* srow = sint + (long)(s + offset) * half + d0;

    if constexpr (TRAD) {
      // interleaved pairs: (2d, 2d+1); 4 pairs = 8 contiguous elements
      const T* xr = x + xrow_off + 2 * d0;
      T* yr = y + yrow_off + 2 * d0;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float c = crow[j], s_ = sgn * srow[j];
        const float a = to_f32(xr[2 * j]), b = to_f32(xr[2 * j + 1]);
        from_f32(&yr[2 * j], a * c - b * s_);
        from_f32(&yr[2 * j + 1], a * s_ + b * c);
      }
    } else {
      const T* xa = x + xrow_off + d0;
      const T* xb = xa + half;
      T* ya = y + yrow_off + d0;
      T* yb = ya + half;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float c = crow[j], s_ = sgn * srow[j];
        const float a = to_f32(xa[j]), b = to_f32(xb[j]);
        from_f32(&ya[j], a * c - b * s_);
        from_f32(&yb[j], a * s_ + b * c);
      }
    }
  }
}

}  // namespace

at::Tensor rope_fwd_out(at::Tensor x, at::Tensor cost, at::Tensor sint, bool traditional,
                        long offset, bool conj, at::Tensor y) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "rope: x must be [B,S,H,D]");
  const int B = x.size(0), S = x.size(1), H = x.size(2), D = x.size(3);
  // allow row-strided views (q/k slices of the fused QKV projection / grad)
  TORCH_CHECK(x.stride(3) == 1 && x.stride(2) == D && x.stride(0) == S * x.stride(1),
              "rope: x must be [B,S,H,D] with contiguous (h,d) inner block");
  const long x_row_stride = x.stride(1);
  TORCH_CHECK(D % 8 == 0, "rope: head_dim must be a multiple of 8");
  TORCH_CHECK(cost.size(0) >= S + offset, "rope table too small");
  if (y.numel() == 0) {
    y = at::empty({B, S, H, D}, x.options());
  } else {
    TORCH_CHECK(y.sizes() == x.sizes() && y.stride(3) == 1 && y.stride(2) == D &&
                    y.stride(0) == S * y.stride(1),
                "rope: out must be [B,S,H,D] with contiguous (h,d) inner block");
  }
  const long y_row_stride = y.stride(1);
  const long total_groups = (long)B * S * H * (D / 8);
  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const long grid = std::min<long>(cdiv(total_groups, block), 2048);
  const float sgn = conj ? -1.f : 1.f;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "rope_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    if constexpr (std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>) {
      if (traditional)
        rope_kernel<T, true><<<grid, block, 0, stream>>>(
            reinterpret_cast<const T*>(x.data_ptr()), reinterpret_cast<T*>(y.data_ptr()),
            cost.data_ptr<float>(), sint.data_ptr<float>(), total_groups, S, H, D, (int)offset, sgn, x_row_stride, y_row_stride);
      else
        rope_kernel<T, false><<<grid, block, 0, stream>>>(
            reinterpret_cast<const T*>(x.data_ptr()), reinterpret_cast<T*>(y.data_ptr()),
            cost.data_ptr<float>(), sint.data_ptr<float>(), total_groups, S, H, D, (int)offset, sgn, x_row_stride, y_row_stride);
    } else {
      TORCH_CHECK(false, "rope: unsupported dtype");
    }
  });
  return y;
}

at::Tensor rope_fwd(at::Tensor x, at::Tensor cost, at::Tensor sint, bool traditional,
                    long offset, bool conj) {
  return rope_fwd_out(x, cost, sint, traditional, offset, conj,
                      at::empty({0}, x.options()));
}
