// RoPE apply (kernel K4). BSHD layout [B,S,H,D].
// Host-precomputed cos/sin tables [Smax, D/2] (on-device trig would turn this
// memory-bound op VALU-bound — guide Appendix B). conj=true applies the
// inverse rotation (backward).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// neox style: pair (d, d+D/2). Each thread handles 8 consecutive d within one
// (b,s,h): 16-B (bf16x8) loads/stores on each half — the 4-pair/8-B version
// ran the op 2x off the HBM limit.
template <typename T, bool TRAD>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ y,
                            const float* __restrict__ cost, const float* __restrict__ sint,
                            long total_groups, int S, int H, int D, int offset, float sgn,
                            long x_row_stride, long y_row_stride) {
  // x and y may be strided views (e.g. the q slice of the fused QKV output /
  // of the fused dQKV grad buffer): element (b,s,h,d) sits at
  // (b*S+s)*row_stride + h*D + d.
  const int half = D / 2;
  const int groups_per_row = half / 8;  // 8 pairs per thread
  for (long g = blockIdx.x * (long)blockDim.x + threadIdx.x; g < total_groups;
       g += gridDim.x * (long)blockDim.x) {
    const int gi = (int)(g % groups_per_row);
    const long row = g / groups_per_row;        // (b*S + s)*H + h
    const int s = (int)((row / H) % S);
    const int h = (int)(row % H);
    const long xrow_off = (row / H) * x_row_stride + (long)h * D;
    const long yrow_off = (row / H) * y_row_stride + (long)h * D;
    const int d0 = gi * 8;
    const float* crow = cost + (long)(s + offset) * half + d0;
    const float* srow = sint + (long)(s + offset) * half + d0;
    float c[8], s_[8];
    *reinterpret_cast<float4*>(c) = *reinterpret_cast<const float4*>(crow);
    *reinterpret_cast<float4*>(c + 4) = *reinterpret_cast<const float4*>(crow + 4);
    *reinterpret_cast<float4*>(s_) = *reinterpret_cast<const float4*>(srow);
    *reinterpret_cast<float4*>(s_ + 4) = *reinterpret_cast<const float4*>(srow + 4);
#pragma unroll
    for (int j = 0; j < 8; ++j) s_[j] *= sgn;

    if constexpr (TRAD) {
      // interleaved pairs: (2d, 2d+1); 8 pairs = 16 contiguous elements
      const T* xr = x + xrow_off + 2 * d0;
      T* yr = y + yrow_off + 2 * d0;
      if constexpr (sizeof(T) == 2) {
        U4 lo, hi, olo, ohi;
        lo.u = *reinterpret_cast<const uint4*>(xr);
        hi.u = *reinterpret_cast<const uint4*>(xr + 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          U4& in = (j < 4) ? lo : hi;
          U4& out = (j < 4) ? olo : ohi;
          const int e = 2 * (j & 3);
          const float a = bf16_bits_to_f32(in.s[e]), b = bf16_bits_to_f32(in.s[e + 1]);
          out.s[e] = f32_to_bf16_bits(a * c[j] - b * s_[j]);
          out.s[e + 1] = f32_to_bf16_bits(a * s_[j] + b * c[j]);
        }
        *reinterpret_cast<uint4*>(yr) = olo.u;
        *reinterpret_cast<uint4*>(yr + 8) = ohi.u;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float a = to_f32(xr[2 * j]), b = to_f32(xr[2 * j + 1]);
          from_f32(&yr[2 * j], a * c[j] - b * s_[j]);
          from_f32(&yr[2 * j + 1], a * s_[j] + b * c[j]);
        }
      }
    } else {
      const T* xa = x + xrow_off + d0;
      const T* xb = xa + half;
      T* ya = y + yrow_off + d0;
      T* yb = ya + half;
      if constexpr (sizeof(T) == 2) {
        U4 ua, ub, oa, ob;
        ua.u = *reinterpret_cast<const uint4*>(xa);
        ub.u = *reinterpret_cast<const uint4*>(xb);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float a = bf16_bits_to_f32(ua.s[j]), b = bf16_bits_to_f32(ub.s[j]);
          oa.s[j] = f32_to_bf16_bits(a * c[j] - b * s_[j]);
          ob.s[j] = f32_to_bf16_bits(a * s_[j] + b * c[j]);
        }
        *reinterpret_cast<uint4*>(ya) = oa.u;
        *reinterpret_cast<uint4*>(yb) = ob.u;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float a = to_f32(xa[j]), b = to_f32(xb[j]);
          from_f32(&ya[j], a * c[j] - b * s_[j]);
          from_f32(&yb[j], a * s_[j] + b * c[j]);
        }
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
  TORCH_CHECK(D % 16 == 0, "rope: head_dim must be a multiple of 16");
  TORCH_CHECK(cost.size(0) >= S + offset, "rope table too small");
  if (y.numel() == 0) {
    y = at::empty({B, S, H, D}, x.options());
  } else {
    TORCH_CHECK(y.sizes() == x.sizes() && y.stride(3) == 1 && y.stride(2) == D &&
                    y.stride(0) == S * y.stride(1),
                "rope: out must be [B,S,H,D] with contiguous (h,d) inner block");
  }
  const long y_row_stride = y.stride(1);
  const long total_groups = (long)B * S * H * (D / 16);
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
