// RMSNorm forward/backward (kernel K3).
// One workgroup per row, fp32 accumulation, bf16x8 vectorized loads.
// Replaces /root/reference/models/llama.py:44-56 (Python-composed RMSNorm).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// ---------------- forward ----------------
// y = x * rsqrt(mean(x^2) + eps) * w ; saves rstd per row for backward.
// HAS_RES: s = x + res is computed in pass 1, written out (it IS the next
// residual stream), and normalized — the separate residual-add kernel and
// its extra HBM round trip disappear (SURVEY.md §2.6 fusion rule).
// amax_partial (nullable): per-BLOCK max|y| written to amax_partial[blockIdx]
// (plain store — an atomic per wave to one address serializes at 32k blocks);
// fp8_quantize_pre reduces the partial array (skips its own amax pass).
template <typename T, int BLOCK, bool HAS_RES>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ res,
                                   T* __restrict__ sum_out, const T* __restrict__ w,
                                   T* __restrict__ y, float* __restrict__ rstd,
                                   int H, float eps, float* __restrict__ amax_partial) {
  __shared__ float scratch[BLOCK / WAVE];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)H;
  T* yr = y + row * (long)H;
  const T* rr = HAS_RES ? res + row * (long)H : nullptr;
  T* sr = HAS_RES ? sum_out + row * (long)H : nullptr;

  float ss = 0.f;
  if constexpr (sizeof(T) == 2) {
    const int HV = H / 8;
    const uint4* xv = reinterpret_cast<const uint4*>(xr);
    for (int i = threadIdx.x; i < HV; i += BLOCK) {
      U4 u; u.u = xv[i];
      if constexpr (HAS_RES) {
        U4 v, o; v.u = reinterpret_cast<const uint4*>(rr)[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf16_bits_to_f32(u.s[j]) + bf16_bits_to_f32(v.s[j]);
          o.s[j] = f32_to_bf16_bits(f);
          f = bf16_bits_to_f32(o.s[j]);  // accumulate what was stored
          ss += f * f;
        }
        reinterpret_cast<uint4*>(sr)[i] = o.u;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) { float f = bf16_bits_to_f32(u.s[j]); ss += f * f; }
      }
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
      float f = to_f32(xr[i]);
      if constexpr (HAS_RES) { f += to_f32(rr[i]); from_f32(&sr[i], f); f = to_f32(sr[i]); }
      ss += f * f;
    }
  } else {
    for (int i = threadIdx.x; i < H; i += BLOCK) {
      float f = to_f32(xr[i]);
      if constexpr (HAS_RES) { f += to_f32(rr[i]); from_f32(&sr[i], f); f = to_f32(sr[i]); }
      ss += f * f;
    }
  }
  ss = block_reduce_sum<BLOCK>(ss, scratch);
  const float r = rsqrtf(ss / H + eps);
  if (threadIdx.x == 0) rstd[row] = r;

  const T* src = HAS_RES ? sr : xr;
  float am = 0.f;
  if constexpr (sizeof(T) == 2) {
    const int HV = H / 8;
    const uint4* xv = reinterpret_cast<const uint4*>(src);
    const uint4* wv = reinterpret_cast<const uint4*>(w);
    uint4* yv = reinterpret_cast<uint4*>(yr);
    for (int i = threadIdx.x; i < HV; i += BLOCK) {
      U4 u, ww, o; u.u = xv[i]; ww.u = wv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float yf = bf16_bits_to_f32(u.s[j]) * r * bf16_bits_to_f32(ww.s[j]);
        o.s[j] = f32_to_bf16_bits(yf);
        am = fmaxf(am, fabsf(yf));
      }
      yv[i] = o.u;
    }
    for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
      const float yf = to_f32(src[i]) * r * to_f32(w[i]);
      from_f32(&yr[i], yf);
      am = fmaxf(am, fabsf(yf));
    }
  } else {
    for (int i = threadIdx.x; i < H; i += BLOCK) {
      const float yf = to_f32(src[i]) * r * to_f32(w[i]);
      from_f32(&yr[i], yf);
      am = fmaxf(am, fabsf(yf));
    }
  }
  if (amax_partial) {
    __syncthreads();  // scratch was read by every thread in pass 1's reduce
    am = block_reduce_max<BLOCK>(am, scratch);
    if (threadIdx.x == 0) amax_partial[row] = am;
  }
}

// ---------------- backward ----------------
// dx = r*dy*w - x * r^3/H * sum(dy*w*x) ; dw_partial[blk] += dy * x * r
// Grid-stride over rows; per-block dw accumulated in LDS then one global add.
template <typename T, int BLOCK, bool HAS_DADD>
__global__ void rmsnorm_bwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                   const float* __restrict__ rstd,
                                   const T* __restrict__ dy, const T* __restrict__ dadd,
                                   T* __restrict__ dx,
                                   float* __restrict__ dw_partial,
                                   long rows, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dw_acc = reinterpret_cast<float*>(smem);          // [H]
  float* scratch = dw_acc + H;                             // [BLOCK/WAVE]

  for (int i = threadIdx.x; i < H; i += BLOCK) dw_acc[i] = 0.f;
  __syncthreads();

  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + row * (long)H;
    const T* dyr = dy + row * (long)H;
    const T* dar = HAS_DADD ? dadd + row * (long)H : nullptr;
    T* dxr = dx + row * (long)H;
    const float r = rstd[row];

    float c = 0.f;
    if constexpr (sizeof(T) == 2) {
      const int HV = H / 8;
      const uint4* xv = reinterpret_cast<const uint4*>(xr);
      const uint4* dyv = reinterpret_cast<const uint4*>(dyr);
      const uint4* wv = reinterpret_cast<const uint4*>(w);
      for (int i = threadIdx.x; i < HV; i += BLOCK) {
        U4 xu, du, wu;
        xu.u = xv[i]; du.u = dyv[i]; wu.u = wv[i];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          c += bf16_bits_to_f32(du.s[j]) * bf16_bits_to_f32(wu.s[j]) * bf16_bits_to_f32(xu.s[j]);
      }
      for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK)
        c += to_f32(dyr[i]) * to_f32(w[i]) * to_f32(xr[i]);
      c = block_reduce_sum<BLOCK>(c, scratch);
      const float kk = r * r * r * c / H;
      uint4* dxv = reinterpret_cast<uint4*>(dxr);
      for (int i = threadIdx.x; i < HV; i += BLOCK) {
        U4 xu, du, wu, ou, au;
        xu.u = xv[i]; du.u = dyv[i]; wu.u = wv[i];
        if constexpr (HAS_DADD) au.u = reinterpret_cast<const uint4*>(dar)[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float xi = bf16_bits_to_f32(xu.s[j]);
          const float dyi = bf16_bits_to_f32(du.s[j]);
          float o = r * dyi * bf16_bits_to_f32(wu.s[j]) - xi * kk;
          if constexpr (HAS_DADD) o += bf16_bits_to_f32(au.s[j]);
          ou.s[j] = f32_to_bf16_bits(o);
          dw_acc[i * 8 + j] += dyi * xi * r;
        }
        dxv[i] = ou.u;
      }
      for (int i = HV * 8 + threadIdx.x; i < H; i += BLOCK) {
        const float xi = to_f32(xr[i]);
        const float dyi = to_f32(dyr[i]);
        float o = r * dyi * to_f32(w[i]) - xi * kk;
        if constexpr (HAS_DADD) o += to_f32(dar[i]);
        from_f32(&dxr[i], o);
        dw_acc[i] += dyi * xi * r;
      }
    } else {
      for (int i = threadIdx.x; i < H; i += BLOCK)
        c += to_f32(dyr[i]) * to_f32(w[i]) * to_f32(xr[i]);
      c = block_reduce_sum<BLOCK>(c, scratch);
      const float kk = r * r * r * c / H;
      for (int i = threadIdx.x; i < H; i += BLOCK) {
        const float xi = to_f32(xr[i]);
        const float dyi = to_f32(dyr[i]);
        from_f32(&dxr[i], r * dyi * to_f32(w[i]) - xi * kk);
        dw_acc[i] += dyi * xi * r;
      }
    }
    __syncthreads();
  }
  float* dwp = dw_partial + blockIdx.x * (long)H;
  for (int i = threadIdx.x; i < H; i += BLOCK) dwp[i] = dw_acc[i];
}

}  // namespace

// res numel()==0 -> plain norm, returns {y, rstd}.
// res present -> fused s = x + res: returns {y, rstd, s}.
// with_amax: also return an int[1] tensor of monotonic float bits holding
// max|y| (consumed by fp8_quantize_pre). Returns {y, rstd[, s][, amax]}.
std::vector<at::Tensor> rmsnorm_fwd_res(at::Tensor x, at::Tensor res, at::Tensor w, double eps,
                                        bool with_amax) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const bool has_res = res.numel() > 0;
  if (has_res) TORCH_CHECK(res.is_contiguous() && res.sizes() == x.sizes());
  const int H = x.size(-1);
  const long rows = x.numel() / H;
  auto y = at::empty_like(x);
  auto s = has_res ? at::empty_like(x) : at::empty({0}, x.options());
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  auto amax = with_amax ? at::empty({rows}, x.options().dtype(at::kFloat))
                        : at::empty({0}, x.options().dtype(at::kFloat));
  float* ab = with_amax ? amax.data_ptr<float>() : nullptr;
  auto stream = at::cuda::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  dim3 grid(rows);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "rmsnorm_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    if constexpr (std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>) {
      if (has_res)
        rmsnorm_fwd_kernel<T, BLOCK, true><<<grid, BLOCK, 0, stream>>>(
            reinterpret_cast<const T*>(x.data_ptr()), reinterpret_cast<const T*>(res.data_ptr()),
            reinterpret_cast<T*>(s.data_ptr()), reinterpret_cast<const T*>(w.data_ptr()),
            reinterpret_cast<T*>(y.data_ptr()), rstd.data_ptr<float>(), H, (float)eps, ab);
      else
        rmsnorm_fwd_kernel<T, BLOCK, false><<<grid, BLOCK, 0, stream>>>(
            reinterpret_cast<const T*>(x.data_ptr()), nullptr, nullptr,
            reinterpret_cast<const T*>(w.data_ptr()),
            reinterpret_cast<T*>(y.data_ptr()), rstd.data_ptr<float>(), H, (float)eps, ab);
    } else {
      TORCH_CHECK(false, "rmsnorm: unsupported dtype");
    }
  });
  std::vector<at::Tensor> out = {y, rstd};
  if (has_res) out.push_back(s);
  if (with_amax) out.push_back(amax);
  return out;
}

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  return rmsnorm_fwd_res(x, at::empty({0}, x.options()), w, eps, false);
}

// dadd numel()==0 -> plain; otherwise dx += dadd (fused residual grad add).
std::vector<at::Tensor> rmsnorm_bwd_add(at::Tensor x, at::Tensor w, at::Tensor rstd,
                                        at::Tensor dy, at::Tensor dadd) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  const bool has_dadd = dadd.numel() > 0;
  if (has_dadd) TORCH_CHECK(dadd.is_contiguous() && dadd.sizes() == x.sizes());
  const int H = x.size(-1);
  const long rows = x.numel() / H;
  auto dx = at::empty_like(x);
  constexpr int BLOCK = 256;
  const int nblocks = (int)std::min<long>(rows, 1024);
  auto dw_partial = at::zeros({nblocks, H}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  const size_t lds = (H + BLOCK / WAVE) * sizeof(float);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, x.scalar_type(), "rmsnorm_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16, float>;
    if constexpr (std::is_same_v<scalar_t, at::BFloat16> || std::is_same_v<scalar_t, float>) {
      if (has_dadd)
        rmsnorm_bwd_kernel<T, BLOCK, true><<<nblocks, BLOCK, lds, stream>>>(
            reinterpret_cast<const T*>(x.data_ptr()), reinterpret_cast<const T*>(w.data_ptr()),
            rstd.data_ptr<float>(), reinterpret_cast<const T*>(dy.data_ptr()),
            reinterpret_cast<const T*>(dadd.data_ptr()),
            reinterpret_cast<T*>(dx.data_ptr()), dw_partial.data_ptr<float>(), rows, H);
      else
        rmsnorm_bwd_kernel<T, BLOCK, false><<<nblocks, BLOCK, lds, stream>>>(
            reinterpret_cast<const T*>(x.data_ptr()), reinterpret_cast<const T*>(w.data_ptr()),
            rstd.data_ptr<float>(), reinterpret_cast<const T*>(dy.data_ptr()), nullptr,
            reinterpret_cast<T*>(dx.data_ptr()), dw_partial.data_ptr<float>(), rows, H);
    } else {
      TORCH_CHECK(false, "rmsnorm: unsupported dtype");
    }
  });
  auto dw = dw_partial.sum(0);
  return {dx, dw};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor x, at::Tensor w, at::Tensor rstd, at::Tensor dy) {
  return rmsnorm_bwd_add(x, w, rstd, dy, at::empty({0}, x.options()));
}
