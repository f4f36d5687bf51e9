// Fused flat-buffer optimizer updates (kernel K7/K10).
// Replaces per-tensor Python tree walks
// (/root/reference/optimizers/enhanced_optimizers.py:121-193 etc.): one
// elementwise pass over the whole model. Global-norm clip reads the
// device-resident sumsq — no host sync.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// ---------------- sum of squares ----------------
template <typename T>
__global__ void sumsq_kernel(const T* __restrict__ g, float* __restrict__ out, long n) {
  float acc = 0.f;
  if constexpr (sizeof(T) == 2) {
    const long nv = n / 8;
    const uint4* gv = reinterpret_cast<const uint4*>(g);
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nv;
         i += gridDim.x * (long)blockDim.x) {
      U4 u; u.u = gv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) { float f = bf16_bits_to_f32(u.s[j]); acc += f * f; }
    }
    for (long i = nv * 8 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (long)blockDim.x) { float f = to_f32(g[i]); acc += f * f; }
  } else {
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * (long)blockDim.x) { float f = to_f32(g[i]); acc += f * f; }
  }
  __shared__ float scratch[256 / WAVE];
  acc = block_reduce_sum<256>(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

__device__ __forceinline__ float clip_coef(const float* sumsq, float max_norm) {
  if (max_norm <= 0.f) return 1.f;
  const float norm = sqrtf(*sumsq);
  return fminf(max_norm / (norm + 1e-6f), 1.f);
}

// ---------------- AdamW ----------------
// param bf16/f32, master f32, grad bf16/f32, moments f32.
template <typename P, typename G>
__global__ void adamw_kernel(P* __restrict__ param, float* __restrict__ master,
                             const G* __restrict__ grad, float* __restrict__ m,
                             float* __restrict__ v, const float* __restrict__ sumsq,
                             long n, long decay_boundary, float lr, float b1, float b2,
                             float eps, float wd, float bc1, float bc2, float max_norm) {
  const float cc = clip_coef(sumsq, max_norm);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (long)blockDim.x) {
    const float g = to_f32(grad[i]) * cc;
    float mi = m[i] = b1 * m[i] + (1.f - b1) * g;
    float vi = v[i] = b2 * v[i] + (1.f - b2) * g * g;
    const float denom = sqrtf(vi / bc2) + eps;
    float p = master[i];
    if (wd > 0.f && i < decay_boundary) p *= (1.f - lr * wd);
    p -= lr * (mi / bc1) / denom;
    master[i] = p;
    from_f32(&param[i], p);
  }
}

// ---------------- Lion ----------------
template <typename P, typename G>
__global__ void lion_kernel(P* __restrict__ param, float* __restrict__ master,
                            const G* __restrict__ grad, float* __restrict__ m,
                            const float* __restrict__ sumsq, long n, long decay_boundary,
                            float lr, float b1, float b2, float wd, float max_norm) {
  const float cc = clip_coef(sumsq, max_norm);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (long)blockDim.x) {
    const float g = to_f32(grad[i]) * cc;
    const float upd = copysignf(1.f, b1 * m[i] + (1.f - b1) * g);
    m[i] = b2 * m[i] + (1.f - b2) * g;
    float p = master[i];
    if (wd > 0.f && i < decay_boundary) p *= (1.f - lr * wd);
    p -= lr * upd;
    master[i] = p;
    from_f32(&param[i], p);
  }
}

// ---------------- SGD (momentum/nesterov) ----------------
template <typename P, typename G>
__global__ void sgd_kernel(P* __restrict__ param, float* __restrict__ master,
                           const G* __restrict__ grad, float* __restrict__ buf,
                           const float* __restrict__ sumsq, long n, long decay_boundary,
                           float lr, float mom, float wd, bool nesterov, float max_norm) {
  const float cc = clip_coef(sumsq, max_norm);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (long)blockDim.x) {
    float g = to_f32(grad[i]) * cc;
    if (wd > 0.f && i < decay_boundary) g += wd * master[i];
    const float b = buf[i] = mom * buf[i] + g;
    const float upd = nesterov ? g + mom * b : b;
    const float p = master[i] - lr * upd;
    master[i] = p;
    from_f32(&param[i], p);
  }
}

template <typename F>
void dispatch_pg(at::Tensor& param, at::Tensor& grad, F&& f) {
  const bool pb = param.scalar_type() == at::kBFloat16;
  const bool gb = grad.scalar_type() == at::kBFloat16;
  if (pb && gb) f(__hip_bfloat16{}, __hip_bfloat16{});
  else if (pb && !gb) f(__hip_bfloat16{}, float{});
  else if (!pb && gb) f(float{}, __hip_bfloat16{});
  else f(float{}, float{});
}

}  // namespace

at::Tensor sumsq(at::Tensor g) {
  TORCH_CHECK(g.is_cuda() && g.is_contiguous());
  auto out = at::zeros({}, g.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  const long n = g.numel();
  const int block = 256;
  const long grid = std::min<long>(cdiv(n / 8 + 1, block), 2048);
  if (g.scalar_type() == at::kBFloat16)
    sumsq_kernel<__hip_bfloat16><<<grid, block, 0, stream>>>(
        reinterpret_cast<const __hip_bfloat16*>(g.data_ptr()), out.data_ptr<float>(), n);
  else
    sumsq_kernel<float><<<grid, block, 0, stream>>>(
        g.data_ptr<float>(), out.data_ptr<float>(), n);
  return out;
}

void adamw_step(at::Tensor param, at::Tensor master, at::Tensor grad, at::Tensor m,
                at::Tensor v, at::Tensor sumsq_t, long step, double lr, double b1,
                double b2, double eps, double wd, long decay_boundary, double max_norm) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous() && master.is_contiguous());
  const long n = param.numel();
  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const long grid = std::min<long>(cdiv(n, block), 4096);
  const float bc1 = 1.f - powf((float)b1, (float)step);
  const float bc2 = 1.f - powf((float)b2, (float)step);
  dispatch_pg(param, grad, [&](auto pt, auto gt) {
    using P = decltype(pt); using G = decltype(gt);
    adamw_kernel<P, G><<<grid, block, 0, stream>>>(
        reinterpret_cast<P*>(param.data_ptr()), master.data_ptr<float>(),
        reinterpret_cast<const G*>(grad.data_ptr()), m.data_ptr<float>(), v.data_ptr<float>(),
        sumsq_t.data_ptr<float>(), n, decay_boundary, (float)lr, (float)b1, (float)b2,
        (float)eps, (float)wd, bc1, bc2, (float)max_norm);
  });
}

void lion_step(at::Tensor param, at::Tensor master, at::Tensor grad, at::Tensor m,
               at::Tensor sumsq_t, double lr, double b1, double b2, double wd,
               long decay_boundary, double max_norm) {
  const long n = param.numel();
  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const long grid = std::min<long>(cdiv(n, block), 4096);
  dispatch_pg(param, grad, [&](auto pt, auto gt) {
    using P = decltype(pt); using G = decltype(gt);
    lion_kernel<P, G><<<grid, block, 0, stream>>>(
        reinterpret_cast<P*>(param.data_ptr()), master.data_ptr<float>(),
        reinterpret_cast<const G*>(grad.data_ptr()), m.data_ptr<float>(),
        sumsq_t.data_ptr<float>(), n, decay_boundary, (float)lr, (float)b1, (float)b2,
        (float)wd, (float)max_norm);
  });
}

void sgd_step(at::Tensor param, at::Tensor master, at::Tensor grad, at::Tensor buf,
              at::Tensor sumsq_t, double lr, double mom, double wd, long decay_boundary,
              bool nesterov, double max_norm) {
  const long n = param.numel();
  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const long grid = std::min<long>(cdiv(n, block), 4096);
  dispatch_pg(param, grad, [&](auto pt, auto gt) {
    using P = decltype(pt); using G = decltype(gt);
    sgd_kernel<P, G><<<grid, block, 0, stream>>>(
        reinterpret_cast<P*>(param.data_ptr()), master.data_ptr<float>(),
        reinterpret_cast<const G*>(grad.data_ptr()), buf.data_ptr<float>(),
        sumsq_t.data_ptr<float>(), n, decay_boundary, (float)lr, (float)mom, (float)wd,
        nesterov, (float)max_norm);
  });
}
