// Fused decode sampling (kernel K12): temperature + min-p
// sampling of one token from [V] f32 logits via the Gumbel-max trick —
// one kernel, no host sync, no sort. (Top-p runs on the torch path; the
// sampler wrapper routes here only when top_p == 1.)
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

__device__ __forceinline__ float xorshift_uniform(unsigned& state) {
  state ^= state << 13; state ^= state >> 17; state ^= state << 5;
  return (state >> 8) * (1.f / 16777216.f) + 1e-12f;
}

__global__ void sample_kernel(const float* __restrict__ logits, long* __restrict__ out,
                              int V, float inv_temp, float min_p, unsigned seed) {
  __shared__ float smax_lds[256 / WAVE];
  // pass 1: max logit (for min-p filtering threshold)
  float mx = -INFINITY;
  for (int i = threadIdx.x; i < V; i += blockDim.x) mx = fmaxf(mx, logits[i]);
  mx = block_reduce_max<256>(mx, smax_lds);
  // min-p: keep tokens with p >= min_p * p_max  <=>  (l - mx)*inv_temp >= log(min_p)
  const float thresh = (min_p > 0.f) ? __logf(min_p) : -INFINITY;
  // pass 2: per-thread best Gumbel-perturbed score
  unsigned rng = seed * 2654435761u + threadIdx.x * 40503u + 1u;
  xorshift_uniform(rng); xorshift_uniform(rng);
  float best = -INFINITY;
  int best_i = 0;
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    const float z = (logits[i] - mx) * inv_temp;
    if (z < thresh) continue;
    const float g = -__logf(-__logf(xorshift_uniform(rng)));
    const float score = z + g;
    if (score > best) { best = score; best_i = i; }
  }
  // block argmax
  __shared__ float bval[256 / WAVE];
  __shared__ int bidx[256 / WAVE];
  const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    const float ov = __shfl_down(best, off, WAVE);
    const int oi = __shfl_down(best_i, off, WAVE);
    if (ov > best) { best = ov; best_i = oi; }
  }
  if (lane == 0) { bval[wid] = best; bidx[wid] = best_i; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float b = bval[0]; int bi = bidx[0];
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w)
      if (bval[w] > b) { b = bval[w]; bi = bidx[w]; }
    *out = bi;
  }
}

// graph-capturable variant: bf16 [B, V] logits, device position salt (the
// captured graph replays with fresh randomness each token), one block/row.
__global__ void sample_dev_kernel(const __hip_bfloat16* __restrict__ logits,
                                  long* __restrict__ out, int V, float inv_temp,
                                  float min_p, unsigned seed,
                                  const int* __restrict__ pos) {
  __shared__ float smax_lds[256 / WAVE];
  const __hip_bfloat16* lr = logits + (long)blockIdx.x * V;
  float mx = -INFINITY;
  for (int i = threadIdx.x; i < V; i += blockDim.x) mx = fmaxf(mx, to_f32(lr[i]));
  mx = block_reduce_max<256>(mx, smax_lds);
  const float thresh = (min_p > 0.f) ? __logf(min_p) : -INFINITY;
  unsigned rng = (seed + (unsigned)*pos * 9781u + blockIdx.x * 613u) * 2654435761u
                 + threadIdx.x * 40503u + 1u;
  xorshift_uniform(rng); xorshift_uniform(rng);
  float best = -INFINITY;
  int best_i = 0;
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    const float z = (to_f32(lr[i]) - mx) * inv_temp;
    if (z < thresh) continue;
    const float g = -__logf(-__logf(xorshift_uniform(rng)));
    if (z + g > best) { best = z + g; best_i = i; }
  }
  __shared__ float bval[256 / WAVE];
  __shared__ int bidx[256 / WAVE];
  const int wid = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    const float ov = __shfl_down(best, off, WAVE);
    const int oi = __shfl_down(best_i, off, WAVE);
    if (ov > best) { best = ov; best_i = oi; }
  }
  if (lane == 0) { bval[wid] = best; bidx[wid] = best_i; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float b = bval[0]; int bi = bidx[0];
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w)
      if (bval[w] > b) { b = bval[w]; bi = bidx[w]; }
    out[blockIdx.x] = bi;
  }
}

}  // namespace

void sample_token_dev(at::Tensor logits, at::Tensor out, double temperature,
                      double min_p, long seed, at::Tensor pos) {
  // logits: [B, V] bf16 contiguous; out: [B] int64; pos: device int32 salt
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 2);
  TORCH_CHECK(logits.scalar_type() == at::kBFloat16);
  const int B = logits.size(0), V = logits.size(1);
  auto stream = at::cuda::getCurrentHIPStream();
  sample_dev_kernel<<<B, 256, 0, stream>>>(
      reinterpret_cast<const __hip_bfloat16*>(logits.data_ptr()),
      out.data_ptr<long>(), V, (float)(1.0 / temperature), (float)min_p,
      (unsigned)seed, pos.data_ptr<int>());
}

at::Tensor sample_token(at::Tensor logits, double temperature, double top_p, double min_p,
                        long seed) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() && logits.dim() == 1);
  TORCH_CHECK(logits.scalar_type() == at::kFloat);
  TORCH_CHECK(top_p >= 1.0, "sample_token kernel handles top_p == 1 only");
  auto out = at::empty({}, logits.options().dtype(at::kLong));
  auto stream = at::cuda::getCurrentHIPStream();
  sample_kernel<<<1, 256, 0, stream>>>(logits.data_ptr<float>(), out.data_ptr<long>(),
                                       (int)logits.numel(), (float)(1.0 / temperature),
                                       (float)min_p, (unsigned)seed);
  return out;
}
