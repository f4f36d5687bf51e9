// Common helpers for the MI355X (gfx950, CDNA4) kernel layer.
// Wave size is 64 everywhere; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

#define HIP_CHECK(cmd)                                                        \
  do {                                                                        \
    hipError_t e_ = (cmd);                                                    \
    if (e_ != hipSuccess) {                                                   \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(e_), " at ",        \
                  __FILE__, ":", __LINE__);                                   \
    }                                                                         \
  } while (0)

// ---------- dtype conversion -------------------------------------------------
__device__ __forceinline__ float to_f32(__hip_bfloat16 x) { return __bfloat162float(x); }
__device__ __forceinline__ float to_f32(float x) { return x; }
__device__ __forceinline__ void from_f32(__hip_bfloat16* out, float v) { *out = __float2bfloat16(v); }
__device__ __forceinline__ void from_f32(float* out, float v) { *out = v; }

// bf16 bit helpers (short-packed vector loads; hipcc does not auto-vectorize
// scalar bf16 loads — guide Guideline 13)
union U4 {
  uint4 u;
  ushort s[8];
  float f[4];
};

__device__ __forceinline__ float bf16_bits_to_f32(ushort b) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)b) << 16;
  return c.f;
}
__device__ __forceinline__ ushort f32_to_bf16_bits(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<ushort*>(&h);
}

// ---------- wave reductions --------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, WAVE));
  return v;
}

// Block reduction: block size up to 1024 (16 waves). Returns result in every
// thread (broadcast via LDS slot 0).
template <int BLOCK>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
  constexpr int NW = BLOCK / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float r = 0.f;
  if (wid == 0) {
    r = (lane < NW) ? lds_scratch[lane] : 0.f;
    r = wave_reduce_sum(r);
    if (lane == 0) lds_scratch[0] = r;
  }
  __syncthreads();
  r = lds_scratch[0];
  __syncthreads();
  return r;
}

template <int BLOCK>
__device__ __forceinline__ float block_reduce_max(float v, float* lds_scratch) {
  constexpr int NW = BLOCK / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float r = -INFINITY;
  if (wid == 0) {
    r = (lane < NW) ? lds_scratch[lane] : -INFINITY;
    r = wave_reduce_max(r);
    if (lane == 0) lds_scratch[0] = r;
  }
  __syncthreads();
  r = lds_scratch[0];
  __syncthreads();
  return r;
}

constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }
