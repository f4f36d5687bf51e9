// Static-shape decode kernels for hipGraph-captured generation (K13 serving
// path). Everything shape-static: the CURRENT sequence
// length lives in a device int32 (`pos`), so one captured graph replays for
// every decode step — no per-token launch overhead (~400 launches/token on
// the eager path made 1B decode launch-bound at 5.3 ms/token).
//
// Kernels:
//   rope_decode  : RoPE at position read from pos (q and k of the new token)
//   kv_append    : write the new token's k/v into the static cache at pos
//   attn_decode  : split-KV flash-decode over the cache (phase A partials
//                  per kv-chunk with own softmax stats; phase B combine)
//   pos_incr     : pos += 1 (end of the captured step)
//   write_token  : out_ring[step_idx] = token (device-side record)
//
// Layout: cache [B, Lmax, Hkv, D] bf16 (same BSHD row layout as training);
// q/k/v of the new token [B, 1, H, D].
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr float LOG2E = 1.4426950408889634f;

// ---- rope at device position (B*H*D/8 threads; 4 pairs per thread) ----
template <bool TRAD>
__global__ void rope_decode_kernel(__hip_bfloat16* __restrict__ x,
                                   const float* __restrict__ cost,
                                   const float* __restrict__ sint,
                                   const int* __restrict__ pos,
                                   int B, int H, int D) {
  const int half = D / 2;
  const int gpr = half / 4;  // thread groups per (b,h)
  const long g = blockIdx.x * (long)blockDim.x + threadIdx.x;
  if (g >= (long)B * H * gpr) return;
  const int gi = (int)(g % gpr);
  const long row = g / gpr;  // b*H + h
  const int p = *pos;
  const int d0 = gi * 4;
  const float* crow = cost + (long)p * half + d0;
  const float* srow = sint + (long)p * half + d0;
  __hip_bfloat16* xr = x + row * (long)D;
  if constexpr (TRAD) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float c = crow[j], s = srow[j];
      const float a = to_f32(xr[2 * (d0 + j)]), b = to_f32(xr[2 * (d0 + j) + 1]);
      from_f32(&xr[2 * (d0 + j)], a * c - b * s);
      from_f32(&xr[2 * (d0 + j) + 1], a * s + b * c);
    }
  } else {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float c = crow[j], s = srow[j];
      const float a = to_f32(xr[d0 + j]), b = to_f32(xr[half + d0 + j]);
      from_f32(&xr[d0 + j], a * c - b * s);
      from_f32(&xr[half + d0 + j], a * s + b * c);
    }
  }
}

// ---- append k/v at pos ----
__global__ void kv_append_kernel(const __hip_bfloat16* __restrict__ k,
                                 const __hip_bfloat16* __restrict__ v,
                                 __hip_bfloat16* __restrict__ kc,
                                 __hip_bfloat16* __restrict__ vc,
                                 const int* __restrict__ pos,
                                 int B, int Lmax, int Hkv, int D) {
  const int p = *pos;
  const long n = (long)B * Hkv * D;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += gridDim.x * (long)blockDim.x) {
    const int b = (int)(i / (Hkv * (long)D));
    const long hd = i % (Hkv * (long)D);
    const long dst = ((long)b * Lmax + p) * Hkv * D + hd;
    kc[dst] = k[i];
    vc[dst] = v[i];
  }
}

// ---- split-KV decode attention, phase A ----
// grid (nchunks, Hq, B), block 256 (4 waves). Each block computes the
// partial o/m/l of its CHUNK of kv rows for one (b, hq). Chunks past the
// current length exit early (l=0). D <= 128. Memory-level parallelism is
// the whole game here (PMC: WAIT_ANY 84%, VALU 0.3% on the naive version):
// both passes keep UNR independent loads in flight per lane.
template <int D, int CHUNK>
__global__ __launch_bounds__(256) void attn_decode_partial_kernel(
    const __hip_bfloat16* __restrict__ q,   // [B, Hq, D] (new token, roped)
    const __hip_bfloat16* __restrict__ kc,  // [B, Lmax, Hkv, D]
    const __hip_bfloat16* __restrict__ vc,
    const int* __restrict__ pos,            // length BEFORE append
    float* __restrict__ part,               // [B, Hq, NC, D+2]
    int B, int Lmax, int Hq, int Hkv, float scale2) {
  const int chunk = blockIdx.x, hq = blockIdx.y, b = blockIdx.z;
  const int NC = gridDim.x;
  const int hkv = hq / (Hq / Hkv);
  const int len = *pos + 1;  // including the just-appended token
  const int k0 = chunk * CHUNK;
  float* out = part + (((long)b * Hq + hq) * NC + chunk) * (D + 2);
  if (k0 >= len) {
    if (threadIdx.x == 0) { out[D] = -INFINITY; out[D + 1] = 0.f; }
    return;
  }
  const int kend = min(len, k0 + CHUNK);

  __shared__ float scratch[256 / WAVE];
  __shared__ float s_row[CHUNK];
  const int tid = threadIdx.x;
  for (int r = tid; r < CHUNK; r += 256) s_row[r] = -INFINITY;
  __syncthreads();

  const __hip_bfloat16* qp = q + ((long)b * Hq + hq) * D;
  const long rs = (long)Hkv * D;  // cache row stride (elements)
  const __hip_bfloat16* kbase = kc + ((long)b * Lmax) * rs + (long)hkv * D;
  const __hip_bfloat16* vbase = vc + ((long)b * Lmax) * rs + (long)hkv * D;

  const int lpr = D / 8;               // lanes per row (16 at D=128)
  const int rpw = WAVE / lpr;          // rows per wave pass (4)
  const int wid = tid / WAVE, lane = tid % WAVE;
  const int sub = lane / lpr;          // row slot within wave
  const int dl = (lane % lpr) * 8;
  U4 qv;
  qv.u = *reinterpret_cast<const uint4*>(qp + dl);

  // score pass: UNR row-groups in flight per wave
  constexpr int UNR = 4;
  for (int r0 = k0 + wid * rpw; r0 < kend; r0 += 4 * rpw * UNR) {
    U4 kv8[UNR];
    int krow[UNR];
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
      krow[u] = r0 + u * 4 * rpw + sub;
      kv8[u].u = (krow[u] < kend)
                     ? *reinterpret_cast<const uint4*>(kbase + (long)krow[u] * rs + dl)
                     : uint4{0, 0, 0, 0};
    }
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
      float acc = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc += bf16_bits_to_f32(qv.s[j]) * bf16_bits_to_f32(kv8[u].s[j]);
      for (int off = lpr / 2; off > 0; off >>= 1) acc += __shfl_xor(acc, off);
      if (krow[u] < kend && (lane % lpr) == 0) s_row[krow[u] - k0] = acc * scale2;
    }
  }
  __syncthreads();

  // block max + expsum (base-2 domain)
  float m = -INFINITY;
  for (int r = tid; r < CHUNK; r += 256) m = fmaxf(m, s_row[r]);
  m = block_reduce_max<256>(m, scratch);
  float l = 0.f;
  for (int r = tid; r < CHUNK; r += 256) {
    if (s_row[r] != -INFINITY) {
      const float p = __builtin_amdgcn_exp2f(s_row[r] - m);
      s_row[r] = p;
      l += p;
    } else {
      s_row[r] = 0.f;
    }
  }
  l = block_reduce_sum<256>(l, scratch);

  // weighted V: all 256 threads (two row-phases at D=128), 8 rows in flight
  const int nrows = kend - k0;
  const int d = tid % D;
  const int ph = tid / D;            // 0 or 1 at D=128
  const int nph = 256 / D;           // phases
  float acc = 0.f;
  {
    int r = ph;
    for (; r + 8 * nph <= nrows; r += 8 * nph) {
      float vals[8];
#pragma unroll
      for (int u = 0; u < 8; ++u)
        vals[u] = to_f32(vbase[(long)(k0 + r + u * nph) * rs + d]);
#pragma unroll
      for (int u = 0; u < 8; ++u) acc += s_row[r + u * nph] * vals[u];
    }
    for (; r < nrows; r += nph)
      acc += s_row[r] * to_f32(vbase[(long)(k0 + r) * rs + d]);
  }
  // combine the phases through LDS (reuse s_row as [(nph-1)*D] scratch;
  // (nph-1)*D <= 192 <= CHUNK for D in {64,128})
  __syncthreads();
  if (ph > 0) s_row[(ph - 1) * D + d] = acc;
  __syncthreads();
  if (ph == 0) {
    for (int p = 1; p < nph; ++p) acc += s_row[(p - 1) * D + d];
    out[d] = acc;
  }
  if (tid == 0) { out[D] = m; out[D + 1] = l; }
}

// ---- phase B: combine partials ----
// grid (Hq, B), block 128. o[b,hq,:] = sum_c exp2(m_c - M) * o_c / L
template <int D>
__global__ void attn_decode_combine_kernel(const float* __restrict__ part,
                                           __hip_bfloat16* __restrict__ o,
                                           const int* __restrict__ pos,
                                           int NC, int Hq, int CHUNK) {
  const int hq = blockIdx.x, b = blockIdx.y;
  const int len = *pos + 1;
  const int nc = min(NC, (len + CHUNK - 1) / CHUNK);
  const float* base = part + (((long)b * Hq + hq) * NC) * (D + 2);
  const int tid = threadIdx.x;

  __shared__ float mM;
  __shared__ float w[64];  // per-chunk weights (NC <= 64)
  if (tid == 0) {
    float M = -INFINITY;
    for (int c = 0; c < nc; ++c) M = fmaxf(M, base[c * (D + 2) + D]);
    float L = 0.f;
    for (int c = 0; c < nc; ++c) {
      const float mc = base[c * (D + 2) + D];
      const float lc = base[c * (D + 2) + D + 1];
      const float wc = (lc > 0.f) ? __builtin_amdgcn_exp2f(mc - M) : 0.f;
      w[c] = wc;
      L += wc * lc;
    }
    mM = (L > 0.f) ? 1.f / L : 0.f;
    for (int c = 0; c < nc; ++c) w[c] *= mM;
  }
  __syncthreads();
  for (int d = tid; d < D; d += blockDim.x) {
    float acc = 0.f;
    for (int c = 0; c < nc; ++c) acc += w[c] * base[c * (D + 2) + d];
    from_f32(&o[((long)b * Hq + hq) * D + d], acc);
  }
}


// ---- int8-quantized KV decode (K13: quantized cache at kernel speed) ----
// Codes: uint8 [B, Lmax, Hkv, D]; per-64-elem group (scale, zero) float2
// [B, Lmax, Hkv, D/64]. Dequant fused into the attention dot loops — the
// eager torch dequant path rebuilt the full bf16 cache every step.

__global__ void kv_append_q8_kernel(const __hip_bfloat16* __restrict__ k,
                                    const __hip_bfloat16* __restrict__ v,
                                    unsigned char* __restrict__ kc, float2* __restrict__ ksz,
                                    unsigned char* __restrict__ vc, float2* __restrict__ vsz,
                                    const int* __restrict__ pos,
                                    int B, int Lmax, int Hkv, int D) {
  // one wave per (b, h, tensor, group-of-64): lanes 0..63 cover the group
  const int G = 64;
  const int groups = D / G;
  const long total = (long)B * Hkv * 2 * groups;
  const long gi = (blockIdx.x * (long)blockDim.x + threadIdx.x) / WAVE;
  if (gi >= total) return;
  const int lane = threadIdx.x % WAVE;
  const int grp = (int)(gi % groups);
  const int t = (int)((gi / groups) % 2);
  const int h = (int)((gi / (2 * groups)) % Hkv);
  const int b = (int)(gi / (2 * groups * Hkv));
  const int p = *pos;
  const __hip_bfloat16* src = (t == 0 ? k : v) + ((long)b * Hkv + h) * D + grp * G;
  const float x = to_f32(src[lane]);
  float lo = x, hi = x;
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    lo = fminf(lo, __shfl_xor(lo, off));
    hi = fmaxf(hi, __shfl_xor(hi, off));
  }
  const float scale = fmaxf((hi - lo) / 255.f, 1e-8f);
  const int code = (int)rintf((x - lo) / scale);
  const long row = ((long)b * Lmax + p) * Hkv + h;
  unsigned char* cdst = (t == 0 ? kc : vc) + row * D + grp * G;
  cdst[lane] = (unsigned char)min(max(code, 0), 255);
  if (lane == 0) (t == 0 ? ksz : vsz)[row * (D / G) + grp] = float2{scale, lo};
}

template <int D, int CHUNK>
__global__ __launch_bounds__(256) void attn_decode_partial_q8_kernel(
    const __hip_bfloat16* __restrict__ q,
    const unsigned char* __restrict__ kc, const float2* __restrict__ ksz,
    const unsigned char* __restrict__ vc, const float2* __restrict__ vsz,
    const int* __restrict__ pos, float* __restrict__ part,
    int B, int Lmax, int Hq, int Hkv, float scale2) {
  // same MLP-pipelined structure as the bf16 kernel (UNR loads in flight),
  // with the int8 group dequant fused into both passes
  const int chunk = blockIdx.x, hq = blockIdx.y, b = blockIdx.z;
  const int NC = gridDim.x;
  const int hkv = hq / (Hq / Hkv);
  const int len = *pos + 1;
  const int k0 = chunk * CHUNK;
  float* out = part + (((long)b * Hq + hq) * NC + chunk) * (D + 2);
  if (k0 >= len) {
    if (threadIdx.x == 0) { out[D] = -INFINITY; out[D + 1] = 0.f; }
    return;
  }
  const int kend = min(len, k0 + CHUNK);
  __shared__ float scratch[256 / WAVE];
  __shared__ float s_row[CHUNK];
  const int tid = threadIdx.x;
  for (int r = tid; r < CHUNK; r += 256) s_row[r] = -INFINITY;
  __syncthreads();

  const __hip_bfloat16* qp = q + ((long)b * Hq + hq) * D;
  const long rs = (long)Hkv * D;
  const unsigned char* kbase = kc + ((long)b * Lmax) * rs + (long)hkv * D;
  const unsigned char* vbase = vc + ((long)b * Lmax) * rs + (long)hkv * D;
  const long szrs = (long)Hkv * (D / 64);
  const float2* kszb = ksz + ((long)b * Lmax) * szrs + (long)hkv * (D / 64);
  const float2* vszb = vsz + ((long)b * Lmax) * szrs + (long)hkv * (D / 64);

  const int lpr = D / 8;
  const int rpw = WAVE / lpr;
  const int wid = tid / WAVE, lane = tid % WAVE;
  const int sub = lane / lpr;
  const int dl = (lane % lpr) * 8;
  U4 qv;
  qv.u = *reinterpret_cast<const uint4*>(qp + dl);

  constexpr int UNR = 4;
  for (int r0 = k0 + wid * rpw; r0 < kend; r0 += 4 * rpw * UNR) {
    uint2 codes[UNR];
    float2 sz[UNR];
    int krow[UNR];
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
      krow[u] = r0 + u * 4 * rpw + sub;
      const bool ok = krow[u] < kend;
      codes[u] = ok ? *reinterpret_cast<const uint2*>(kbase + (long)krow[u] * rs + dl)
                    : uint2{0, 0};
      sz[u] = ok ? kszb[(long)krow[u] * szrs + dl / 64] : float2{0.f, 0.f};
    }
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
      const unsigned char* cb = reinterpret_cast<const unsigned char*>(&codes[u]);
      float acc = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc += bf16_bits_to_f32(qv.s[j]) * (cb[j] * sz[u].x + sz[u].y);
      for (int off = lpr / 2; off > 0; off >>= 1) acc += __shfl_xor(acc, off);
      if (krow[u] < kend && (lane % lpr) == 0) s_row[krow[u] - k0] = acc * scale2;
    }
  }
  __syncthreads();

  float m = -INFINITY;
  for (int r = tid; r < CHUNK; r += 256) m = fmaxf(m, s_row[r]);
  m = block_reduce_max<256>(m, scratch);
  float l = 0.f;
  for (int r = tid; r < CHUNK; r += 256) {
    if (s_row[r] != -INFINITY) {
      const float p = __builtin_amdgcn_exp2f(s_row[r] - m);
      s_row[r] = p;
      l += p;
    } else {
      s_row[r] = 0.f;
    }
  }
  l = block_reduce_sum<256>(l, scratch);

  const int nrows = kend - k0;
  const int d = tid % D;
  const int ph = tid / D;
  const int nph = 256 / D;
  const int grp = d / 64;
  float acc = 0.f;
  {
    int r = ph;
    for (; r + 8 * nph <= nrows; r += 8 * nph) {
      float vals[8];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const long row = (long)(k0 + r + u * nph);
        const float2 s = vszb[row * szrs + grp];
        vals[u] = vbase[row * rs + d] * s.x + s.y;
      }
#pragma unroll
      for (int u = 0; u < 8; ++u) acc += s_row[r + u * nph] * vals[u];
    }
    for (; r < nrows; r += nph) {
      const long row = (long)(k0 + r);
      const float2 s = vszb[row * szrs + grp];
      acc += s_row[r] * (vbase[row * rs + d] * s.x + s.y);
    }
  }
  __syncthreads();
  if (ph > 0) s_row[(ph - 1) * D + d] = acc;
  __syncthreads();
  if (ph == 0) {
    for (int p = 1; p < nph; ++p) acc += s_row[(p - 1) * D + d];
    out[d] = acc;
  }
  if (tid == 0) { out[D] = m; out[D + 1] = l; }
}

__global__ void pos_incr_kernel(int* pos) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *pos += 1;
}

__global__ void write_token_kernel(const long* __restrict__ tok,
                                   long* __restrict__ ring,
                                   const int* __restrict__ idx, int B, int cap) {
  const int b = threadIdx.x;
  if (b < B) ring[(long)(*idx % cap) * B + b] = tok[b];
}

}  // namespace

void rope_decode_(at::Tensor x, at::Tensor cost, at::Tensor sint, bool traditional,
                  at::Tensor pos) {
  // x: [B, 1, H, D] contiguous, modified in place
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16);
  const int B = x.size(0), H = x.size(2), D = x.size(3);
  const long total = (long)B * H * (D / 8);
  auto stream = at::cuda::getCurrentHIPStream();
  const int block = 256;
  const long grid = cdiv(total, block);
  auto* xp = reinterpret_cast<__hip_bfloat16*>(x.data_ptr());
  if (traditional)
    rope_decode_kernel<true><<<grid, block, 0, stream>>>(
        xp, cost.data_ptr<float>(), sint.data_ptr<float>(), pos.data_ptr<int>(), B, H, D);
  else
    rope_decode_kernel<false><<<grid, block, 0, stream>>>(
        xp, cost.data_ptr<float>(), sint.data_ptr<float>(), pos.data_ptr<int>(), B, H, D);
}

void kv_append_(at::Tensor k, at::Tensor v, at::Tensor kc, at::Tensor vc, at::Tensor pos) {
  TORCH_CHECK(k.is_cuda() && kc.is_contiguous() && vc.is_contiguous());
  const int B = kc.size(0), Lmax = kc.size(1), Hkv = kc.size(2), D = kc.size(3);
  auto stream = at::cuda::getCurrentHIPStream();
  const long n = (long)B * Hkv * D;
  kv_append_kernel<<<cdiv(n, 256), 256, 0, stream>>>(
      reinterpret_cast<const __hip_bfloat16*>(k.contiguous().data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(v.contiguous().data_ptr()),
      reinterpret_cast<__hip_bfloat16*>(kc.data_ptr()),
      reinterpret_cast<__hip_bfloat16*>(vc.data_ptr()),
      pos.data_ptr<int>(), B, Lmax, Hkv, D);
}

at::Tensor attn_decode(at::Tensor q, at::Tensor kc, at::Tensor vc, at::Tensor pos,
                       at::Tensor part, double scale) {
  // q: [B, 1, Hq, D]; kc/vc: [B, Lmax, Hkv, D]; part: fp32 workspace
  // [B, Hq, NC, D+2] with NC = cdiv(Lmax, CHUNK). Returns o [B, 1, Hq, D].
  constexpr int CHUNK = 256;
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  const int B = kc.size(0), Lmax = kc.size(1), Hkv = kc.size(2), D = kc.size(3);
  const int Hq = q.size(2);
  const int NC = cdiv(Lmax, CHUNK);
  TORCH_CHECK(NC <= 64, "attn_decode: Lmax too large for the combine kernel");
  TORCH_CHECK(part.numel() >= (long)B * Hq * NC * (D + 2), "attn_decode: workspace too small");
  auto o = at::empty({B, 1, Hq, D}, q.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const float scale2 = (float)scale * LOG2E;
  dim3 gA(NC, Hq, B);
  auto* qp = reinterpret_cast<const __hip_bfloat16*>(q.contiguous().data_ptr());
  auto* kp = reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr());
  auto* vp = reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr());
  auto* op = reinterpret_cast<__hip_bfloat16*>(o.data_ptr());
  if (D == 128) {
    attn_decode_partial_kernel<128, CHUNK><<<gA, 256, 0, stream>>>(
        qp, kp, vp, pos.data_ptr<int>(), part.data_ptr<float>(), B, Lmax, Hq, Hkv, scale2);
    attn_decode_combine_kernel<128><<<dim3(Hq, B), 128, 0, stream>>>(
        part.data_ptr<float>(), op, pos.data_ptr<int>(), NC, Hq, CHUNK);
  } else if (D == 64) {
    attn_decode_partial_kernel<64, CHUNK><<<gA, 256, 0, stream>>>(
        qp, kp, vp, pos.data_ptr<int>(), part.data_ptr<float>(), B, Lmax, Hq, Hkv, scale2);
    attn_decode_combine_kernel<64><<<dim3(Hq, B), 128, 0, stream>>>(
        part.data_ptr<float>(), op, pos.data_ptr<int>(), NC, Hq, CHUNK);
  } else {
    TORCH_CHECK(false, "attn_decode: head_dim must be 64 or 128");
  }
  return o;
}


void kv_append_q8_(at::Tensor k, at::Tensor v, at::Tensor kc, at::Tensor ksz,
                   at::Tensor vc, at::Tensor vsz, at::Tensor pos) {
  const int B = kc.size(0), Lmax = kc.size(1), Hkv = kc.size(2), D = kc.size(3);
  TORCH_CHECK(D % 64 == 0, "kv_append_q8: D must be a multiple of 64");
  auto stream = at::cuda::getCurrentHIPStream();
  const long waves = (long)B * Hkv * 2 * (D / 64);
  kv_append_q8_kernel<<<cdiv(waves * WAVE, 256), 256, 0, stream>>>(
      reinterpret_cast<const __hip_bfloat16*>(k.contiguous().data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(v.contiguous().data_ptr()),
      kc.data_ptr<unsigned char>(), reinterpret_cast<float2*>(ksz.data_ptr<float>()),
      vc.data_ptr<unsigned char>(), reinterpret_cast<float2*>(vsz.data_ptr<float>()),
      pos.data_ptr<int>(), B, Lmax, Hkv, D);
}

at::Tensor attn_decode_q8(at::Tensor q, at::Tensor kc, at::Tensor ksz, at::Tensor vc,
                          at::Tensor vsz, at::Tensor pos, at::Tensor part, double scale) {
  constexpr int CHUNK = 256;
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16);
  const int B = kc.size(0), Lmax = kc.size(1), Hkv = kc.size(2), D = kc.size(3);
  const int Hq = q.size(2);
  const int NC = cdiv(Lmax, CHUNK);
  TORCH_CHECK(NC <= 64, "attn_decode_q8: Lmax too large");
  auto o = at::empty({B, 1, Hq, D}, q.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const float scale2 = (float)scale * LOG2E;
  dim3 gA(NC, Hq, B);
  auto* qp = reinterpret_cast<const __hip_bfloat16*>(q.contiguous().data_ptr());
  auto* op = reinterpret_cast<__hip_bfloat16*>(o.data_ptr());
#define Q8LAUNCH(D_)                                                                     \
  do {                                                                                   \
    attn_decode_partial_q8_kernel<D_, CHUNK><<<gA, 256, 0, stream>>>(                    \
        qp, kc.data_ptr<unsigned char>(),                                                \
        reinterpret_cast<const float2*>(ksz.data_ptr<float>()),                          \
        vc.data_ptr<unsigned char>(),                                                    \
        reinterpret_cast<const float2*>(vsz.data_ptr<float>()), pos.data_ptr<int>(),     \
        part.data_ptr<float>(), B, Lmax, Hq, Hkv, scale2);                               \
    attn_decode_combine_kernel<D_><<<dim3(Hq, B), 128, 0, stream>>>(                     \
        part.data_ptr<float>(), op, pos.data_ptr<int>(), NC, Hq, CHUNK);                 \
  } while (0)
  if (D == 128) Q8LAUNCH(128);
  else if (D == 64) Q8LAUNCH(64);
  else TORCH_CHECK(false, "attn_decode_q8: head_dim must be 64 or 128");
#undef Q8LAUNCH
  return o;
}

void pos_incr_(at::Tensor pos) {
  auto stream = at::cuda::getCurrentHIPStream();
  pos_incr_kernel<<<1, 64, 0, stream>>>(pos.data_ptr<int>());
}

void write_token_(at::Tensor tok, at::Tensor ring, at::Tensor idx) {
  // tok: [B] int64; ring: [cap, B] int64; idx: device int32 (ring row)
  const int B = tok.numel();
  const int cap = ring.size(0);
  auto stream = at::cuda::getCurrentHIPStream();
  write_token_kernel<<<1, std::max(64, B), 0, stream>>>(
      tok.data_ptr<long>(), ring.data_ptr<long>(), idx.data_ptr<int>(), B, cap);
}
