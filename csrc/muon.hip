// K8: Muon Newton-Schulz orthogonalization — hand-written gfx950 MFMA GEMM
// chain (reference algorithm
// /root/reference/optimizers/muon.py:54-83).
//
// The NS-5 iteration per step is   A = X Xᵀ;  B = b·A + c·A·Aᵀ;  X = a·X + B·X
// (A symmetric, so A² = A·Aᵀ).  Two kernels cover the chain:
//   muon_gemm_nt   : C[M,N] = alpha·X[M,K]·Y[N,K]ᵀ + beta·E[M,N]   (NT form —
//                    both operands row-major with contiguous K, staged by
//                    global_load_lds with a source-side XOR swizzle; the
//                    quintic combine b·A + c·A² runs as the fused beta·E
//                    epilogue instead of two extra elementwise kernels)
//   muon_gemm_nn_ax: C[M,N] = Bm[M,K]·X[K,N] + a·X[M,N]             (NN form —
//                    X's K index is its row, so the X tile is staged as a
//                    transposed LDS image exactly like attention's V tile,
//                    and a·X is the fused epilogue)
//
// Design notes (measured rationale, guide §5):
//  - A triangle-only syrk would halve the off-diagonal FLOPs, but at the NS
//    shapes (m ≤ 2048 → T = m/128 ≤ 16 → T(T+1)/2 ≤ 136 blocks) the upper
//    triangle under-fills the 256-CU chip by ~2x; the full (ti,tj) grid does
//    2x the FLOPs at 2x the occupancy — same wall clock, no mirror pass, and
//    the operands of X·Xᵀ are the SAME tensor so L2/L3 serve the re-reads.
//  - 128x128 tile, BK=64, 4 waves (2x2), 64x64 per wave, 16 MFMAs
//    (32x32x16 bf16) per wave per K-step, double-buffered LDS, the guide's
//    minimum-2-phase schedule (stage t+1 issued before the ds_read+MFMA of
//    tile t, one vmcnt(0)+barrier per K-step).
//  - fp32 accumulation, bf16 storage; shapes padded to 128/64 multiples by
//    the caller (zero padding is exact for the whole NS chain: padded rows
//    and columns stay zero through XXᵀ, the combine and B·X).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "attn_common.h"

namespace {

constexpr int BMN = 128;  // square output tile
constexpr int BK = 64;    // K step
constexpr int TPB = 256;  // 4 waves

// byte-level XOR swizzle for the row-major [128][BK] bf16 images: 16-B chunk
// index (3 bits) XOR (row & 7) — ds_read_b128 lane groups read 16 different
// rows at one chunk column, which a linear image makes an 8..16-way bank
// conflict (guide §6 G4).
__device__ __forceinline__ int swz_chunk(int chunk, int row) { return chunk ^ (row & 7); }

// stage a [128][BK] row-major bf16 tile into a swizzled lane-linear LDS image
// via global_load_lds (16 B per lane; the swizzle is applied to the SOURCE
// address — guide §5.4 rule 21). src points at element [0][0] of the tile,
// ld = row stride in elements. lds_base is the wave-uniform image base.
__device__ __forceinline__ void glds_tile(const __hip_bfloat16* src, long ld,
                                          __hip_bfloat16* lds_base) {
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  // each wave stages 32 rows: 4 glds of 8 rows (64 lanes x 16 B = 1 KiB)
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    const int row = w * 32 + g * 8 + (lane >> 3);
    const int chunk = swz_chunk(lane & 7, row);
    const __hip_bfloat16* gp = src + (long)row * ld + chunk * 8;
    __hip_bfloat16* lp = lds_base + (w * 32 + g * 8) * BK;  // wave-uniform
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gp,
        (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
  }
}

// read one 32x16 A/B-operand fragment: rows r0+lane&31, k-chunk kc*2+hi of
// the swizzled [128][BK] image (b128, conflict <= 2-way)
__device__ __forceinline__ bf16x8 img_frag(const __hip_bfloat16* img, int r0,
                                           int kc, int lane) {
  const int row = r0 + (lane & 31);
  const int chunk = swz_chunk(kc * 2 + (lane >> 5), row);
  Bf16x8U u;
  *reinterpret_cast<uint4*>(u.s) =
      *reinterpret_cast<const uint4*>(img + row * BK + chunk * 8);
  return u.v;
}

// ---------------------------------------------------------------------------
// C[M,N] = alpha * X[M,K] @ Y[N,K]^T + beta * E[M,N]
// SPLIT: blockIdx.z = K-slice; partial tiles land as fp32 slabs in WS
// [slices, M, N] and muon_combine_kernel applies alpha/beta/E. Chosen when
// the (M/128)x(N/128) grid under-fills the 256 CUs (the NS shapes at
// m <= 1024 run 64 blocks -> 25% occupancy without it; the guide's M=256
// GEMM lesson: pick SPLITK so blocks ~ 0.5-1x the CU count).
template <bool HAS_E, typename TO = __hip_bfloat16, bool PIPE3 = false, bool SPLIT = false>
__global__ __launch_bounds__(TPB) void muon_gemm_nt_kernel(
    const __hip_bfloat16* __restrict__ X, const __hip_bfloat16* __restrict__ Y,
    const TO* __restrict__ E, TO* __restrict__ C,
    int M, int N, int K, float alpha, float beta,
    float* __restrict__ WS = nullptr, int ksteps_per_slice = 0) {
  constexpr int TILE = BMN * BK;          // elements per image
  constexpr int SLOTS = PIPE3 ? 3 : 2;
  __shared__ __hip_bfloat16 smem[SLOTS * 2 * TILE];

  const int tm = blockIdx.x, tn = blockIdx.y;
  const int m0 = tm * BMN, n0 = tn * BMN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;
  const int lq = lane & 31, hi = lane >> 5;

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x16{};

  int k0step = 0, ksteps = K / BK;
  if constexpr (SPLIT) {
    k0step = blockIdx.z * ksteps_per_slice;
    ksteps = min(ksteps, k0step + ksteps_per_slice);
    if (k0step >= ksteps) {
      // empty tail slice: the combine sums EVERY slab, so write zeros
      // (returning left recycled-allocation garbage — order-dependent)
      for (int r = threadIdx.x; r < BMN; r += TPB) {
        float* row = WS + ((long)blockIdx.z * M + m0 + r) * N + n0;
        for (int c = 0; c < BMN; ++c) row[c] = 0.f;
      }
      return;
    }
  }
  const long kb0 = (long)k0step * BK;
  // prologue: stage tile 0 (and, 3-slot form, tile 1)
  glds_tile(X + (long)m0 * K + kb0, K, smem);
  glds_tile(Y + (long)n0 * K + kb0, K, smem + TILE);
  if constexpr (PIPE3) {
    if (ksteps - k0step > 1) {
      glds_tile(X + (long)m0 * K + kb0 + BK, K, smem + 2 * TILE);
      glds_tile(Y + (long)n0 * K + kb0 + BK, K, smem + 3 * TILE);
    }
  } else {
    __syncthreads();
  }

  for (int kt = k0step; kt < ksteps; ++kt) {
    const int rel = kt - k0step;
    const int buf = PIPE3 ? rel % 3 : (rel & 1);
    if constexpr (PIPE3) {
      // T3/T4 counted-vmcnt 3-slot ring: wait for tile kt's 8 glds (leave
      // tile kt+1's 8 in flight), raw barrier (no vmcnt(0) drain), THEN
      // issue tile kt+2 into the slot whose readers finished before the
      // PREVIOUS barrier. Each wave's own pieces are covered by its own
      // vmcnt; the barrier publishes them to the other waves.
      if (kt + 1 < ksteps)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");  // tile kt landed
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // tail: nothing behind
      __builtin_amdgcn_s_barrier();
      if (kt + 2 < ksteps) {
        const int ns = (rel + 2) % 3;
        glds_tile(X + (long)m0 * K + (kt + 2) * BK, K, smem + ns * 2 * TILE);
        glds_tile(Y + (long)n0 * K + (kt + 2) * BK, K, smem + ns * 2 * TILE + TILE);
      }
    } else {
      if (kt + 1 < ksteps) {  // issue next-tile staging FIRST (T3 min-2-phase)
        glds_tile(X + (long)m0 * K + (long)(kt + 1) * BK, K, smem + (buf ^ 1) * 2 * TILE);
        glds_tile(Y + (long)n0 * K + (long)(kt + 1) * BK, K, smem + (buf ^ 1) * 2 * TILE + TILE);
      }
    }
    const __hip_bfloat16* ax = smem + buf * 2 * TILE;
    const __hip_bfloat16* by = ax + TILE;
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kc = 0; kc < BK / 16; ++kc) {
      bf16x8 af[2], bf[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) af[i] = img_frag(ax, wr + i * 32, kc, lane);
#pragma unroll
      for (int j = 0; j < 2; ++j) bf[j] = img_frag(by, wc + j * 32, kc, lane);
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    if constexpr (!PIPE3) {
      __syncthreads();  // drains the in-flight glds (vmcnt(0)) + buffer swap
    }
  }
  if constexpr (PIPE3) __syncthreads();  // epilogue joins before C stores

  // epilogue: C = alpha*acc + beta*E, element (m,n) at lane n=lq, reg row
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int m = m0 + wr + i * 32 + acc_row(reg, hi);
        const int n = n0 + wc + j * 32 + lq;
        if constexpr (SPLIT) {
          WS[((long)blockIdx.z * M + m) * N + n] = acc[i][j][reg];
        } else {
          float v = alpha * acc[i][j][reg];
          if constexpr (HAS_E) v += beta * to_f32(E[(long)m * N + n]);
          from_f32(&C[(long)m * N + n], v);
        }
      }
}

// combine the split-K slabs: C = alpha * sum_s WS[s] + beta * E
template <bool HAS_E, typename TO>
__global__ void muon_combine_kernel(const float* __restrict__ WS, const TO* __restrict__ E,
                                    TO* __restrict__ C, long mn, int slices,
                                    float alpha, float beta) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < mn;
       i += gridDim.x * (long)blockDim.x) {
    float v = 0.f;
    for (int s = 0; s < slices; ++s) v += WS[(long)s * mn + i];
    v *= alpha;
    if constexpr (HAS_E) v += beta * to_f32(E[i]);
    from_f32(&C[i], v);
  }
}

// ---------------------------------------------------------------------------
// C[M,N] = Bm[M,K] @ X[K,N] + a * X2[M,N]   (X2 = the same X when M == K)
template <bool SPLIT = false>
__global__ __launch_bounds__(TPB) void muon_gemm_nn_ax_kernel(
    const __hip_bfloat16* __restrict__ Bm, const __hip_bfloat16* __restrict__ X,
    const __hip_bfloat16* __restrict__ X2, __hip_bfloat16* __restrict__ C,
    int M, int N, int K, float a,
    float* __restrict__ WS = nullptr, int ksteps_per_slice = 0) {
  constexpr int TILE = BMN * BK;   // A image [128 m][64 k]
  constexpr int VROW = 64;         // transposed X image rows [128 n][64 k]
  __shared__ __hip_bfloat16 smem[2 * (TILE + BMN * VROW)];

  const int tm = blockIdx.x, tn = blockIdx.y;
  const int m0 = tm * BMN, n0 = tn * BMN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = (wave >> 1) * 64, wc = (wave & 1) * 64;
  const int lq = lane & 31, hi = lane >> 5;

  // register staging for the transposed X image (glds cannot transpose):
  // chunks of (2 k-rows x 8 n-cols), pair-packed b32 writes (attention's V
  // staging pattern; conflict-free reads via swzt)
  constexpr int CH = (BK / 2) * (BMN / 8) / TPB;  // = 4 chunks per thread
  uint4 xreg[CH][2];

  auto xstage_load = [&](int k0) {
#pragma unroll
    for (int c = 0; c < CH; ++c) {
      const int u = tid + c * TPB;
      const int kr = (u / (BMN / 8)) * 2;
      const int n = (u % (BMN / 8)) * 8;
      const __hip_bfloat16* base = X + (long)(k0 + kr) * N + n0 + n;
      xreg[c][0] = *reinterpret_cast<const uint4*>(base);
      xreg[c][1] = *reinterpret_cast<const uint4*>(base + N);
    }
  };
  auto xstage_write = [&](__hip_bfloat16* xt) {
#pragma unroll
    for (int c = 0; c < CH; ++c) {
      const int u = tid + c * TPB;
      const int kr = (u / (BMN / 8)) * 2;
      const int n = (u % (BMN / 8)) * 8;
      Bf16x8U v0, v1;
      *reinterpret_cast<uint4*>(v0.s) = xreg[c][0];
      *reinterpret_cast<uint4*>(v1.s) = xreg[c][1];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int nrow = n + j;
        const uint pair = (uint)v0.s[j] | ((uint)v1.s[j] << 16);
        *reinterpret_cast<uint*>(xt + nrow * VROW + (kr ^ swzt(nrow))) = pair;
      }
    }
  };

  f32x16 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = f32x16{};

  int k0step = 0, ksteps = K / BK;
  if constexpr (SPLIT) {
    k0step = blockIdx.z * ksteps_per_slice;
    ksteps = min(ksteps, k0step + ksteps_per_slice);
    if (k0step >= ksteps) {
      for (int r = threadIdx.x; r < BMN; r += TPB) {
        float* row = WS + ((long)blockIdx.z * M + m0 + r) * N + n0;
        for (int c = 0; c < BMN; ++c) row[c] = 0.f;
      }
      return;
    }
  }
  glds_tile(Bm + (long)m0 * K + (long)k0step * BK, K, smem);
  xstage_load(k0step * BK);
  xstage_write(smem + TILE);
  __syncthreads();

  for (int kt = k0step; kt < ksteps; ++kt) {
    const int buf = (kt - k0step) & 1;
    __hip_bfloat16* cur = smem + buf * (TILE + BMN * VROW);
    __hip_bfloat16* nxt = smem + (buf ^ 1) * (TILE + BMN * VROW);
    const bool has_next = kt + 1 < ksteps;
    if (has_next) {
      glds_tile(Bm + (long)m0 * K + (kt + 1) * BK, K, nxt);
      xstage_load((kt + 1) * BK);  // loads in flight under the MFMAs (T14)
    }
    const __hip_bfloat16* aimg = cur;
    const __hip_bfloat16* xt = cur + TILE;
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kc = 0; kc < BK / 16; ++kc) {
      bf16x8 af[2], bf[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) af[i] = img_frag(aimg, wr + i * 32, kc, lane);
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int nrow = wc + j * 32 + lq;
        Bf16x8U u;
        *reinterpret_cast<uint4*>(u.s) = *reinterpret_cast<const uint4*>(
            xt + nrow * VROW + ((kc * 16 + hi * 8) ^ swzt(nrow)));
        bf[j] = u.v;
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    if (has_next) xstage_write(nxt + TILE);
    __syncthreads();
  }

#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int m = m0 + wr + i * 32 + acc_row(reg, hi);
        const int n = n0 + wc + j * 32 + lq;
        if constexpr (SPLIT) {
          WS[((long)blockIdx.z * M + m) * N + n] = acc[i][j][reg];
        } else {
          const float v = acc[i][j][reg] + a * __bfloat162float(X2[(long)m * N + n]);
          C[(long)m * N + n] = __float2bfloat16(v);
        }
      }
}


// ---------------------------------------------------------------------------
// 256x256 quadrant-phased NT GEMM: C[M,N] = alpha * X[M,K] @ Y[N,K]^T.
// A step toward the guide's 8-phase template sized for what is provably
// race-free without the template's region-level buffer reuse (a 3-tile ring
// does not fit 160 KiB at this tile size):
//   - BK=64, 8 waves (2Mx4N, 512 thr), 128x64 per wave, 64 MFMAs
//     (16x16x32 bf16) per wave per K-tile in FOUR quadrant phases of
//     {12 ds_read_b128 | stage 1 half-tile of tile kt+1 | barrier |
//      lgkmcnt | 16 MFMA | barrier};
//   - 2 K-tile LDS buffers (128 KiB): tile kt+1 is staged into buffer
//     (kt+1)&1, whose readers finished before tile kt's first barrier;
//   - st_16x32 XOR swizzle (byte bit5 ^= bit9) on the glds SOURCE address
//     and the ds_read address; LDS destination stays lane-linear
//     (guide §5.4 rule 21);
//   - the full counted-vmcnt 8-phase schedule (2 tiles in flight) is the
//     known next rung; it needs the template's region-reuse proof.
// Requires M,N % 256 == 0, K % 64 == 0 (the wrapper pads); SPLIT writes
// fp32 slabs for muon_combine_kernel.
template <bool SPLIT>
__global__ __launch_bounds__(512) void muon_gemm_nt8_kernel(
    const __hip_bfloat16* __restrict__ X, const __hip_bfloat16* __restrict__ Y,
    __hip_bfloat16* __restrict__ C, int M, int N, int K,
    float alpha, float* __restrict__ WS, int ksteps_per_slice) {
  typedef __attribute__((ext_vector_type(4))) float f32x4;
  extern __shared__ __attribute__((aligned(16))) __hip_bfloat16 smem8[];
  constexpr int HT = 128 * 64;  // elements per half-tile image ([128][64])
  const int tm = blockIdx.x, tn = blockIdx.y;
  const int m0 = tm * 256, n0 = tn * 256;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const int wr = (wave >> 2) * 128;  // 2 M-wave rows
  const int wc = (wave & 3) * 64;    // 4 N-wave cols
  const int l15 = lane & 15, l4 = lane >> 4;

  int k0step = 0, ktiles = K / 64;
  if constexpr (SPLIT) {
    k0step = blockIdx.z * ksteps_per_slice;
    ktiles = min(ktiles, k0step + ksteps_per_slice);
    if (k0step >= ktiles) {
      for (int r = tid; r < 256; r += 512) {
        float* row = WS + ((long)blockIdx.z * M + m0 + r) * N + n0;
        for (int c = 0; c < 256; ++c) row[c] = 0.f;
      }
      return;
    }
  }

  // stage one [128][64] half-tile by glds: each wave moves 2 KiB (2 x 1 KiB)
  auto stage_half = [&](const __hip_bfloat16* src_rows, int kt, __hip_bfloat16* dst) {
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      const int off16 = (wave * 2 + g) * 64 + lane;      // 16-B unit in image
      const unsigned pb = (unsigned)off16 * 16u;          // physical byte
      const unsigned lb = pb ^ (((pb >> 9) & 1u) << 5);   // logical byte
      const int row = (int)(lb >> 7);
      const int col = (int)(lb & 127) >> 1;               // element
      const __hip_bfloat16* gp = src_rows + (long)row * K + kt * 64 + col;
      __hip_bfloat16* lp = dst + (long)(wave * 2 + g) * 512;  // wave-uniform
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gp,
          (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
    }
  };
  // half h of tile t: {0:A rows 0-127, 1:A 128-255, 2:Y 0-127, 3:Y 128-255}
  auto stage_phase = [&](int h, int t) {
    if (t >= ktiles) return;
    const __hip_bfloat16* base = (h < 2) ? X + (long)(m0 + (h & 1) * 128) * K
                                         : Y + (long)(n0 + (h & 1) * 128) * K;
    stage_half(base, t, smem8 + ((t & 1) * 4 + h) * HT);
  };
  // fragment read (A-style for both operands: row-per-lane, 16 contiguous B)
  auto frag = [&](const __hip_bfloat16* img, int r, int chunk) -> bf16x8 {
    const unsigned lbyte = (unsigned)r * 128u + (unsigned)chunk * 16u;
    const unsigned pbyte = lbyte ^ (((lbyte >> 9) & 1u) << 5);
    Bf16x8U u;
    *reinterpret_cast<uint4*>(u.s) = *reinterpret_cast<const uint4*>(
        reinterpret_cast<const char*>(img) + pbyte);
    return u.v;
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{};

  // prologue: stage tile k0 fully; barrier includes the vmcnt(0) drain
  for (int h = 0; h < 4; ++h) stage_phase(h, k0step);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int kt = k0step; kt < ktiles; ++kt) {
    const __hip_bfloat16* Ah = smem8 + ((kt & 1) * 4 + (wr ? 1 : 0)) * HT;
    const __hip_bfloat16* Bh = smem8 + ((kt & 1) * 4 + 2 + (wc >> 7)) * HT;
    const int brow = wc & 127;
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      // quadrant q: m-frags 2q, 2q+1 x 4 n-frags x K=64 (2 k-steps)
      bf16x8 af[2][2], bfr[4][2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          af[mi][kk] = frag(Ah, (q * 2 + mi) * 16 + l15, kk * 4 + l4);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          bfr[ni][kk] = frag(Bh, brow + ni * 16 + l15, kk * 4 + l4);
      stage_phase(q, kt + 1);  // buffer (kt+1)&1: readers done last tile
      // no per-quadrant barrier needed: reads target the stable buffer,
      // writes the other one; hipcc places counted lgkm waits for the
      // ds_read->MFMA edges itself
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[q * 2 + mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[mi][kk], bfr[ni][kk], acc[q * 2 + mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    // tile kt+1's 8 glds (per wave) must land before its first read
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: C/D 16x16 layout: col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wr + i * 16 + l4 * 4 + r;
        const int n = n0 + wc + j * 16 + l15;
        if constexpr (SPLIT) {
          WS[((long)blockIdx.z * M + m) * N + n] = acc[i][j][r];
        } else {
          from_f32(&C[(long)m * N + n], alpha * acc[i][j][r]);
        }
      }
}

void check_2d_bf16(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16 && t.dim() == 2 &&
                  t.is_contiguous(),
              "muon: ", name, " must be a contiguous 2-D bf16 CUDA tensor");
}

}  // namespace

void muon_gemm_nt(at::Tensor X, at::Tensor Y, at::Tensor C, double alpha,
                  double beta, at::Tensor E) {
  check_2d_bf16(X, "X");
  check_2d_bf16(Y, "Y");
  check_2d_bf16(C, "C");
  const int M = X.size(0), K = X.size(1), N = Y.size(0);
  TORCH_CHECK(Y.size(1) == K && C.size(0) == M && C.size(1) == N,
              "muon_gemm_nt: shape mismatch");
  TORCH_CHECK(M % BMN == 0 && N % BMN == 0 && K % BK == 0,
              "muon_gemm_nt: pad to 128/128/64 multiples");
  const bool has_e = E.numel() > 0;
  if (has_e) check_2d_bf16(E, "E");
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(M / BMN, N / BMN), block(TPB);
  auto* xp = reinterpret_cast<const __hip_bfloat16*>(X.data_ptr());
  auto* yp = reinterpret_cast<const __hip_bfloat16*>(Y.data_ptr());
  auto* ep = has_e ? reinterpret_cast<const __hip_bfloat16*>(E.data_ptr()) : nullptr;
  auto* cp = reinterpret_cast<__hip_bfloat16*>(C.data_ptr());
  static const bool pipe3 = []() {
    const char* e = getenv("MCDP_MUON_PIPE3");
    return e && atoi(e) != 0;
  }();
  // split-K when the tile grid under-fills the chip (env override: 0 = off)
  const int nblocks = (M / BMN) * (N / BMN);
  int splitk = 1;
  static const int sk_env = []() {
    const char* e = getenv("MCDP_MUON_SPLITK");
    return e ? atoi(e) : -1;
  }();
  if (sk_env >= 0) splitk = sk_env > 0 ? sk_env : 1;
  else if (nblocks < 384 && (K / BK) >= 2) {  // < 1.5x CUs: 2 blocks/CU co-residency needs >= 384
    splitk = 2;
    while (nblocks * splitk * 2 <= 512 && splitk < 8 && (K / BK) / (splitk * 2) >= 2)
      splitk *= 2;
  }
  // 256^2 quadrant-phase kernel for aligned shapes (bigger tile halves the
  // staging traffic per FLOP; env MCDP_MUON_NT8=0 disables)
  // default OFF: correct but measured SLOWER than the 128^2 split-K kernel
  // (0.56-0.97x vs 0.75-1.04x of hipBLASLt, r2_nt8.log) — the 256^2 tile
  // only pays with the full counted-vmcnt region-reuse schedule (the
  // guide's 8-phase template), which needs its own race proof; this kernel
  // is the scaffold for that rung.
  static const bool use_nt8 = []() {
    const char* e = getenv("MCDP_MUON_NT8");
    return e && atoi(e) != 0;
  }();
  const bool nt8_ok = use_nt8 && (M % 256 == 0) && (N % 256 == 0);
  const size_t nt8_lds = 2 * 4 * 128 * 64 * sizeof(__hip_bfloat16);  // 128 KiB
  static const bool nt8_attr = []() {  // >64 KiB dynamic LDS needs the opt-in
    hipFuncSetAttribute(reinterpret_cast<const void*>(&muon_gemm_nt8_kernel<false>),
                        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    hipFuncSetAttribute(reinterpret_cast<const void*>(&muon_gemm_nt8_kernel<true>),
                        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    return true;
  }();
  (void)nt8_attr;
  if (nt8_ok) {
    const int nblk8 = (M / 256) * (N / 256);
    int sk8 = 1;
    if (nblk8 < 192 && (K / BK) >= 2) {
      sk8 = 2;
      while (nblk8 * sk8 * 2 <= 512 && sk8 < 16 && (K / BK) / (sk8 * 2) >= 2)
        sk8 *= 2;
    }
    dim3 g8(M / 256, N / 256, sk8), b8(512);
    if (sk8 > 1 || has_e) {
      const int kps = (K / BK + sk8 - 1) / sk8;
      auto ws = at::empty({(long)sk8 * M * N}, X.options().dtype(at::kFloat));
      muon_gemm_nt8_kernel<true><<<g8, b8, nt8_lds, stream>>>(
          xp, yp, nullptr, M, N, K, 1.f, ws.data_ptr<float>(), kps);
      const long mn = (long)M * N;
      const int cg = (int)std::min<long>((mn + 1023) / 1024, 2048);
      if (has_e)
        muon_combine_kernel<true, __hip_bfloat16><<<cg, 1024, 0, stream>>>(
            ws.data_ptr<float>(), ep, cp, mn, sk8, (float)alpha, (float)beta);
      else
        muon_combine_kernel<false, __hip_bfloat16><<<cg, 1024, 0, stream>>>(
            ws.data_ptr<float>(), ep, cp, mn, sk8, (float)alpha, (float)beta);
    } else {
      muon_gemm_nt8_kernel<false><<<g8, b8, nt8_lds, stream>>>(
          xp, yp, cp, M, N, K, (float)alpha, nullptr, K / BK);
    }
    return;
  }
  if (splitk > 1) {
    const int ksteps = K / BK;
    const int kps = (ksteps + splitk - 1) / splitk;
    auto ws = at::empty({(long)splitk * M * N}, X.options().dtype(at::kFloat));
    dim3 gs(M / BMN, N / BMN, splitk);
    muon_gemm_nt_kernel<false, __hip_bfloat16, false, true><<<gs, block, 0, stream>>>(
        xp, yp, nullptr, nullptr, M, N, K, 1.f, 0.f, ws.data_ptr<float>(), kps);
    const long mn = (long)M * N;
    const int cg = (int)std::min<long>((mn + 1023) / 1024, 2048);
    if (has_e)
      muon_combine_kernel<true, __hip_bfloat16><<<cg, 1024, 0, stream>>>(
          ws.data_ptr<float>(), ep, cp, mn, splitk, (float)alpha, (float)beta);
    else
      muon_combine_kernel<false, __hip_bfloat16><<<cg, 1024, 0, stream>>>(
          ws.data_ptr<float>(), ep, cp, mn, splitk, (float)alpha, (float)beta);
    return;
  }
  if (has_e) {
    if (pipe3)
      muon_gemm_nt_kernel<true, __hip_bfloat16, true><<<grid, block, 0, stream>>>(
          xp, yp, ep, cp, M, N, K, (float)alpha, (float)beta);
    else
      muon_gemm_nt_kernel<true><<<grid, block, 0, stream>>>(xp, yp, ep, cp, M, N, K,
                                                            (float)alpha, (float)beta);
  } else {
    if (pipe3)
      muon_gemm_nt_kernel<false, __hip_bfloat16, true><<<grid, block, 0, stream>>>(
          xp, yp, ep, cp, M, N, K, (float)alpha, (float)beta);
    else
      muon_gemm_nt_kernel<false><<<grid, block, 0, stream>>>(xp, yp, ep, cp, M, N, K,
                                                             (float)alpha, (float)beta);
  }
}

// K9 Shampoo statistics EMA on the same MFMA NT kernel, fp32 state:
// S = beta*S + (1-beta) * G @ G^T   (left stats; pass G^T for right stats).
// In-place E==C is safe: each thread reads its E element once before the
// write. Parity: /root/reference/optimizers/shampoo.py:229-255.
void shampoo_stats_update(at::Tensor G, at::Tensor S, double beta) {
  check_2d_bf16(G, "G");
  TORCH_CHECK(S.is_cuda() && S.scalar_type() == at::kFloat && S.dim() == 2 &&
                  S.is_contiguous(), "shampoo: S must be contiguous fp32 2-D");
  const int M = G.size(0), K = G.size(1);
  TORCH_CHECK(S.size(0) == M && S.size(1) == M, "shampoo: S must be [M,M]");
  TORCH_CHECK(M % BMN == 0 && K % BK == 0, "shampoo: pad to 128/64 multiples");
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(M / BMN, M / BMN), block(TPB);
  auto* gp = reinterpret_cast<const __hip_bfloat16*>(G.data_ptr());
  auto* sp = S.data_ptr<float>();
  muon_gemm_nt_kernel<true, float><<<grid, block, 0, stream>>>(
      gp, gp, sp, sp, M, M, K, (float)(1.0 - beta), (float)beta);
}

void muon_gemm_nn_ax(at::Tensor Bm, at::Tensor X, at::Tensor C, double a) {
  check_2d_bf16(Bm, "Bm");
  check_2d_bf16(X, "X");
  check_2d_bf16(C, "C");
  const int M = Bm.size(0), K = Bm.size(1), N = X.size(1);
  TORCH_CHECK(X.size(0) == K && C.size(0) == M && C.size(1) == N,
              "muon_gemm_nn_ax: shape mismatch");
  TORCH_CHECK(M % BMN == 0 && N % BMN == 0 && K % BK == 0,
              "muon_gemm_nn_ax: pad to 128/128/64 multiples");
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(M / BMN, N / BMN), block(TPB);
  auto* bp = reinterpret_cast<const __hip_bfloat16*>(Bm.data_ptr());
  auto* xp = reinterpret_cast<const __hip_bfloat16*>(X.data_ptr());
  auto* cp = reinterpret_cast<__hip_bfloat16*>(C.data_ptr());
  const int nblocks = (M / BMN) * (N / BMN);
  int splitk = 1;
  if (nblocks < 384 && (K / BK) >= 2) {
    splitk = 2;
    while (nblocks * splitk * 2 <= 512 && splitk < 8 && (K / BK) / (splitk * 2) >= 2)
      splitk *= 2;
  }
  if (splitk > 1) {
    const int kps = (K / BK + splitk - 1) / splitk;
    auto ws = at::empty({(long)splitk * M * N}, X.options().dtype(at::kFloat));
    dim3 gs(M / BMN, N / BMN, splitk);
    muon_gemm_nn_ax_kernel<true><<<gs, block, 0, stream>>>(
        bp, xp, xp, cp, M, N, K, (float)a, ws.data_ptr<float>(), kps);
    const long mn = (long)M * N;
    const int cg = (int)std::min<long>((mn + 1023) / 1024, 2048);
    // combine with the +a*X epilogue: reuse muon_combine with E = X, beta = a
    muon_combine_kernel<true, __hip_bfloat16><<<cg, 1024, 0, stream>>>(
        ws.data_ptr<float>(), xp, cp, mn, splitk, 1.f, (float)a);
    return;
  }
  muon_gemm_nn_ax_kernel<false><<<grid, block, 0, stream>>>(bp, xp, xp, cp, M, N, K, (float)a);
}
