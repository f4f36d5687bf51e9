// Shared pieces of the tiled attention kernels (K1/K2).
//
// MFMA tile convention (gfx950 mfma_f32_32x32x16_bf16, guide §3):
//   A[32][16]: lane l holds A[l&31][8*(l>>5)+j], j=0..7  (bf16x8)
//   B[16][32]: lane l holds B[8*(l>>5)+j][l&31]
//   C/D[32][32] f32x16: element (r,c) lives in lane c+32*((r>>2)&1),
//                       reg (r&3)+4*(r>>3); i.e. row r(reg,hi)=(reg&3)+8*(reg>>2)+4*hi.
//
// The workhorse trick (guide T12): QK^T is computed SWAPPED — mfma(K, Q)
// yields S^T with the q index lane-local (lane&31 = q column), so the online
// softmax runs entirely in registers; acc_to_afrag() then converts the
// exp'd accumulator into the A-operand fragments of the P·V MFMA with
// 8 v_cvt_pk_bf16_f32 + 4 v_permlane32_swap per 32x32 tile.
#pragma once
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// attention mask/score-mod variant codes (must match ops/attention.py)
#define MOD_NONE 0
#define MOD_CAUSAL 1
#define MOD_SLIDING_WINDOW 2
#define MOD_PREFIX_LM 3
#define MOD_ALIBI 4
#define MOD_BLOCKMASK 5  // arbitrary mask_mod via device block-mask + packed bits

__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned packed;
  // s_nop 1 = the 2 wait states of the gfx950 "VALU write -> v_permlane*
  // read" hazard (guide T21): the consumer is often permlane32_swap and
  // hipcc cannot pad an inline-asm write (it bit under sched_group_barrier
  // interleaving: every fwd numerics test failed until padded here).
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1" : "=v"(packed) : "v"(lo), "v"(hi));
  return packed;
}

union Bf16x8U {
  bf16x8 v;
  unsigned u[4];
  __hip_bfloat16 h[8];
  ushort s[8];
};

// Convert a 32x32 f32 accumulator (element M[r][c] at lane c+32*((r>>2)&1))
// into the two A-operand fragments of A = M^T (A[i][j] = M[j][i]):
// frag[ks] lane l = A[l&31][16*ks + 8*(l>>5) + j]. Guide T12.
__device__ __forceinline__ void acc_to_afrag(const float (&p)[16], bf16x8& f0, bf16x8& f1) {
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    unsigned a0 = cvt_pk_bf16(p[8 * ks + 0], p[8 * ks + 1]);
    unsigned a1 = cvt_pk_bf16(p[8 * ks + 2], p[8 * ks + 3]);
    unsigned b0 = cvt_pk_bf16(p[8 * ks + 4], p[8 * ks + 5]);
    unsigned b1 = cvt_pk_bf16(p[8 * ks + 6], p[8 * ks + 7]);
    auto r0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
    auto r1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
    Bf16x8U out;
    out.u[0] = r0[0];
    out.u[1] = r1[0];
    out.u[2] = r0[1];
    out.u[3] = r1[1];
    if (ks == 0) f0 = out.v; else f1 = out.v;
  }
}

// Cross-half (lane ^ 32) exchange WITHOUT ds_bpermute: __shfl_xor(x, 32)
// compiles to ds_bpermute + an lgkmcnt(0) drain, which ALSO waits for any
// prefetched ds_reads in flight (lgkmcnt counts both) — a serializer in the
// online-softmax critical path. v_permlane32_swap_b32 is a VALU op:
// permlane32_swap(x, x) returns r0 = {x.lo | x.lo}, r1 = {x.hi | x.hi}
// (each lane's r0/r1 hold the low/high-half values of its column), so
// max(r0, r1) / r0 + r1 IS the cross-half reduction in every lane.
__device__ __forceinline__ float cross32_max(float x) {
  union { float f; unsigned u; } a, r0, r1;
  a.f = x;
  auto r = __builtin_amdgcn_permlane32_swap(a.u, a.u, false, false);
  r0.u = r[0];
  r1.u = r[1];
  return fmaxf(r0.f, r1.f);
}

__device__ __forceinline__ float cross32_sum(float x) {
  union { float f; unsigned u; } a, r0, r1;
  a.f = x;
  auto r = __builtin_amdgcn_permlane32_swap(a.u, a.u, false, false);
  r0.u = r[0];
  r1.u = r[1];
  return r0.f + r1.f;
}

// row index held by (reg, hi) in a 32x32 accumulator
__device__ __forceinline__ int acc_row(int reg, int hi) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * hi;
}

// ---- ds_read_b64_tr_b16 (gfx950 hardware transpose read, guide T10) ----
//
// Semantics (probed on hardware, probe/tr16_probe.hip): per 16-lane group,
// each lane reads 4 contiguous bf16 (8 B, address MUST be 8-B aligned:
// misaligned returns the aligned address's data with no fault); the group's
// 64 elements form a window and dest lane i slot j receives window[16j + i]
// (window order = source lane i4 contributes window[4*i4 .. 4*i4+3]).
//
// Used here to read an MFMA B-fragment X[16 rows x 32 cols] DIRECTLY from a
// row-major LDS image (no transposed copy): dest lane l wants
// X[base + 8*(l>>5) + j][c = l&31]; per tr_read the window is a
// [4 row][16 col] block, so source lane i4 = l&15 (col group gd = (l>>4)&1)
// reads rows base + (i4>>2), 4 contiguous cols at 4*(i4&3).
//
// Bank structure: the image swizzle below makes BOTH the b128 row reads and
// the tr_read gather conflict-free (simulated + PMC-checked):
//   elem(q, d) at q*D + ((d & ~7) ^ (s_tr<D>(q) << 3)) + (d & 7)
// Constraints satisfied: s_tr bijective on every 16-row window (b128 phase
// covers all 64 banks) and s_tr>>1 distinct on every aligned 4-row window
// (the [4 row][16 col] tr gather covers 16 distinct 8-B bank slots).
template <int D>
__device__ __forceinline__ int s_tr(int q) {
  if constexpr (D >= 128) return ((q & 7) << 1) | ((q >> 3) & 1);
  else return (((q >> 1) & 3) << 1) | ((q >> 3) & 1);
}

// address (in elements) of the 8-aligned chunk holding cols d8..d8+7 of row q
template <int D>
__device__ __forceinline__ int rm_swz(int q, int d8) {
  return q * D + (d8 ^ (s_tr<D>(q) << 3));
}

// two tr_reads -> one bf16x8 B-operand fragment (rows qb..qb+7, col lane&31)
template <int D>
__device__ __forceinline__ bf16x8 tr16_frag(const __hip_bfloat16* img, int qb,
                                            int dcol0, int lane) {
  const int i4 = lane & 15, gd = (lane >> 4) & 1;
  const int dch = dcol0 + gd * 16 + 4 * (i4 & 3);
  const int q0 = qb + (i4 >> 2);
  const unsigned o0 = (unsigned)(unsigned long long)(
      img + q0 * D + ((dch & ~7) ^ (s_tr<D>(q0) << 3)) + (dch & 4));
  const unsigned o1 = (unsigned)(unsigned long long)(
      img + (q0 + 4) * D + ((dch & ~7) ^ (s_tr<D>(q0 + 4) << 3)) + (dch & 4));
  uint2 a, b;
  // "=&v" (early-clobber) is REQUIRED: ds_read is asynchronous, so without it
  // the allocator may alias an output pair with the second read's address
  // register, and the first read's writeback can corrupt that address if the
  // wave stalls between the two issues (observed: non-deterministic scattered
  // errors under multi-wave load; single-wave self-tests pass).
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(a), "=&v"(b)
      : "v"(o0), "v"(o1));
  Bf16x8U out;
  out.u[0] = a.x;
  out.u[1] = a.y;
  out.u[2] = b.x;
  out.u[3] = b.y;
  return out.v;
}

// Swizzle (element units) for TRANSPOSED LDS images stored as [D][64] rows
// (128 B rows). ds_write bank = (byte/4) % 32, so the row base (drow*32
// dwords) contributes nothing — the spread must come from the column offset.
// The staging writes touch d-rows stride 8 within a lane group ((drow&7)
// constant there), so the swizzle mixes BOTH drow&7 and (drow>>3)&7:
// conflict-free b128 reads (16-lane group covers all banks via the
// drow-parity split), 2-way b32 writes. Columns used must be < 64 elements.
__device__ __forceinline__ int swzt(int drow) {
  return (((drow & 7) ^ ((drow >> 3) & 7)) << 3);
}

// Causal block-skip makes per-block work a linear function of the x-tile
// index (q-tile: 1..Sq/KVB live kv tiles; kv-tile in the dKV kernel:
// mirrored). These kernels run ~1 block/CU, so the dispatcher refills CUs
// at stride gridDim.x*gridDim.y*gridDim.z / 256-ish = exactly 256 apart in
// linear id; with gridDim.x | 256 every refill hands a CU the SAME
// blockIdx.x — at S=2048 the all-late-tile CUs do 8x the mean work and set
// the wall clock. Rotating x by the (y,z)-group index (constant within a
// group -> bijective per group) makes each CU's successive blocks cycle
// through the x values instead.
__device__ __forceinline__ int balance_x() {
  const int gx = gridDim.x;
  int x = blockIdx.x;
  if ((256 % gx) == 0) {
    const int gid = blockIdx.y + gridDim.y * blockIdx.z;
    x = (x + gid / (256 / gx)) % gx;
  }
  return x;
}

// tile_map(): linear block id -> (tile, head, batch) with TWO goals on top
// of balance_x()'s rotation:
//   (a) every tile-block of one (batch, head) lands on ONE XCD (dispatch
//       places linear id i on XCD i%8), so the per-(b,h) operand stream
//       (K/V for fwd+dQ, Q/dO for dKV — ~1 MB at S=2048/D=128) is pulled
//       into that XCD's private 4 MiB L2 once and re-read from L2 instead
//       of being re-fetched by all 8 XCDs (attention bwd is memory-bound on
//       exactly this re-streaming);
//   (b) a CU's successive blocks rotate through tile indices (causal work
//       balance, same mechanism as balance_x).
// Head-major layout gives (a) when gridDim.y (heads) is a multiple of 8;
// bijectivity of the tile rotation needs nt | (256/nh). Falls back to
// balance_x() + natural (y,z) otherwise.
struct TileMap { int tile, head, batch; };
__device__ __forceinline__ TileMap tile_map() {
  const int nt = gridDim.x, nh = gridDim.y;
  TileMap m;
  if ((nh & 7) == 0 && (256 % nh) == 0 && ((256 / nh) % nt) == 0) {
    const long lin = blockIdx.x + (long)nt * (blockIdx.y + (long)nh * blockIdx.z);
    m.head = (int)(lin % nh);          // consecutive ids differ in head -> XCD = head % 8
    const long rest = lin / nh;        // = t0 + nt * batch
    const int t0 = (int)(rest % nt);
    m.batch = (int)(rest / nt);
    m.tile = (int)((t0 + rest / (256 / nh)) % nt);
  } else {
    m.tile = balance_x();
    m.head = blockIdx.y;
    m.batch = blockIdx.z;
  }
  return m;
}

// keep/mask decision for one score element. q_pos/k_pos are ABSOLUTE
// positions (q_pos = q_row + Skv - Sq handles KV-cache decode).
template <int MOD>
__device__ __forceinline__ bool attn_keep(int q_pos, int k_pos, int skv, int modarg) {
  if (k_pos >= skv) return false;
  if constexpr (MOD == MOD_NONE) return true;
  if constexpr (MOD == MOD_CAUSAL || MOD == MOD_ALIBI) return k_pos <= q_pos;
  if constexpr (MOD == MOD_SLIDING_WINDOW)
    return k_pos <= q_pos && (q_pos - k_pos) < modarg;
  if constexpr (MOD == MOD_PREFIX_LM) return k_pos <= q_pos || k_pos < modarg;
  return true;
}
