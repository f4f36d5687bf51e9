// Layout self-checks: a single 32x32x16 MFMA tile exposed to Python so the
// A/B/C lane mappings (attn_common.h) and the acc_to_afrag transform can be
// verified on hardware against torch matmul with ASYMMETRIC inputs
// (guide §5.4 rule 16: symmetric inputs miss transposes).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "attn_common.h"

namespace {

// C[32,32] = A[32,16] @ B[16,32] using the documented fragment layouts.
__global__ void mfma_tile_kernel(const __hip_bfloat16* __restrict__ A,
                                 const __hip_bfloat16* __restrict__ B,
                                 float* __restrict__ C) {
  const int lane = threadIdx.x;
  const int hi = lane >> 5;
  Bf16x8U a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a.h[j] = A[(lane & 31) * 16 + hi * 8 + j];      // A[l&31][8*hi + j]
    b.h[j] = B[(hi * 8 + j) * 32 + (lane & 31)];    // B[8*hi + j][l&31]
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a.v, b.v, acc, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 16; ++reg)
    C[acc_row(reg, hi) * 32 + (lane & 31)] = acc[reg];
}

// Verify acc_to_afrag: load M[32][32] into the acc layout, transform, and
// write out the claimed A-fragments as the matrix A[32][32] = M^T they are
// supposed to represent (slot ks covers columns 16ks..16ks+15).
__global__ void afrag_transform_kernel(const float* __restrict__ M, float* __restrict__ Aout) {
  const int lane = threadIdx.x;
  const int hi = lane >> 5;
  float p[16];
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) p[reg] = M[acc_row(reg, hi) * 32 + (lane & 31)];
  bf16x8 f0, f1;
  acc_to_afrag(p, f0, f1);
  Bf16x8U u0, u1;
  u0.v = f0; u1.v = f1;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    Aout[(lane & 31) * 32 + 0 * 16 + hi * 8 + j] = to_f32(u0.h[j]);
    Aout[(lane & 31) * 32 + 1 * 16 + hi * 8 + j] = to_f32(u1.h[j]);
  }
}

// Verify tr16_frag: stage X[32][D] into the rm_swz image exactly as the bwd
// kernels do, then emit every (ks, dc) B-fragment; out[ks][dc][k8][c] should
// equal X[ks*16 + k8][dc*32 + c] if the hardware transpose-read delivery
// matches the probed semantics (probe/tr16_probe.hip).
template <int D>
__global__ void tr16_frag_kernel(const __hip_bfloat16* __restrict__ X,
                                 float* __restrict__ out) {
  __shared__ __align__(16) __hip_bfloat16 img[32 * D];
  const int lane = threadIdx.x % WAVE;
  const int tid = threadIdx.x;
  // stage: thread u writes rows 2u/(D/8) pair, 8 cols
  for (int u = tid; u < (32 / 2) * (D / 8); u += blockDim.x) {
    const int row = (u / (D / 8)) * 2;
    const int d0 = (u % (D / 8)) * 8;
    *reinterpret_cast<uint4*>(img + rm_swz<D>(row, d0)) =
        *reinterpret_cast<const uint4*>(X + row * D + d0);
    *reinterpret_cast<uint4*>(img + rm_swz<D>(row + 1, d0)) =
        *reinterpret_cast<const uint4*>(X + (row + 1) * D + d0);
  }
  __syncthreads();
  if (tid >= WAVE) return;
  const int hi = lane >> 5;
#pragma unroll
  for (int ks = 0; ks < 2; ++ks)
    for (int dc = 0; dc < D / 32; ++dc) {
      bf16x8 f = tr16_frag<D>(img, ks * 16 + hi * 8, dc * 32, lane);
      Bf16x8U u;
      u.v = f;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        out[((ks * (D / 32) + dc) * 16 + hi * 8 + j) * 32 + (lane & 31)] = to_f32(u.h[j]);
    }
}

}  // namespace

at::Tensor tr16_frag_test(at::Tensor X) {
  TORCH_CHECK(X.is_cuda() && X.scalar_type() == at::kBFloat16 && X.dim() == 2 && X.size(0) == 32);
  const int D = X.size(1);
  TORCH_CHECK(D == 64 || D == 128);
  auto out = at::zeros({2 * (D / 32), 16, 32}, X.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  if (D == 64)
    tr16_frag_kernel<64><<<1, 256, 0, stream>>>(
        reinterpret_cast<const __hip_bfloat16*>(X.contiguous().data_ptr()), out.data_ptr<float>());
  else
    tr16_frag_kernel<128><<<1, 256, 0, stream>>>(
        reinterpret_cast<const __hip_bfloat16*>(X.contiguous().data_ptr()), out.data_ptr<float>());
  return out;
}

at::Tensor mfma_tile_test(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.sizes() == at::IntArrayRef({32, 16}) && B.sizes() == at::IntArrayRef({16, 32}));
  auto C = at::zeros({32, 32}, A.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  mfma_tile_kernel<<<1, 64, 0, stream>>>(
      reinterpret_cast<const __hip_bfloat16*>(A.contiguous().data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(B.contiguous().data_ptr()),
      C.data_ptr<float>());
  return C;
}

at::Tensor afrag_transform_test(at::Tensor M) {
  TORCH_CHECK(M.is_cuda() && M.scalar_type() == at::kFloat);
  TORCH_CHECK(M.sizes() == at::IntArrayRef({32, 32}));
  auto A = at::zeros({32, 32}, M.options());
  auto stream = at::cuda::getCurrentHIPStream();
  afrag_transform_kernel<<<1, 64, 0, stream>>>(M.contiguous().data_ptr<float>(),
                                               A.data_ptr<float>());
  return A;
}
