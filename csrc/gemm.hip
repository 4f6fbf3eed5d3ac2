// f32 MFMA GEMM for the ML path (normal equations / linear models).
//
// Uses gfx950's exact-f32 matrix instruction v_mfma_f32_16x16x4_f32
// (guide §3: f32-in/f32-accumulate at the 157 TF f32 vector rate, ≈2.4x a
// VALU f32 GEMM, bitwise equal to an fmaf chain).  LDS-tiled 64x64 block,
// 4 waves per block each owning a 32x32 sub-tile (2x2 fragments of 16x16),
// +1-element LDS row padding against bank conflicts (guide Guideline 4).
// Correctness-first: the ML fits it backs are K-reduction bound, not
// GEMM-peak bound.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));

#define BM 64
#define BN 64
#define BK 32

__global__ __launch_bounds__(256) void gemm_f32_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, float beta) {
  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BK][BN + 1];
  int block_row = blockIdx.y * BM;
  int block_col = blockIdx.x * BN;
  int wave = threadIdx.x / WAVE;  // 0..3 -> (wr, wc) in 2x2
  int lane = threadIdx.x % WAVE;
  int wr = wave / 2, wc = wave % 2;
  f32x4 acc[2][2] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    // cooperative staging: 256 threads load 64x32 A tile and 32x64 B tile
    for (int idx = threadIdx.x; idx < BM * BK; idx += 256) {
      int r = idx / BK, c = idx % BK;
      int gr = block_row + r, gc = k0 + c;
      As[r][c] = (gr < M && gc < K) ? A[(int64_t)gr * K + gc] : 0.0f;
    }
    for (int idx = threadIdx.x; idx < BK * BN; idx += 256) {
      int r = idx / BN, c = idx % BN;
      int gr = k0 + r, gc = block_col + c;
      Bs[r][c] = (gr < K && gc < N) ? B[(int64_t)gr * N + gc] : 0.0f;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      // lane mapping for v_mfma_f32_16x16x4_f32 (guide §3):
      //   A operand: A[i = lane&15][k = lane>>4]
      //   B operand: B[k = lane>>4][j = lane&15]
      int i = lane & 15, kq = lane >> 4;
      float a0 = As[wr * 32 + i][kk + kq];
      float a1 = As[wr * 32 + 16 + i][kk + kq];
      float b0 = Bs[kk + kq][wc * 32 + i];
      float b1 = Bs[kk + kq][wc * 32 + 16 + i];
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }
  // C/D layout (guide §3, dtype-independent): col = lane&15,
  // row = (lane>>4)*4 + reg
  int jc = lane & 15, rbase = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int n = 0; n < 2; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = block_row + wr * 32 + m * 16 + rbase + r;
        int col = block_col + wc * 32 + n * 16 + jc;
        if (row < M && col < N) {
          int64_t off = (int64_t)row * N + col;
          C[off] = acc[m][n][r] + (beta != 0.0f ? beta * C[off] : 0.0f);
        }
      }
    }
  }
}

torch::Tensor gemm_f32(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2 && A.size(1) == B.size(0));
  TORCH_CHECK(A.dtype() == torch::kFloat32 && B.dtype() == torch::kFloat32);
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  int M = A.size(0), K = A.size(1), N = B.size(1);
  auto C = torch::empty({M, N},
                        torch::dtype(torch::kFloat32).device(A.device()));
  dim3 grid((N + BN - 1) / BN, (M + BM - 1) / BM);
  hipLaunchKernelGGL(gemm_f32_kernel, grid, dim3(256), 0,
                     c10::hip::getCurrentHIPStream().stream(),
                     (const float*)Ac.data_ptr(), (const float*)Bc.data_ptr(),
                     (float*)C.data_ptr(), M, N, K, 0.0f);
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));
  return C;
}
