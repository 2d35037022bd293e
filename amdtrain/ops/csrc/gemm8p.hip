// 256x256-tile bf16 BT-GEMM for gfx950 — staged port of the measured
// "8-phase" CDNA4 schedule (cdna_hip_programming.md §5: 1563-1728 TF vs the
// ~900 TF ceiling of the 128x128 2-barrier structure).
//
// Build-up is incremental and each stage is refchecked on hardware before
// the next lands (the guide's m152 lesson: naive combination races):
//   stage 1 (this file's baseline): 256x256 tile, BK=64, 8 waves (2m x 4n),
//     linear LDS, global_load_lds staging, one barrier pair per K-step,
//     64 MFMA per wave between barriers.
//   stage 2: st-swizzle (pre-swizzled global source + swizzled ds_read).
//   stage 3: phase-split with counted vmcnt + s_setprio.
//
// Opt-in: the dispatcher uses it only when AMDTRAIN_GEMM8P=1 and the shape
// qualifies (K % 64 == 0, N >= 128); the proven 128x128 kernel stays the
// default until this one measures faster end-to-end.
#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int TPB8 = 512;  // 8 waves
constexpr int BM8 = 256, BN8 = 256, BK8 = 64;

__device__ __forceinline__ int xcd_swz8(int bid, int nwg) {
  constexpr int NXCD = 8;
  if (nwg < NXCD) return bid;
  int xcd = bid % NXCD, idx = bid / NXCD;
  int q = nwg / NXCD, r = nwg % NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

// stage a [256 rows][64 cols] bf16 tile: 32 KiB = 2048 x 16B units,
// 4 units per thread; rows are 128 B (8 units per row)
__device__ __forceinline__ void stage_256x64(
    const bf16* __restrict__ g, long ld, long row0, long rows, long k0,
    bf16* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int rnd = 0; rnd < 4; ++rnd) {
    int unit = rnd * TPB8 + t;           // 0..2047
    long row = row0 + (unit >> 3);       // 8 x 16B units per 128B row
    if (row >= rows) row = rows - 1;
    int koff = (unit & 7) * 8;           // 8 bf16 per unit
    const bf16* src = g + row * ld + k0 + koff;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(lds + unit * 8), 16,
        0, 0);
  }
}

__global__ void __launch_bounds__(TPB8, 1)
gemm_bt_8p_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                  bf16* __restrict__ C, long M, long N, long K, int nbm,
                  int nbn) {
  __shared__ bf16 As[BM8 * BK8];
  __shared__ bf16 Bs[BN8 * BK8];

  const int bid = xcd_swz8(blockIdx.x, nbm * nbn);
  const int bm = bid / nbn, bn = bid % nbn;
  const long m0 = (long)bm * BM8, n0 = (long)bn * BN8;

  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE;       // 0..7
  const int lane = t % AMD_WAVE;
  const int wr = (wave >> 2) * 128;    // wave row offset (2 m-waves)
  const int wc = (wave & 3) * 64;      // wave col offset (4 n-waves)
  const int fr = lane & 15;
  const int fq = lane >> 4;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const long ksteps = K / BK8;
  for (long kt = 0; kt < ksteps; ++kt) {
    __syncthreads();
    stage_256x64(A, K, m0, M, kt * BK8, As);
    stage_256x64(B, K, n0, N, kt * BK8, Bs);
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 b[4];
#pragma unroll
      for (int j = 0; j < 4; ++j)
        b[j] = *(const bf16x8*)&Bs[(wc + j * 16 + fr) * BK8 + kk * 32 +
                                   fq * 8];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        bf16x8 a = *(const bf16x8*)&As[(wr + i * 16 + fr) * BK8 + kk * 32 +
                                       fq * 8];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b[j], acc[i][j], 0, 0, 0);
      }
    }
  }

#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wr + i * 16 + fq * 4 + r;
        long col = n0 + wc + j * 16 + fr;
        if (row < M && col < N)
          C[row * N + col] = __float2bfloat16(acc[i][j][r]);
      }
}

}  // namespace

at::Tensor gemm_bt_8p(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16);
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  long M = Ac.size(0), K = Ac.size(1), N = Bc.size(0);
  TORCH_CHECK(Bc.size(1) == K && K % BK8 == 0,
              "gemm_bt_8p needs K % 64 == 0");
  auto C = at::empty({M, N}, Ac.options());
  int nbm = (int)((M + BM8 - 1) / BM8), nbn = (int)((N + BN8 - 1) / BN8);
  auto stream = at::cuda::getCurrentCUDAStream();
  gemm_bt_8p_kernel<<<nbm * nbn, TPB8, 0, stream>>>(
      (const bf16*)Ac.const_data_ptr(), (const bf16*)Bc.const_data_ptr(),
      (bf16*)C.data_ptr(), M, N, K, nbm, nbn);
  CHECK_CUDA_OK();
  return C;
}
