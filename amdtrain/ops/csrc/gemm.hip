// MFMA bf16 GEMM kernels for gfx950 — the compute core of the hand-written
// 1x1-convolution path ("Conv2d 1x1 (cuDNN) ... pointwise conv
// == GEMM", ~36 call sites per ResNet-50 forward).
//
// Structure follows the measured CDNA4 recipe (cdna_hip_programming.md §5,
// "m97" ladder step 3): 128x128 output tile, BK=32, 4 waves (2x2) per block,
// each wave computing a 64x64 sub-tile as 4x4 fragments of
// v_mfma_f32_16x16x32_bf16; A/B tiles staged to LDS with
// __builtin_amdgcn_global_load_lds (16 B per lane, the direct-to-LDS DMA);
// XCD-aware bijective blockIdx swizzle for L2 locality.
//
// Kernels:
//   gemm_bt:  C[M,N] = A[M,K] x B[N,K]^T  (bf16 in, bf16 or f32 out)
//             -> conv1x1 forward (A = NHWC activation rows, B = weight)
//             -> conv1x1 dgrad   (A = grad rows, B = pre-transposed weight)
//   gemm_tn:  C[N,K] += A[M,N]^T x B[M,K]  (f32 accumulation via split-M
//             atomics) -> conv1x1 wgrad
//   transpose_2d: bf16 [N,K] -> [K,N] (per-step weight transpose for dgrad)
#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) short;  // MFMA A/B frag
using f32x4 = __attribute__((ext_vector_type(4))) float;   // MFMA C/D frag

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int GEMM_TPB = 256;  // 4 waves, 2x2 wave grid, 64x64 per wave

__device__ __forceinline__ short bf16_bits(bf16 v) {
  short s;
  __builtin_memcpy(&s, &v, 2);
  return s;
}

// bijective XCD-aware swizzle (guide §5 "m204"): contiguous grid chunks map
// to one XCD so neighboring tiles share that XCD's L2
__device__ __forceinline__ int xcd_swizzle(int bid, int nwg) {
  constexpr int NXCD = 8;
  if (nwg < NXCD) return bid;
  int xcd = bid % NXCD, idx = bid / NXCD;
  int q = nwg / NXCD, r = nwg % NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

// stage a BMxBK bf16 tile (rows `row0..row0+127` of a [Rows x ld] matrix,
// cols k0..k0+31) into linear LDS [128][32] via global_load_lds.
// 128*32*2B = 8 KiB = 512 lanes x 16 B -> 2 rounds of 256 threads.
// Out-of-range rows are clamped (values unused, must just be finite).
__device__ __forceinline__ void stage_tile_128x32(
    const bf16* __restrict__ g, long ld, long row0, long rows, long k0,
    bf16* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    int unit = rnd * GEMM_TPB + t;       // 16B unit index 0..511
    long row = row0 + (unit >> 2);       // 4 units per 64B row
    if (row >= rows) row = rows - 1;     // clamp (finite garbage)
    int koff = (unit & 3) * 8;           // 8 bf16 per 16B
    const bf16* src = g + row * ld + k0 + koff;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(lds + unit * 8), 16,
        0, 0);
  }
}

// same, with explicit per-unit source rows (strided 1x1-conv gather)
__device__ __forceinline__ void stage_tile_rows(
    const bf16* __restrict__ g, long ld, const long* arow, long k0,
    bf16* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    int unit = rnd * GEMM_TPB + t;
    int koff = (unit & 3) * 8;
    const bf16* src = g + arow[rnd] * ld + k0 + koff;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(lds + unit * 8), 16,
        0, 0);
  }
}

// Double-buffered K-loop variants: the single-buffer loop serializes DMA
// and MFMA within a block and leans on the 2-blocks/CU interleave for
// overlap.  Measured (b512 shape sweep + e2e): +1-9% on the streaming
// 1x1 shapes, -15% on the deep-K small-M shapes (K=2048: 64 k-steps of
// compute-heavy reuse where the doubled LDS hurts more than overlap
// helps) -> ON by default for K <= GEMM_DB_KMAX, AMDTRAIN_GEMM_DB=0/1
// forces either path.
constexpr long GEMM_DB_KMAX = 1024;
// AMDTRAIN_GEMM_ADIR=1: opt-in direct-from-global A fragments (see kernel)
inline bool gemm_adir_enabled() {
  static const bool v = []() {
    const char* e = std::getenv("AMDTRAIN_GEMM_ADIR");
    return e && e[0] == '1';
  }();
  return v;
}
inline bool gemm_db_enabled(long K) {
  static const int v = []() {
    const char* e = std::getenv("AMDTRAIN_GEMM_DB");
    return e ? (e[0] == '1' ? 1 : 0) : -1;
  }();
  if (v >= 0) return v == 1;
  return K <= GEMM_DB_KMAX;
}

// compact output row m=(n,ho,wo) -> strided input row (n, ho*s, wo*s)
struct StrideMap {
  int H, W, Hout, Wout, stride;  // input spatial dims + output dims
};

__device__ __forceinline__ long stride_row(long m, const StrideMap& sm) {
  long t = m;
  const int wo = (int)(t % sm.Wout); t /= sm.Wout;
  const int ho = (int)(t % sm.Hout); t /= sm.Hout;
  return (t * sm.H + (long)ho * sm.stride) * sm.W + (long)wo * sm.stride;
}

// C/D fragment mapping for mfma_f32_16x16x32_bf16 (guide §3, m89-verified):
//   col = lane & 15, row = (lane >> 4) * 4 + reg
// A fragment: lane holds A[row = lane&15][k = (lane>>4)*8 + j]
// B fragment: lane holds B[k = (lane>>4)*8 + j][col = lane&15]
//   (from LDS B^T tile [n][k] this is the same contiguous 16B read as A)

// per-block column (sum, sumsq) of the fp32 accumulator tile -> one row of
// stats[bm][2N]: feeds BN forward without re-reading the conv output.
// acc layout: lane holds col = base+fr, rows (i*16 + fq*4 + r).
template <int NFRAG>
__device__ __forceinline__ void epilogue_stats(
    float* __restrict__ stats, const float* acc_flat, long m0, long M,
    long n0, long N, int bm, int wm, int wn, int fr, int fq, int nbm,
    bf16* lds_scratch) {
  // acc_flat: [4 m-frags][NFRAG n-frags][4 rows] per lane
  float* srow = stats + (long)bm * 2 * N;
  float s1[NFRAG], s2[NFRAG];
#pragma unroll
  for (int j = 0; j < NFRAG; ++j) {
    s1[j] = 0.f;
    s2[j] = 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wm + i * 16 + fq * 4 + r;
        float v = (row < M) ? acc_flat[(i * NFRAG + j) * 4 + r] : 0.f;
        s1[j] += v;
        s2[j] += v * v;
      }
  }
  // reduce over the 4 fq lanes of each column (lanes 16 apart)
#pragma unroll
  for (int j = 0; j < NFRAG; ++j) {
#pragma unroll
    for (int off = 32; off >= 16; off >>= 1) {
      s1[j] += __shfl_down(s1[j], off, AMD_WAVE);
      s2[j] += __shfl_down(s2[j], off, AMD_WAVE);
    }
  }
  // cross-wave: the two m-waves (wm 0/64) cover the same columns; stage in
  // LDS (reusing the staging buffer, fp32 [2 wavesM][N tile=128][2])
  float* red = (float*)lds_scratch;
  __syncthreads();  // tile LDS no longer needed for staging
  if (fq == 0) {
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) {
      int col = wn + j * 16 + fr;
      red[(wm ? 1 : 0) * 256 + col * 2 + 0] = s1[j];
      red[(wm ? 1 : 0) * 256 + col * 2 + 1] = s2[j];
    }
  }
  __syncthreads();
  const int t = threadIdx.x;
  for (int col = t; col < 128; col += GEMM_TPB) {
    long n = n0 + col;
    if (n < N) {
      srow[n] = red[col * 2] + red[256 + col * 2];
      srow[N + n] = red[col * 2 + 1] + red[256 + col * 2 + 1];
    }
  }
}

// NT: nontemporal C stores (global_store_* nt) — A/B experiment for the
// ~4.2 TB/s streaming wall on the BW-bound shapes (AMDTRAIN_GEMM_NT=1)
// NT=1: nontemporal C stores — measured 2x SLOWER (breaks write combining
// of the 2 B scattered stores); kept env-gated as a documented negative.
// LDSE=1: stage the C tile through LDS and emit coalesced 16 B stores
// instead of per-element column-strided 2 B stores.
// DB=1: double-buffered K-loop — stage chunk kt+1 while the MFMAs consume
// chunk kt (one barrier per chunk instead of two, per-block load/compute
// overlap instead of relying on the 2-blocks/CU interleave drifting into
// opposite phase).  The barrier's implicit vmcnt(0) wait is what drains
// the global_load_lds queue, exactly as in the single-buffer path.
// ADIR=1: A fragments are read DIRECTLY from global memory into VGPRs —
// the mfma A fragment is 16 contiguous bytes of one A row (lane l reads
// A[row=l&15][k=(l>>4)*8..+7]), so no LDS transit is needed at all: the
// A-side DMA and half the LDS footprint disappear; the second wave on
// the same rows hits L1.  A/B experiment for the streaming shapes
// (AMDTRAIN_GEMM_ADIR).  Plain (non-strided, non-LDSE) path only.
template <bool F32OUT, bool STRIDED, bool NT = false, bool LDSE = false,
          bool DB = false, bool ADIR = false>
__global__ void __launch_bounds__(GEMM_TPB, 2)
gemm_bt_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
               void* __restrict__ C, long M, long N, long K, int nbm,
               int nbn, StrideMap sm, float* __restrict__ stats) {
  __shared__ bf16 SMEM[(DB ? 2 : 1) *
                       ((ADIR ? 0 : BM) + BN) * BK];  // [As |] Bs
  bf16* const As = SMEM;    // (contiguous: the LDSE epilogue reuses 16 KB)
  bf16* const Bs = SMEM + (ADIR ? 0 : BM * BK);

  const int nwg = nbm * nbn;
  const int bid = xcd_swizzle(blockIdx.x, nwg);
  const int bm = bid / nbn, bn = bid % nbn;
  const long m0 = (long)bm * BM, n0 = (long)bn * BN;

  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE;      // 0..3 -> 2x2 wave grid
  const int lane = t % AMD_WAVE;
  const int wm = (wave >> 1) * 64;    // wave row offset in tile
  const int wn = (wave & 1) * 64;
  const int fr = lane & 15;           // fragment row/col within 16
  const int fq = lane >> 4;           // quad index 0..3

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  long arow[2];
  if (STRIDED) {  // per-unit gather rows are K-invariant: hoist
#pragma unroll
    for (int rnd = 0; rnd < 2; ++rnd) {
      long m = m0 + (((rnd * GEMM_TPB) + t) >> 2);
      if (m >= M) m = M - 1;
      arow[rnd] = stride_row(m, sm);
    }
  }

  // ADIR: per-lane clamped global row pointers for the 4 A fragments
  const bf16* Ap[4];
  if (ADIR) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      long row = m0 + wm + i * 16 + fr;
      if (row >= M) row = M - 1;  // finite garbage, epilogue masks it
      Ap[i] = A + row * K + fq * 8;
    }
  }
  const long ksteps = K / BK;
  auto stage = [&](long kt, bf16* as) {
    if (!ADIR) {
      if (STRIDED)
        stage_tile_rows(A, K, arow, kt * BK, as);
      else
        stage_tile_128x32(A, K, m0, M, kt * BK, as);
    }
    stage_tile_128x32(B, K, n0, N, kt * BK,
                      as + (ADIR ? 0 : BM * BK));
  };
  auto compute = [&](const bf16* as, long kt) {
    const bf16* bs = as + (ADIR ? 0 : BM * BK);
    bf16x8 a[4], b[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = ADIR ? *(const bf16x8*)(Ap[i] + kt * BK)
                  : *(const bf16x8*)&as[(wm + i * 16 + fr) * BK + fq * 8];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b[j] = *(const bf16x8*)&bs[(wn + j * 16 + fr) * BK + fq * 8];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[i], b[j], acc[i][j], 0, 0, 0);
  };
  if (DB) {
    constexpr int HB = ((ADIR ? 0 : BM) + BN) * BK;
    if (ksteps > 0) stage(0, SMEM);
    for (long kt = 0; kt < ksteps; ++kt) {
      bf16* const as = SMEM + (kt & 1) * HB;
      __syncthreads();  // drains this chunk's DMA queue (implicit
                        // vmcnt(0)); all waves are past reading the
                        // other buffer, which kt+1 overwrites below
      if (kt + 1 < ksteps) stage(kt + 1, SMEM + ((kt + 1) & 1) * HB);
      compute(as, kt);
    }
  } else {
    for (long kt = 0; kt < ksteps; ++kt) {
      __syncthreads();  // previous compute done before overwriting LDS
      stage(kt, SMEM);
      __syncthreads();  // barrier drains the global_load_lds queue
      compute(SMEM, kt);
    }
  }

  // epilogue: C[m0+wm+i*16+fq*4+r][n0+wn+j*16+fr]
  if (LDSE && !F32OUT && n0 + 128 <= N) {
    // stage the bf16 C tile through LDS (reusing the 16 KB staging buffer,
    // 64 rows per round) and emit coalesced 16 B row-major stores instead
    // of the fragment-shaped column-strided 2 B stores
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      __syncthreads();
      if (wm == h * 64) {
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
#pragma unroll
            for (int r = 0; r < 4; ++r)
              SMEM[(i * 16 + fq * 4 + r) * 128 + wn + j * 16 + fr] =
                  __float2bfloat16(acc[i][j][r]);
      }
      __syncthreads();
      for (int u2 = t; u2 < 1024; u2 += GEMM_TPB) {
        const int row_l = u2 >> 3;
        const long row = m0 + h * 64 + row_l;
        if (row < M)
          *(uint4*)((bf16*)C + row * N + n0 + (u2 & 7) * 16) =
              ((const uint4*)SMEM)[u2];
      }
    }
  } else {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          long row = m0 + wm + i * 16 + fq * 4 + r;
          long col = n0 + wn + j * 16 + fr;
          if (row < M && col < N) {
            if (F32OUT) {
              ((float*)C)[row * N + col] = acc[i][j][r];
            } else if (NT) {
              bf16 v = __float2bfloat16(acc[i][j][r]);
              short b;
              __builtin_memcpy(&b, &v, 2);
              __builtin_nontemporal_store(b, (short*)C + row * N + col);
            } else {
              ((bf16*)C)[row * N + col] = __float2bfloat16(acc[i][j][r]);
            }
          }
        }
      }
    }
  }
  if (stats != nullptr)
    epilogue_stats<4>(stats, (const float*)acc, m0, M, n0, N, bm, wm, wn,
                      fr, fq, nbm, As);
}

// ---- dual-N-tile BT GEMM (A-panel reuse) ---------------------------------
// The wide-N, small-K conv shapes (N>=256, K<=256: the bottleneck expand
// convs) sit at ~4.2 TB/s of min-traffic (tools/bench_conv1x1.py).  This
// variant computes TWO n-tiles per block from ONE staged A tile, halving
// nominal A traffic.  MEASURED NEUTRAL on those shapes (the ceiling is a
// uniform streaming-efficiency wall, not the A re-reads — the rr-floor
// coincidence in the first analysis was misleading); kept because it also
// halves launch width and is covered by tests.
template <bool DB = false>
__global__ void __launch_bounds__(GEMM_TPB, 2)
gemm_bt_n2_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                  bf16* __restrict__ C, long M, long N, long K, int nbm,
                  int nbn2, float* __restrict__ stats, int nbm_stats) {
  __shared__ bf16 SMEM[(DB ? 2 : 1) * (BM + 256) * BK];
  bf16* const As = SMEM;
  bf16* const Bs = SMEM + BM * BK;

  const int bid = xcd_swizzle(blockIdx.x, nbm * nbn2);
  const int bm = bid / nbn2, bn = bid % nbn2;
  const long m0 = (long)bm * BM, n0 = (long)bn * 256;

  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE;
  const int lane = t % AMD_WAVE;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int fr = lane & 15;
  const int fq = lane >> 4;

  f32x4 acc[4][8];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const long ksteps = K / BK;
  auto stage = [&](long kt, bf16* as) {
    bf16* bs = as + BM * BK;
    stage_tile_128x32(A, K, m0, M, kt * BK, as);
    stage_tile_128x32(B, K, n0, N, kt * BK, bs);
    stage_tile_128x32(B, K, n0 + 128, N, kt * BK, bs + 128 * BK);
  };
  auto compute = [&](const bf16* as) {
    const bf16* bs = as + BM * BK;
    bf16x8 a[4], b[8];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = *(const bf16x8*)&as[(wm + i * 16 + fr) * BK + fq * 8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int jn = (j < 4) ? wn + j * 16 : 128 + wn + (j - 4) * 16;
      b[j] = *(const bf16x8*)&bs[(jn + fr) * BK + fq * 8];
    }
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[i], b[j], acc[i][j], 0, 0, 0);
  };
  if (DB) {
    constexpr int HB = (BM + 256) * BK;
    if (ksteps > 0) stage(0, SMEM);
    for (long kt = 0; kt < ksteps; ++kt) {
      bf16* const as = SMEM + (kt & 1) * HB;
      __syncthreads();  // implicit vmcnt(0) drains this chunk's DMA
      if (kt + 1 < ksteps) stage(kt + 1, SMEM + ((kt + 1) & 1) * HB);
      compute(as);
    }
  } else {
    for (long kt = 0; kt < ksteps; ++kt) {
      __syncthreads();
      stage(kt, SMEM);
      __syncthreads();
      compute(SMEM);
    }
  }

#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wm + i * 16 + fq * 4 + r;
        long col = n0 + ((j < 4) ? wn + j * 16 : 128 + wn + (j - 4) * 16)
                   + fr;
        if (row < M && col < N)
          C[row * N + col] = __float2bfloat16(acc[i][j][r]);
      }
  if (stats != nullptr) {
    // per-block column (sum, sumsq) partials for the fused BN pipeline —
    // two 128-col halves through the shared epilogue helper
    float accl[4][4][4], acch[4][4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          accl[i][j][r] = acc[i][j][r];
          acch[i][j][r] = acc[i][j + 4][r];
        }
    epilogue_stats<4>(stats, (const float*)accl, m0, M, n0, N, bm, wm, wn,
                      fr, fq, nbm_stats, As);
    __syncthreads();
    epilogue_stats<4>(stats, (const float*)acch, m0, M, n0 + 128, N, bm,
                      wm, wn, fr, fq, nbm_stats, As);
  }
}

// ---- split-K BT GEMM for tiny-M shapes (the FC classifier) ---------------
// At [B,2048]x[2048,1000] the plain kernel has only ceil(B/128)*8 blocks —
// a fraction of the 256-CU chip — while K=2048 runs 64 serial K-steps.
// Split the K range across ksplit block-groups into fp32 partials, then
// collapse (+bias/addend, cast) in one small pass.
__global__ void __launch_bounds__(GEMM_TPB, 2)
gemm_bt_ks_kernel(const bf16* __restrict__ A, const bf16* __restrict__ B,
                  float* __restrict__ parts, long M, long N, long K, int nbm,
                  int nbn, int ksplit) {
  __shared__ bf16 As[BM * BK];
  __shared__ bf16 Bs[BN * BK];

  const int tiles = nbm * nbn;
  const int bid = xcd_swizzle(blockIdx.x, tiles * ksplit);
  const int tile = bid % tiles;
  const int kp = bid / tiles;
  const int bm = tile / nbn, bn = tile % nbn;
  const long m0 = (long)bm * BM, n0 = (long)bn * BN;

  const long ksteps = K / BK;
  const long kper = (ksteps + ksplit - 1) / ksplit;
  const long kt0 = kp * kper;
  const long kt1 = min(kt0 + kper, ksteps);

  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE;
  const int lane = t % AMD_WAVE;
  const int wm = (wave >> 1) * 64;
  const int wn = (wave & 1) * 64;
  const int fr = lane & 15;
  const int fq = lane >> 4;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (long kt = kt0; kt < kt1; ++kt) {
    __syncthreads();
    stage_tile_128x32(A, K, m0, M, kt * BK, As);
    stage_tile_128x32(B, K, n0, N, kt * BK, Bs);
    __syncthreads();
    bf16x8 a[4], b[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = *(const bf16x8*)&As[(wm + i * 16 + fr) * BK + fq * 8];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b[j] = *(const bf16x8*)&Bs[(wn + j * 16 + fr) * BK + fq * 8];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[i], b[j], acc[i][j], 0, 0, 0);
  }

  float* out = parts + (long)kp * M * N;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wm + i * 16 + fq * 4 + r;
        long col = n0 + wn + j * 16 + fr;
        if (row < M && col < N) out[row * N + col] = acc[i][j][r];
      }
}

__global__ void __launch_bounds__(AMD_TPB)
gemm_ks_collapse_kernel(const float* __restrict__ parts,
                        bf16* __restrict__ out, int ksplit, long MN, long N,
                        const float* __restrict__ bias,
                        const bf16* __restrict__ gadd) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < MN;
       i += (long)gridDim.x * blockDim.x) {
    float v = parts[i];
    for (int p = 1; p < ksplit; ++p) v += parts[(long)p * MN + i];
    if (bias != nullptr) v += bias[i % N];
    if (gadd != nullptr) v += __bfloat162float(gadd[i]);
    out[i] = __float2bfloat16(v);
  }
}

// ---- TN GEMM for wgrad: dW[N,K] += sum_m dY[m,n] * X[m,k] ----------------
// Output tiles 128n x 128k, M chunked by 32 and split across blocks
// (split-M) with fp32 atomic accumulation into dW.
//
// The MFMA fragments need the m-major (transposed) view of both chunks, so
// staging is global->reg->TRANSPOSED LDS write ([col][32 m] layout with an
// 8-element-block XOR swizzle to spread write banks); fragment loads are
// then contiguous ds_read_b128.  Out-of-range rows/cols stage zeros, which
// also removes per-fragment masking.
__device__ __forceinline__ int tn_swz(int c, int m) {
  // XOR c-bits 1-2 as well: spreads the stride-1 fragment reads over
  // 8 banks (2-way, free) while keeping the stride-8 writes at 4-way
  return c * 32 + (m ^ (((((c >> 3) & 3) ^ ((c >> 1) & 3))) << 3));
}

// stage a [32 m][128 col] chunk of a [M x ld] matrix, transposed, into LDS
__device__ __forceinline__ void tn_stage(const bf16* __restrict__ g, long ld,
                                         long m0, long M, long col0,
                                         bf16* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    const int unit = rnd * GEMM_TPB + t;  // 16 units (of 8 cols) per m-row
    const long m = m0 + (unit >> 4);
    const int c0 = (unit & 15) * 8;
    bf16 vals[8];
    if (m < M && col0 + c0 + 8 <= ld) {
      uint4 raw = *(const uint4*)(g + m * ld + col0 + c0);
      __builtin_memcpy(vals, &raw, 16);
    } else if (m < M && col0 + c0 < ld) {  // ragged col tail
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vals[j] = (col0 + c0 + j < ld) ? g[m * ld + col0 + c0 + j]
                                       : bf16(0.f);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] = bf16(0.f);
    }
    const int mloc = unit >> 4;
#pragma unroll
    for (int j = 0; j < 8; ++j) lds[tn_swz(c0 + j, mloc)] = vals[j];
  }
}

// tn_stage over strided-gathered rows (row = stride_row(m))
__device__ __forceinline__ void tn_stage_strided(
    const bf16* __restrict__ g, long ld, long m0, long M, long col0,
    const StrideMap& sm, bf16* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    const int unit = rnd * GEMM_TPB + t;
    const long m = m0 + (unit >> 4);
    const int c0 = (unit & 15) * 8;
    bf16 vals[8];
    if (m < M && col0 + c0 + 8 <= ld) {
      const long row = stride_row(m, sm);
      uint4 raw = *(const uint4*)(g + row * ld + col0 + c0);
      __builtin_memcpy(vals, &raw, 16);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[j] = bf16(0.f);
    }
    const int mloc = unit >> 4;
#pragma unroll
    for (int j = 0; j < 8; ++j) lds[tn_swz(c0 + j, mloc)] = vals[j];
  }
}

template <bool STRIDED>
__global__ void __launch_bounds__(GEMM_TPB, 2)
gemm_tn_kernel(const bf16* __restrict__ dY, const bf16* __restrict__ X,
               float* __restrict__ dW, long M, long N, long K, int nbn,
               int nbk, int msplit, StrideMap sm) {
  __shared__ bf16 Ys[128 * 32];
  __shared__ bf16 Xs[128 * 32];

  const int tiles = nbn * nbk;
  const int tile = blockIdx.x % tiles;
  const int mpart = blockIdx.x / tiles;
  const int bn = tile / nbk, bk = tile % nbk;
  const long n0 = (long)bn * BM, k0 = (long)bk * BN;

  const long mchunks = (M + 31) / 32;
  const long chunks_per_part = (mchunks + msplit - 1) / msplit;
  const long mc0 = (long)mpart * chunks_per_part;
  const long mc1 = min(mc0 + chunks_per_part, mchunks);

  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE;
  const int lane = t % AMD_WAVE;
  const int wn = (wave >> 1) * 64;
  const int wk = (wave & 1) * 64;
  const int fr = lane & 15;
  const int fq = lane >> 4;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (long mc = mc0; mc < mc1; ++mc) {
    const long m0 = mc * 32;
    __syncthreads();
    tn_stage(dY, N, m0, M, n0, Ys);
    if (STRIDED)
      tn_stage_strided(X, K, m0, M, k0, sm, Xs);
    else
      tn_stage(X, K, m0, M, k0, Xs);
    __syncthreads();

    // A fragment: dY^T[n][m]; B fragment: X^T -> both contiguous b128 reads
    bf16x8 a[4], b[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = *(const bf16x8*)&Ys[tn_swz(wn + i * 16 + fr, fq * 8)];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b[j] = *(const bf16x8*)&Xs[tn_swz(wk + j * 16 + fr, fq * 8)];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[i], b[j], acc[i][j], 0, 0, 0);
  }

  // atomic split-M accumulation (measured cheaper than a partial-buffer
  // second pass, whose traffic scales with block count)
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long n = n0 + wn + i * 16 + fq * 4 + r;
        long k = k0 + wk + j * 16 + fr;
        if (n < N && k < K) atomicAdd(&dW[n * K + k], acc[i][j][r]);
      }
}

// ---- small bf16 2D transpose (per-step weight transpose for dgrad) -------
__global__ void __launch_bounds__(AMD_TPB)
transpose_2d_kernel(const bf16* __restrict__ in, bf16* __restrict__ out,
                    int R, int Ccols) {
  __shared__ bf16 tile[32][33];
  const int bx = blockIdx.x % ((Ccols + 31) / 32);
  const int by = blockIdx.x / ((Ccols + 31) / 32);
  const int c0 = bx * 32, r0 = by * 32;
  const int tx = threadIdx.x % 32, ty = threadIdx.x / 32;  // 32x8
  for (int dy = 0; dy < 32; dy += 8) {
    int r = r0 + ty + dy, c = c0 + tx;
    if (r < R && c < Ccols) tile[ty + dy][tx] = in[(long)r * Ccols + c];
  }
  __syncthreads();
  for (int dy = 0; dy < 32; dy += 8) {
    int c = c0 + ty + dy, r = r0 + tx;  // transposed coords
    if (r < R && c < Ccols) out[(long)c * R + r] = tile[tx][ty + dy];
  }
}

}  // namespace

at::Tensor gemm_bt(at::Tensor A, at::Tensor B, bool f32_out,
                   std::optional<at::Tensor> addend,
                   std::optional<at::Tensor> bias) {
  TORCH_CHECK(A.is_cuda() && A.dim() == 2 && B.dim() == 2);
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16,
              "gemm_bt expects bf16 inputs");
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  long M = Ac.size(0), K = Ac.size(1), N = Bc.size(0);
  TORCH_CHECK(Bc.size(1) == K, "K mismatch");
  TORCH_CHECK(K % BK == 0, "K must be a multiple of ", BK);
  auto C = at::empty({M, N},
                     Ac.options().dtype(f32_out ? at::kFloat : at::kBFloat16));
  int nbm = (int)((M + BM - 1) / BM), nbn = (int)((N + BN - 1) / BN);
  auto stream = at::cuda::getCurrentCUDAStream();
  StrideMap sm{0, 0, 0, 0, 1};
  const bf16* gadd = nullptr;
  at::Tensor ac, bc2;
  if (addend) {
    ac = addend->contiguous();
    TORCH_CHECK(ac.scalar_type() == at::kBFloat16 &&
                ac.numel() == M * N, "addend must be bf16 [M,N]");
    gadd = (const bf16*)ac.const_data_ptr();
  }
  const float* biasp = nullptr;
  if (bias) {
    bc2 = bias->to(at::kFloat).contiguous();
    TORCH_CHECK(bc2.numel() == N, "bias must be [N]");
    biasp = (const float*)bc2.const_data_ptr();
  }
  const long tiles = (long)nbm * nbn;
  const long ksteps = K / BK;
  if (!f32_out && !addend && !bias && N % 256 == 0 && K <= 256 &&
      (long)nbm * (nbn / 2) >= 256) {
    // wide-N small-K: dual-n-tile variant (see kernel comment; measured
    // perf-neutral, halves nominal A traffic and launch width)
    int nbn2 = nbn / 2;
    if (gemm_db_enabled(K))
      gemm_bt_n2_kernel<true><<<nbm * nbn2, GEMM_TPB, 0, stream>>>(
          (const bf16*)Ac.const_data_ptr(),
          (const bf16*)Bc.const_data_ptr(), (bf16*)C.data_ptr(), M, N, K,
          nbm, nbn2, nullptr, nbm);
    else
      gemm_bt_n2_kernel<<<nbm * nbn2, GEMM_TPB, 0, stream>>>(
          (const bf16*)Ac.const_data_ptr(),
          (const bf16*)Bc.const_data_ptr(), (bf16*)C.data_ptr(), M, N, K,
          nbm, nbn2, nullptr, nbm);
    CHECK_CUDA_OK();
    return C;
  }
  if (!f32_out && tiles <= 128 && ksteps >= 2 && 512 / tiles >= 2) {
    // tiny-M path (FC classifier fwd/dgrad): split K to fill the chip
    const int ksplit = (int)std::min<long>(ksteps, 512 / tiles);
    auto parts = at::empty({ksplit, M * N},
                           Ac.options().dtype(at::kFloat));
    gemm_bt_ks_kernel<<<(int)(tiles * ksplit), GEMM_TPB, 0, stream>>>(
        (const bf16*)Ac.const_data_ptr(), (const bf16*)Bc.const_data_ptr(),
        parts.data_ptr<float>(), M, N, K, nbm, nbn, ksplit);
    CHECK_CUDA_OK();
    gemm_ks_collapse_kernel<<<amd_grid(M * N), AMD_TPB, 0, stream>>>(
        (const float*)parts.const_data_ptr(), (bf16*)C.data_ptr(), ksplit,
        M * N, N, biasp, gadd);
    CHECK_CUDA_OK();
    return C;
  }
  static const bool use_nt = []() {
    const char* v = std::getenv("AMDTRAIN_GEMM_NT");
    return v && v[0] == '1';
  }();
  const bool use_db = gemm_db_enabled(K);
  static const bool use_ldse = []() {
    const char* v = std::getenv("AMDTRAIN_GEMM_LDSE");
    return v && v[0] == '1';
  }();
  if (f32_out)
    gemm_bt_kernel<true, false><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)Ac.const_data_ptr(), (const bf16*)Bc.const_data_ptr(),
        C.data_ptr(), M, N, K, nbm, nbn, sm, nullptr);
  else if (use_ldse && N % 8 == 0)
    gemm_bt_kernel<false, false, false, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)Ac.const_data_ptr(),
            (const bf16*)Bc.const_data_ptr(), C.data_ptr(), M, N, K, nbm,
            nbn, sm, nullptr);
  else if (use_nt)
    gemm_bt_kernel<false, false, true><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)Ac.const_data_ptr(), (const bf16*)Bc.const_data_ptr(),
        C.data_ptr(), M, N, K, nbm, nbn, sm, nullptr);
  else if (gemm_adir_enabled())
    gemm_bt_kernel<false, false, false, false, true, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)Ac.const_data_ptr(),
            (const bf16*)Bc.const_data_ptr(), C.data_ptr(), M, N, K, nbm,
            nbn, sm, nullptr);
  else if (use_db)
    gemm_bt_kernel<false, false, false, false, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)Ac.const_data_ptr(),
            (const bf16*)Bc.const_data_ptr(), C.data_ptr(), M, N, K, nbm,
            nbn, sm, nullptr);
  else
    gemm_bt_kernel<false, false><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)Ac.const_data_ptr(), (const bf16*)Bc.const_data_ptr(),
        C.data_ptr(), M, N, K, nbm, nbn, sm, nullptr);
  CHECK_CUDA_OK();
  // rare big-M path with bias/addend (the FC shapes take the split-K
  // collapse above; keep the hot conv epilogue branch-free)
  if (biasp) C.add_(*bias);
  if (gadd) C.add_(ac);
  return C;
}

// 1x1 conv forward + fused per-block BN statistics partials [nbm][2N]
std::vector<at::Tensor> gemm_bt_stats(at::Tensor A, at::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16);
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  long M = Ac.size(0), K = Ac.size(1), N = Bc.size(0);
  TORCH_CHECK(Bc.size(1) == K && K % BK == 0);
  auto C = at::empty({M, N}, Ac.options());
  int nbm = (int)((M + BM - 1) / BM), nbn = (int)((N + BN - 1) / BN);
  auto stats = at::empty({nbm, 2 * N}, Ac.options().dtype(at::kFloat));
  StrideMap sm{0, 0, 0, 0, 1};
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool use_db = gemm_db_enabled(K);
  if (N % 256 == 0 && K <= 256 && (long)nbm * (nbn / 2) >= 256) {
    int nbn2 = nbn / 2;
    if (use_db)
      gemm_bt_n2_kernel<true><<<nbm * nbn2, GEMM_TPB, 0, stream>>>(
          (const bf16*)Ac.const_data_ptr(),
          (const bf16*)Bc.const_data_ptr(), (bf16*)C.data_ptr(), M, N, K,
          nbm, nbn2, stats.data_ptr<float>(), nbm);
    else
      gemm_bt_n2_kernel<<<nbm * nbn2, GEMM_TPB, 0, stream>>>(
          (const bf16*)Ac.const_data_ptr(),
          (const bf16*)Bc.const_data_ptr(), (bf16*)C.data_ptr(), M, N, K,
          nbm, nbn2, stats.data_ptr<float>(), nbm);
  } else if (gemm_adir_enabled()) {
    gemm_bt_kernel<false, false, false, false, true, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)Ac.const_data_ptr(),
            (const bf16*)Bc.const_data_ptr(), C.data_ptr(), M, N, K, nbm,
            nbn, sm, stats.data_ptr<float>());
  } else if (use_db) {
    gemm_bt_kernel<false, false, false, false, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)Ac.const_data_ptr(),
            (const bf16*)Bc.const_data_ptr(), C.data_ptr(), M, N, K, nbm,
            nbn, sm, stats.data_ptr<float>());
  } else {
    gemm_bt_kernel<false, false><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)Ac.const_data_ptr(), (const bf16*)Bc.const_data_ptr(),
        C.data_ptr(), M, N, K, nbm, nbn, sm, stats.data_ptr<float>());
  }
  CHECK_CUDA_OK();
  return {C, stats};
}

// strided 1x1-conv forward: y[m=(n,ho,wo), n'] = sum_k x[(n,ho*s,wo*s), k]
// * w[n', k] — the even-row gather happens inside the A staging (the
// separate .contiguous() gather copy this replaces cost ~0.5 ms/step)
at::Tensor gemm_bt_strided(at::Tensor A, at::Tensor B, long Nn, long H,
                           long W, long stride) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16);
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  long K = Ac.size(1), N = Bc.size(0);
  TORCH_CHECK(Bc.size(1) == K && K % BK == 0);
  long Hout = (H + stride - 1) / stride, Wout = (W + stride - 1) / stride;
  long M = Nn * Hout * Wout;
  auto C = at::empty({M, N}, Ac.options());
  int nbm = (int)((M + BM - 1) / BM), nbn = (int)((N + BN - 1) / BN);
  StrideMap sm{(int)H, (int)W, (int)Hout, (int)Wout, (int)stride};
  auto stream = at::cuda::getCurrentCUDAStream();
  if (gemm_db_enabled(K))
    gemm_bt_kernel<false, true, false, false, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)Ac.const_data_ptr(),
            (const bf16*)Bc.const_data_ptr(), C.data_ptr(), M, N, K, nbm,
            nbn, sm, nullptr);
  else
    gemm_bt_kernel<false, true><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)Ac.const_data_ptr(), (const bf16*)Bc.const_data_ptr(),
        C.data_ptr(), M, N, K, nbm, nbn, sm, nullptr);
  CHECK_CUDA_OK();
  return C;
}

at::Tensor gemm_tn(at::Tensor dY, at::Tensor X, long msplit) {
  TORCH_CHECK(dY.is_cuda() && dY.dim() == 2 && X.dim() == 2);
  TORCH_CHECK(dY.scalar_type() == at::kBFloat16 &&
              X.scalar_type() == at::kBFloat16);
  auto Yc = dY.contiguous();
  auto Xc = X.contiguous();
  long M = Yc.size(0), N = Yc.size(1), K = Xc.size(1);
  TORCH_CHECK(Xc.size(0) == M, "M mismatch");
  TORCH_CHECK(N % 8 == 0 && K % 8 == 0, "N,K must be multiples of 8");
  auto dW = at::zeros({N, K}, Yc.options().dtype(at::kFloat));
  int nbn = (int)((N + 127) / 128), nbk = (int)((K + 127) / 128);
  long tiles = (long)nbn * nbk;
  const bool det = std::getenv("AMDTRAIN_DETERMINISTIC") != nullptr;
  if (det)
    msplit = 1;  // single-accumulator: bitwise-deterministic wgrad
  else if (msplit <= 0)
    msplit = std::max<long>(1, std::min<long>((M + 31) / 32, 512 / tiles));
  auto stream = at::cuda::getCurrentCUDAStream();
  StrideMap sm{0, 0, 0, 0, 1};
  gemm_tn_kernel<false><<<(int)(tiles * msplit), GEMM_TPB, 0, stream>>>(
      (const bf16*)Yc.const_data_ptr(), (const bf16*)Xc.const_data_ptr(),
      dW.data_ptr<float>(), M, N, K, nbn, nbk, (int)msplit, sm);
  CHECK_CUDA_OK();
  return dW;
}

// wgrad of the strided 1x1 conv: X rows gathered in-kernel
at::Tensor gemm_tn_strided(at::Tensor dY, at::Tensor X, long Nn, long H,
                           long W, long stride) {
  auto Yc = dY.contiguous();
  auto Xc = X.contiguous();
  long M = Yc.size(0), N = Yc.size(1), K = Xc.size(1);
  TORCH_CHECK(N % 8 == 0 && K % 8 == 0);
  long Hout = (H + stride - 1) / stride, Wout = (W + stride - 1) / stride;
  TORCH_CHECK(M == Nn * Hout * Wout, "M mismatch for strided wgrad");
  auto dW = at::zeros({N, K}, Yc.options().dtype(at::kFloat));
  int nbn = (int)((N + 127) / 128), nbk = (int)((K + 127) / 128);
  long tiles = (long)nbn * nbk;
  const bool det2 = std::getenv("AMDTRAIN_DETERMINISTIC") != nullptr;
  long msplit = det2 ? 1 : std::max<long>(
      1, std::min<long>((M + 31) / 32, 512 / tiles));
  StrideMap sm{(int)H, (int)W, (int)Hout, (int)Wout, (int)stride};
  auto stream = at::cuda::getCurrentCUDAStream();
  gemm_tn_kernel<true><<<(int)(tiles * msplit), GEMM_TPB, 0, stream>>>(
      (const bf16*)Yc.const_data_ptr(), (const bf16*)Xc.const_data_ptr(),
      dW.data_ptr<float>(), M, N, K, nbn, nbk, (int)msplit, sm);
  CHECK_CUDA_OK();
  return dW;
}

at::Tensor transpose_2d(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 &&
              x.scalar_type() == at::kBFloat16);
  auto xc = x.contiguous();
  int R = (int)xc.size(0), C = (int)xc.size(1);
  auto out = at::empty({C, R}, xc.options());
  int gx = (C + 31) / 32, gy = (R + 31) / 32;
  auto stream = at::cuda::getCurrentCUDAStream();
  transpose_2d_kernel<<<gx * gy, AMD_TPB, 0, stream>>>(
      (const bf16*)xc.const_data_ptr(), (bf16*)out.data_ptr(), R, C);
  CHECK_CUDA_OK();
  return out;
}
