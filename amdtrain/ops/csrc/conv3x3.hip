// Implicit-GEMM 3x3 convolution (NHWC, pad 1, stride 1/2) for gfx950 —
// forward, dgrad and wgrad (SURVEY §2c "Conv2d 3x3 ... 16 call sites/fwd").
//
// GEMM view over rows m = (n, ho, wo):
//   fwd:   Y[m, cout] = sum_{tap, cin} X[gather(m, tap), cin] * W[cout, tap, cin]
//   dgrad: dX[m, cin] = sum_{tap, cout} dY[gather'(m, tap), cout] * W'[cin, tap, cout]
//          (W' is the 180-degree-rotated, [Cin, 9*Cout]-permuted weight)
//   wgrad: dW[cout, tap, cin] = sum_m dY[m, cout] * X[gather(m, tap), cin]
//          (one split-M TN GEMM per tap, atomic fp32 accumulation)
//
// Same block structure as gemm.hip (128x128 tile, BK=32 within one tap —
// all ResNet channel counts are multiples of 32 — 4 waves of 4x4
// mfma_f32_16x16x32_bf16 fragments, global_load_lds staging).  Padding is
// handled with a 16 B zero page: out-of-bounds gather rows load from it, so
// no LDS pre-zeroing pass and no boundary branches in the MFMA loop.
#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int GEMM_TPB = 256;
constexpr int BK = 32;

__device__ __forceinline__ int xcd_swz(int bid, int nwg) {
  constexpr int NXCD = 8;
  if (nwg < NXCD) return bid;
  int xcd = bid % NXCD, idx = bid / NXCD;
  int q = nwg / NXCD, r = nwg % NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

struct ConvGeom {
  int H, W, Hout, Wout, stride;  // input / output spatial dims
};

// forward gather: output row m=(n,ho,wo), tap (kh,kw) -> input row or -1
__device__ __forceinline__ long fwd_gather(long m, int kh, int kw,
                                           const ConvGeom& g) {
  long t = m;
  const int wo = (int)(t % g.Wout); t /= g.Wout;
  const int ho = (int)(t % g.Hout); t /= g.Hout;
  const int h = ho * g.stride - 1 + kh;
  const int w = wo * g.stride - 1 + kw;
  if (h < 0 || h >= g.H || w < 0 || w >= g.W) return -1;
  return (t * g.H + h) * g.W + w;
}

// dgrad gather: input row m=(n,h,w), tap (kh,kw) -> dY row or -1
// (H,W = input dims; Hout,Wout = dY dims)
__device__ __forceinline__ long dgrad_gather(long m, int kh, int kw,
                                             const ConvGeom& g) {
  long t = m;
  const int w = (int)(t % g.W); t /= g.W;
  const int h = (int)(t % g.H); t /= g.H;
  const int ho2 = h + 1 - kh, wo2 = w + 1 - kw;
  if (g.stride == 1) {
    if (ho2 < 0 || ho2 >= g.Hout || wo2 < 0 || wo2 >= g.Wout) return -1;
    return (t * g.Hout + ho2) * g.Wout + wo2;
  }
  if ((ho2 & 1) || (wo2 & 1)) return -1;
  const int ho = ho2 >> 1, wo = wo2 >> 1;
  if (ho < 0 || ho >= g.Hout || wo < 0 || wo >= g.Wout) return -1;
  return (t * g.Hout + ho) * g.Wout + wo;
}

// per-unit precomputed pixel coordinates: the output-row decode (div/mod
// chains) is tap-invariant, so it is hoisted out of the K-loop entirely
struct UnitCoord {
  long n_off;  // n * H * W (input-row base for this image)
  int hb, wb;  // tap-0 pixel coords (may be negative / out of range)
};

template <bool DGRAD>
__device__ __forceinline__ UnitCoord decode_unit(long m, const ConvGeom& g) {
  UnitCoord u;
  long t = m;
  if (DGRAD) {  // m ranges over INPUT pixels; gather reads dY [Hout,Wout]
    const int w = (int)(t % g.W); t /= g.W;
    const int h = (int)(t % g.H); t /= g.H;
    u.n_off = t * (long)g.Hout * g.Wout;
    u.hb = h + 1;  // minus kh per tap
    u.wb = w + 1;
  } else {      // m ranges over OUTPUT pixels; gather reads X [H,W]
    const int wo = (int)(t % g.Wout); t /= g.Wout;
    const int ho = (int)(t % g.Hout); t /= g.Hout;
    u.n_off = t * (long)g.H * g.W;
    u.hb = ho * g.stride - 1;  // plus kh per tap
    u.wb = wo * g.stride - 1;
  }
  return u;
}

template <bool DGRAD>
__device__ __forceinline__ long unit_row(const UnitCoord& u, int kh, int kw,
                                         const ConvGeom& g) {
  if (DGRAD) {
    int ho2 = u.hb - kh, wo2 = u.wb - kw;
    if (g.stride == 2) {
      if ((ho2 | wo2) & 1) return -1;
      ho2 >>= 1;
      wo2 >>= 1;
    }
    if (ho2 < 0 || ho2 >= g.Hout || wo2 < 0 || wo2 >= g.Wout) return -1;
    return u.n_off + (long)ho2 * g.Wout + wo2;
  }
  const int h = u.hb + kh, w = u.wb + kw;
  if (h < 0 || h >= g.H || w < 0 || w >= g.W) return -1;
  return u.n_off + (long)h * g.W + w;
}

// BKT = K-step depth (32 or 64): BKT/8 16-B units per m-row, BKT/16 rounds
template <bool DGRAD, int BKT>
__device__ __forceinline__ void stage_gathered(
    const bf16* __restrict__ src, int ld, const UnitCoord* uc, int c0, int kh,
    int kw, const ConvGeom& g, const bf16* __restrict__ zero_page,
    bf16* lds) {
  const int t = threadIdx.x;
  constexpr int UPR = BKT / 8;
#pragma unroll
  for (int rnd = 0; rnd < BKT / 16; ++rnd) {
    int unit = rnd * GEMM_TPB + t;
    long row = unit_row<DGRAD>(uc[rnd], kh, kw, g);
    const bf16* p = row < 0 ? zero_page
                            : src + row * (long)ld + c0 +
                                  (unit % UPR) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)p,
        (__attribute__((address_space(3))) unsigned int*)(lds + unit * 8), 16,
        0, 0);
  }
}

// stage a [128 rows][32 cols] tile of a plain [Rows x ld] matrix at column
// offset koff (weights)
// stage ROWS x BKT (ROWS=64 for the NT=64 tile: half the units)
template <int BKT, int ROWS>
__device__ __forceinline__ void stage_plain_rows(
    const bf16* __restrict__ gsrc, long ld, long row0, long rows, long koff,
    bf16* lds) {
  const int t = threadIdx.x;
  constexpr int UPR = BKT / 8;
  constexpr int UNITS = ROWS * UPR;
#pragma unroll
  for (int rnd = 0; rnd < UNITS / GEMM_TPB; ++rnd) {
    int unit = rnd * GEMM_TPB + t;
    long row = row0 + unit / UPR;
    if (row >= rows) row = rows - 1;
    const bf16* src = gsrc + row * ld + koff + (unit % UPR) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(lds + unit * 8), 16,
        0, 0);
  }
}

template <int BKT>
__device__ __forceinline__ void stage_plain(
    const bf16* __restrict__ gsrc, long ld, long row0, long rows, long koff,
    bf16* lds) {
  const int t = threadIdx.x;
  constexpr int UPR = BKT / 8;
#pragma unroll
  for (int rnd = 0; rnd < BKT / 16; ++rnd) {
    int unit = rnd * GEMM_TPB + t;
    long row = row0 + unit / UPR;
    if (row >= rows) row = rows - 1;
    const bf16* src = gsrc + row * ld + koff + (unit % UPR) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(lds + unit * 8), 16,
        0, 0);
  }
}

// per-block column (sum, sumsq) of the fp32 accumulator tile (see
// gemm.hip); NTE = n-tile width (128, or 64 for the banded variant)
template <int NTE = 128>
__device__ __forceinline__ void conv_epilogue_stats(
    float* __restrict__ stats, const float* acc_flat, long m0, long M,
    long n0, long N, int bm, int wm, int wn, int fr, int fq,
    bf16* lds_scratch) {
  constexpr int NF = NTE / 32;  // n fragments per wave
  float* srow = stats + (long)bm * 2 * N;
  float s1[NF], s2[NF];
#pragma unroll
  for (int j = 0; j < NF; ++j) {
    s1[j] = 0.f;
    s2[j] = 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wm + i * 16 + fq * 4 + r;
        float v = (row < M) ? acc_flat[(i * NF + j) * 4 + r] : 0.f;
        s1[j] += v;
        s2[j] += v * v;
      }
  }
#pragma unroll
  for (int j = 0; j < NF; ++j) {
#pragma unroll
    for (int off = 32; off >= 16; off >>= 1) {
      s1[j] += __shfl_down(s1[j], off, AMD_WAVE);
      s2[j] += __shfl_down(s2[j], off, AMD_WAVE);
    }
  }
  float* red = (float*)lds_scratch;
  __syncthreads();
  if (fq == 0) {
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      int col = wn + j * 16 + fr;
      red[(wm ? 1 : 0) * (2 * NTE) + col * 2 + 0] = s1[j];
      red[(wm ? 1 : 0) * (2 * NTE) + col * 2 + 1] = s2[j];
    }
  }
  __syncthreads();
  const int t = threadIdx.x;
  for (int col = t; col < NTE; col += GEMM_TPB) {
    long n = n0 + col;
    if (n < N) {
      srow[n] = red[col * 2] + red[2 * NTE + col * 2];
      srow[N + n] = red[col * 2 + 1] + red[2 * NTE + col * 2 + 1];
    }
  }
}

// fwd / dgrad main kernel.  DGRAD only changes the gather map; operand roles:
//   fwd:   A = x rows (AC channels), B = w [NC, 9*AC], C = y [M, NC]
//   dgrad: A = dy rows (AC = Cout), B = w' [NC = Cin, 9*Cout], C = dx
// BAND: block-diagonal weight (grouped conv via the dense path, Cin==NC,
// both % 128): only the K-steps whose NT-channel window matches this
// n-tile's group window are nonzero — skip the rest.
// NT: output-tile width (128 default; 64 halves the band window and thus
// the off-diagonal waste for grouped convs — wave grid becomes 2m x 2n
// over [128m x 64n], 4x2 fragments per wave).
// DB: double-buffered (tap, ks) pipeline — stage iteration it+1's A/B
// tiles while the MFMAs consume iteration it's (one barrier per iteration
// instead of two; the barrier's implicit vmcnt(0) drains the DMA queue),
// mirroring the gemm.hip DB loop.
template <bool DGRAD, int BKT = BK, bool BAND = false, int NT = 128,
          bool DB = false>
__global__ void __launch_bounds__(GEMM_TPB, 2)
conv3x3_kernel(const bf16* __restrict__ A, const bf16* __restrict__ Bw,
               bf16* __restrict__ C, long M, int AC, int NC, ConvGeom g,
               int nbm, int nbn, const bf16* __restrict__ zero_page,
               float* __restrict__ stats) {
  __shared__ bf16 SMEM[(DB ? 2 : 1) * 2 * 128 * BKT];
  bf16* const As = SMEM;
  bf16* const Bs = SMEM + 128 * BKT;

  const int bid = xcd_swz(blockIdx.x, nbm * nbn);
  const int bm = bid / nbn, bn = bid % nbn;
  const long m0 = (long)bm * 128, n0 = (long)bn * NT;
  constexpr int NFR = NT / 32;  // n fragments per wave (4 at NT=128)

  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE, lane = t % AMD_WAVE;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * (NT / 2);
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][NFR];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < NFR; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // hoist the per-unit coordinate decode out of the K-loop
  UnitCoord uc[BKT / 16];
#pragma unroll
  for (int rnd = 0; rnd < BKT / 16; ++rnd) {
    long m = m0 + (rnd * GEMM_TPB + t) / (BKT / 8);
    if (m >= M) m = M - 1;
    uc[rnd] = decode_unit<DGRAD>(m, g);
  }

  const int ks_lo = BAND ? (int)(n0 / BKT) : 0;
  const int ks_hi = BAND ? ks_lo + NT / BKT : AC / BKT;
  const int nks = ks_hi - ks_lo;
  const int total = 9 * nks;
  auto stage_it = [&](int it, bf16* as) {
    const int tap = it / nks, ks = ks_lo + it % nks;
    const int kh = tap / 3, kw = tap % 3, c0 = ks * BKT;
    stage_gathered<DGRAD, BKT>(A, AC, uc, c0, kh, kw, g, zero_page, as);
    if (NT == 128)
      stage_plain<BKT>(Bw, (long)9 * AC, n0, NC, (long)tap * AC + c0,
                       as + 128 * BKT);
    else  // 64-row B tile: half the staging rounds
      stage_plain_rows<BKT, 64>(Bw, (long)9 * AC, n0, NC,
                                (long)tap * AC + c0, as + 128 * BKT);
  };
  auto compute_it = [&](const bf16* as) {
    const bf16* bs = as + 128 * BKT;
#pragma unroll
    for (int kk = 0; kk < BKT / 32; ++kk) {
      bf16x8 a[4], b[NFR];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        a[i] = *(const bf16x8*)
            &as[(wm + i * 16 + fr) * BKT + kk * 32 + fq * 8];
#pragma unroll
      for (int j = 0; j < NFR; ++j)
        b[j] = *(const bf16x8*)
            &bs[(wn + j * 16 + fr) * BKT + kk * 32 + fq * 8];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NFR; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
  };
  if (DB) {
    constexpr int HB = 2 * 128 * BKT;
    if (total > 0) stage_it(0, SMEM);
    for (int it = 0; it < total; ++it) {
      bf16* const as = SMEM + (it & 1) * HB;
      __syncthreads();  // implicit vmcnt(0) drains this iter's DMA; all
                        // waves are past reading the other buffer
      if (it + 1 < total) stage_it(it + 1, SMEM + ((it + 1) & 1) * HB);
      compute_it(as);
    }
  } else {
    for (int it = 0; it < total; ++it) {
      __syncthreads();
      stage_it(it, SMEM);
      __syncthreads();
      compute_it(SMEM);
    }
  }

#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < NFR; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wm + i * 16 + fq * 4 + r;
        long col = n0 + wn + j * 16 + fr;
        if (row < M && col < NC)
          C[row * NC + col] = __float2bfloat16(acc[i][j][r]);
      }
  if (stats != nullptr)
    conv_epilogue_stats<NT>(stats, (const float*)acc, m0, M, n0, NC, bm,
                            wm, wn, fr, fq, As);
}

// wgrad per tap: dW[cout, tap*Cin + cin] += sum_m dY[m, cout] * Xg[m, cin]
// Transposed reg-staging + contiguous b128 fragment reads (see gemm.hip
// gemm_tn); gathered X rows that fall in the padding stage zeros.
__device__ __forceinline__ int wg_swz(int cc, int m) {
  return cc * 32 + (m ^ (((((cc >> 3) & 3) ^ ((cc >> 1) & 3))) << 3));
}

__global__ void __launch_bounds__(GEMM_TPB, 2)
conv3x3_wgrad_kernel(const bf16* __restrict__ dY, const bf16* __restrict__ X,
                     float* __restrict__ dW, long M, int Cout, int Cin,
                     ConvGeom g, int tap, int nbn, int nbk, int msplit) {
  __shared__ bf16 Ys[128 * 32];
  __shared__ bf16 Xs[128 * 32];

  const int kh = tap / 3, kw = tap % 3;
  const int tiles = nbn * nbk;
  const int tile = blockIdx.x % tiles;
  const int mpart = blockIdx.x / tiles;
  const int bn = tile / nbk, bk = tile % nbk;
  const long n0 = (long)bn * 128, k0 = (long)bk * 128;

  const long mchunks = (M + 31) / 32;
  const long cpp = (mchunks + msplit - 1) / msplit;
  const long mc0 = (long)mpart * cpp;
  const long mc1 = min(mc0 + cpp, mchunks);

  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE, lane = t % AMD_WAVE;
  const int wn = (wave >> 1) * 64, wk = (wave & 1) * 64;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (long mc = mc0; mc < mc1; ++mc) {
    const long m0 = mc * 32;
    __syncthreads();
#pragma unroll
    for (int rnd = 0; rnd < 2; ++rnd) {
      const int unit = rnd * GEMM_TPB + t;  // 16 units (8 cols) per m-row
      const long m = m0 + (unit >> 4);
      const int c0 = (unit & 15) * 8;
      const int mloc = unit >> 4;
      // dY chunk (cols may exceed Cout for Cout<128 -> zeros)
      bf16 yv[8];
      if (m < M && n0 + c0 + 8 <= Cout) {
        uint4 raw = *(const uint4*)(dY + m * Cout + n0 + c0);
        __builtin_memcpy(yv, &raw, 16);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) yv[j] = bf16(0.f);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) Ys[wg_swz(c0 + j, mloc)] = yv[j];
      // gathered X chunk (padding rows and col overflow -> zeros)
      long xrow = m < M ? fwd_gather(m, kh, kw, g) : -1;
      bf16 xv[8];
      if (xrow >= 0 && k0 + c0 + 8 <= Cin) {
        uint4 raw = *(const uint4*)(X + xrow * (long)Cin + k0 + c0);
        __builtin_memcpy(xv, &raw, 16);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) xv[j] = bf16(0.f);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) Xs[wg_swz(c0 + j, mloc)] = xv[j];
    }
    __syncthreads();

    bf16x8 a[4], b[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = *(const bf16x8*)&Ys[wg_swz(wn + i * 16 + fr, fq * 8)];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b[j] = *(const bf16x8*)&Xs[wg_swz(wk + j * 16 + fr, fq * 8)];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[i], b[j], acc[i][j], 0, 0, 0);
  }

#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long n = n0 + wn + i * 16 + fq * 4 + r;   // cout
        long k = k0 + wk + j * 16 + fr;           // cin
        if (n < Cout && k < Cin)
          atomicAdd(&dW[n * (long)(9 * Cin) + (long)tap * Cin + k],
                    acc[i][j][r]);
      }
}

// permute weight [Cout, 9, Cin] -> [Cin, 9, Cout] for dgrad.
// NO 180-degree rotation here: the dgrad gather (ho = h+1-kh) already
// encodes it, so tap k pairs with W[., k, .] directly.
__global__ void __launch_bounds__(AMD_TPB)
rotate_weight_kernel(const bf16* __restrict__ w, bf16* __restrict__ out,
                     int Cout, int Cin) {
  long total = (long)Cout * 9 * Cin;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long t = i;
    const int cin = (int)(t % Cin); t /= Cin;
    const int tap = (int)(t % 9); t /= 9;
    const int cout = (int)t;
    out[((long)cin * 9 + tap) * Cout + cout] =
        w[((long)cout * 9 + tap) * Cin + cin];
  }
}

}  // namespace

// AMDTRAIN_CONV3X3_DB=0 forces the single-buffer (tap, ks) loop; default
// is the double-buffered pipeline (see kernel comment / gemm.hip).
static bool conv3x3_db_enabled() {
  static const bool v = []() {
    const char* e = std::getenv("AMDTRAIN_CONV3X3_DB");
    return !(e && e[0] == '0');
  }();
  return v;
}

// host-side 16B zero page (device memory), one per device, created lazily
static at::Tensor zero_page_for(const at::Tensor& like) {
  static thread_local at::Tensor zp;
  if (!zp.defined() || zp.device() != like.device())
    zp = at::zeros({16}, like.options().dtype(at::kBFloat16));
  return zp;
}

std::vector<at::Tensor> conv3x3_fwd_stats_impl(at::Tensor x2d, long Nn,
                                                long H, long W, long stride,
                                                at::Tensor w2d,
                                                bool want_stats,
                                                bool banded = false) {
  // x2d: [Nn*H*W, Cin] bf16 NHWC rows; w2d: [Cout, 9*Cin]
  TORCH_CHECK(x2d.is_cuda() && x2d.scalar_type() == at::kBFloat16);
  long Cin = x2d.size(1), Cout = w2d.size(0);
  TORCH_CHECK(w2d.size(1) == 9 * Cin);
  TORCH_CHECK(Cin % BK == 0, "Cin must be multiple of 32");
  long Hout = (H + 2 - 3) / stride + 1, Wout = (W + 2 - 3) / stride + 1;
  long M = Nn * Hout * Wout;
  auto y = at::empty({M, Cout}, x2d.options());
  ConvGeom g{(int)H, (int)W, (int)Hout, (int)Wout, (int)stride};
  int nbm = (int)((M + 127) / 128), nbn = (int)((Cout + 127) / 128);
  auto zp = zero_page_for(x2d);
  at::Tensor stats;
  float* stats_ptr = nullptr;
  if (want_stats) {
    stats = at::empty({nbm, 2 * Cout}, x2d.options().dtype(at::kFloat));
    stats_ptr = stats.data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  static const bool bk64 = []() {
    const char* v = std::getenv("AMDTRAIN_CONV3X3_BK64");
    return !(v && v[0] == '0');
  }();
  // A/B (tools/bench_conv3x3.py + in-context traces): BK64 fwd +17% at
  // layer2 (Cin=128, M=400k); layer1 (Cin=64) and layer3/4 (small M)
  // measured WORSE in-context -> Cin>=128 && large-M only
  if (banded) {  // requires Cin == Cout, both % 128 (host-checked)
    // NT=64 tile halves the group-band window (and the off-diag waste)
    int nbn64 = (int)((Cout + 63) / 64);
    if (conv3x3_db_enabled())
      conv3x3_kernel<false, BK, true, 64, true>
          <<<nbm * nbn64, GEMM_TPB, 0, stream>>>(
              (const bf16*)x2d.const_data_ptr(),
              (const bf16*)w2d.const_data_ptr(), (bf16*)y.data_ptr(), M,
              (int)Cin, (int)Cout, g, nbm, nbn64,
              (const bf16*)zp.const_data_ptr(), stats_ptr);
    else
      conv3x3_kernel<false, BK, true, 64>
          <<<nbm * nbn64, GEMM_TPB, 0, stream>>>(
              (const bf16*)x2d.const_data_ptr(),
              (const bf16*)w2d.const_data_ptr(), (bf16*)y.data_ptr(), M,
              (int)Cin, (int)Cout, g, nbm, nbn64,
              (const bf16*)zp.const_data_ptr(), stats_ptr);
  } else if (bk64 && Cin % 64 == 0 && Cin >= 128 && M >= 200000) {
    if (conv3x3_db_enabled())
      conv3x3_kernel<false, 64, false, 128, true>
          <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
              (const bf16*)x2d.const_data_ptr(),
              (const bf16*)w2d.const_data_ptr(), (bf16*)y.data_ptr(), M,
              (int)Cin, (int)Cout, g, nbm, nbn,
              (const bf16*)zp.const_data_ptr(), stats_ptr);
    else
      conv3x3_kernel<false, 64><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
          (const bf16*)x2d.const_data_ptr(),
          (const bf16*)w2d.const_data_ptr(), (bf16*)y.data_ptr(), M,
          (int)Cin, (int)Cout, g, nbm, nbn,
          (const bf16*)zp.const_data_ptr(), stats_ptr);
  } else if (conv3x3_db_enabled())
    conv3x3_kernel<false, BK, false, 128, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)x2d.const_data_ptr(),
            (const bf16*)w2d.const_data_ptr(), (bf16*)y.data_ptr(), M,
            (int)Cin, (int)Cout, g, nbm, nbn,
            (const bf16*)zp.const_data_ptr(), stats_ptr);
  else
    conv3x3_kernel<false><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)x2d.const_data_ptr(), (const bf16*)w2d.const_data_ptr(),
        (bf16*)y.data_ptr(), M, (int)Cin, (int)Cout, g, nbm, nbn,
        (const bf16*)zp.const_data_ptr(), stats_ptr);
  CHECK_CUDA_OK();
  if (want_stats) return {y, stats};
  return {y};
}

at::Tensor conv3x3_fwd(at::Tensor x2d, long Nn, long H, long W, long stride,
                       at::Tensor w2d) {
  return conv3x3_fwd_stats_impl(x2d, Nn, H, W, stride, w2d, false)[0];
}

std::vector<at::Tensor> conv3x3_fwd_stats(at::Tensor x2d, long Nn, long H,
                                          long W, long stride,
                                          at::Tensor w2d, bool banded) {
  TORCH_CHECK(!banded || (x2d.size(1) == w2d.size(0) &&
                          x2d.size(1) % 128 == 0),
              "banded conv3x3 needs Cin == Cout, both % 128");
  return conv3x3_fwd_stats_impl(x2d, Nn, H, W, stride, w2d, true, banded);
}

at::Tensor conv3x3_dgrad(at::Tensor dy2d, long Nn, long H, long W,
                         long stride, at::Tensor w2d, bool banded) {
  // dy2d: [Nn*Hout*Wout, Cout]; returns dx2d [Nn*H*W, Cin]
  long Cout = dy2d.size(1), Cin = w2d.size(1) / 9;
  TORCH_CHECK(w2d.size(0) == Cout && w2d.size(1) == 9 * Cin);
  TORCH_CHECK(Cout % BK == 0);
  long Hout = (H + 2 - 3) / stride + 1, Wout = (W + 2 - 3) / stride + 1;
  long M = Nn * H * W;
  auto stream = at::cuda::getCurrentCUDAStream();
  // build rotated-permuted weight W' [Cin, 9*Cout]
  auto wrot = at::empty({Cin, 9 * Cout}, w2d.options());
  rotate_weight_kernel<<<amd_grid(Cout * 9 * Cin), AMD_TPB, 0, stream>>>(
      (const bf16*)w2d.const_data_ptr(), (bf16*)wrot.data_ptr(), (int)Cout,
      (int)Cin);
  CHECK_CUDA_OK();
  auto dx = at::empty({M, Cin}, dy2d.options());
  ConvGeom g{(int)H, (int)W, (int)Hout, (int)Wout, (int)stride};
  int nbm = (int)((M + 127) / 128), nbn = (int)((Cin + 127) / 128);
  auto zp = zero_page_for(dy2d);
  if (banded) {
    TORCH_CHECK(Cin == Cout && Cin % 128 == 0);
    // NT=64 tile: the group-diagonal K window shrinks to 64 channels,
    // halving the off-diagonal waste vs the 128-wide tile
    int nbn64 = (int)(Cin / 64);
    if (conv3x3_db_enabled())
      conv3x3_kernel<true, BK, true, 64, true>
          <<<nbm * nbn64, GEMM_TPB, 0, stream>>>(
              (const bf16*)dy2d.const_data_ptr(),
              (const bf16*)wrot.const_data_ptr(), (bf16*)dx.data_ptr(), M,
              (int)Cout, (int)Cin, g, nbm, nbn64,
              (const bf16*)zp.const_data_ptr(), nullptr);
    else
      conv3x3_kernel<true, BK, true, 64>
          <<<nbm * nbn64, GEMM_TPB, 0, stream>>>(
              (const bf16*)dy2d.const_data_ptr(),
              (const bf16*)wrot.const_data_ptr(), (bf16*)dx.data_ptr(), M,
              (int)Cout, (int)Cin, g, nbm, nbn64,
              (const bf16*)zp.const_data_ptr(), nullptr);
    CHECK_CUDA_OK();
    return dx;
  }
  static const bool bk64d = []() {  // measured negative for dgrad: off
    const char* v = std::getenv("AMDTRAIN_CONV3X3_BK64D");
    return v && v[0] == '1';
  }();
  if (bk64d && Cout % 64 == 0)
    conv3x3_kernel<true, 64><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)dy2d.const_data_ptr(),
        (const bf16*)wrot.const_data_ptr(), (bf16*)dx.data_ptr(), M,
        (int)Cout, (int)Cin, g, nbm, nbn,
        (const bf16*)zp.const_data_ptr(), nullptr);
  else if (conv3x3_db_enabled())
    conv3x3_kernel<true, BK, false, 128, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)dy2d.const_data_ptr(),
            (const bf16*)wrot.const_data_ptr(), (bf16*)dx.data_ptr(), M,
            (int)Cout, (int)Cin, g, nbm, nbn,
            (const bf16*)zp.const_data_ptr(), nullptr);
  else
    conv3x3_kernel<true><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)dy2d.const_data_ptr(),
        (const bf16*)wrot.const_data_ptr(), (bf16*)dx.data_ptr(), M,
        (int)Cout, (int)Cin, g, nbm, nbn,
        (const bf16*)zp.const_data_ptr(), nullptr);
  CHECK_CUDA_OK();
  return dx;
}

at::Tensor conv3x3_wgrad(at::Tensor dy2d, at::Tensor x2d, long Nn, long H,
                         long W, long stride) {
  long Cout = dy2d.size(1), Cin = x2d.size(1);
  long Hout = (H + 2 - 3) / stride + 1, Wout = (W + 2 - 3) / stride + 1;
  long M = Nn * Hout * Wout;
  TORCH_CHECK(dy2d.size(0) == M);
  auto dW = at::zeros({Cout, 9 * Cin}, dy2d.options().dtype(at::kFloat));
  ConvGeom g{(int)H, (int)W, (int)Hout, (int)Wout, (int)stride};
  int nbn = (int)((Cout + 127) / 128), nbk = (int)((Cin + 127) / 128);
  long tiles = (long)nbn * nbk;
  const bool det = std::getenv("AMDTRAIN_DETERMINISTIC") != nullptr;
  int msplit = det ? 1
                   : (int)std::max<long>(
                         1, std::min<long>((M + 31) / 32, 512 / tiles));
  auto stream = at::cuda::getCurrentCUDAStream();
  for (int tap = 0; tap < 9; ++tap) {
    conv3x3_wgrad_kernel<<<(int)(tiles * msplit), GEMM_TPB, 0, stream>>>(
        (const bf16*)dy2d.const_data_ptr(), (const bf16*)x2d.const_data_ptr(),
        dW.data_ptr<float>(), M, (int)Cout, (int)Cin, g, tap, nbn, nbk,
        msplit);
    CHECK_CUDA_OK();
  }
  return dW;
}
