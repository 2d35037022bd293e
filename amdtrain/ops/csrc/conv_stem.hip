// Generic implicit-GEMM convolution for gfx950 — covers the ResNet stem
// (7x7 stride-2 pad-3, Cin=3) and any other odd configuration.
//
// Motivation (measured): torch-ROCm's MIOpen path for the stem wgrad
// nondeterministically selects `naive_conv_..._wrw` at ~1.2 s/call
// (profiles/rocprof_kernel_stats_allcustom.txt) — the hot path must not
// depend on MIOpen find.  Here the im2col K dimension (KH*KW*Cin, e.g. 147)
// is padded to a multiple of 32 and A-tiles are staged by per-element
// gather (scalar loads + LDS writes; the data is tiny — Cin=3 — so staging
// amplification is ~940 MB/step at b256, ~0.15 ms of HBM time), then the
// same 128x128 MFMA block structure as gemm.hip runs the math.
//
//   fwd:   Y[m, cout] = sum_col im2col[m, col] * W2[cout, col]
//   wgrad: dW2[cout, col] = sum_m dY[m, cout] * im2col[m, col]
//   dgrad: dX[m=(n,h,w), ci] = sum_col2 im2colT[m, col2] * W2p[ci, col2]
//          with col2 = tap*Cout + cout, im2colT gathering dY through the
//          transposed-conv map ho = (h + pad - kh) / stride (exact
//          divisions only), W2p = weight permuted to [Cin, KH*KW*Cout].
//          (The stem itself never needs it — x is data — but any other
//          odd-shaped conv does)
#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int GEMM_TPB = 256;
constexpr int BK = 32;

struct StemGeom {
  int H, W, Hout, Wout;    // input / output spatial
  int KH, KW, Cin;         // filter
  int stride, pad;
  int Kpad;                // padded im2col K (multiple of 32)
};

// per-row (m) decode, hoisted out of the per-element gather: output pixel ->
// input-space base coordinates
struct StemCoord {
  long n_off;
  int hb, wb;
};

template <bool DGRAD = false>
__device__ __forceinline__ StemCoord stem_decode(long m, const StemGeom& g) {
  StemCoord u;
  long t = m;
  if (DGRAD) {  // m ranges over INPUT pixels; gather reads dY [Hout,Wout]
    const int w = (int)(t % g.W); t /= g.W;
    const int h = (int)(t % g.H); t /= g.H;
    u.n_off = t * (long)g.Hout * g.Wout;
    u.hb = h + g.pad;
    u.wb = w + g.pad;
    return u;
  }
  const int wo = (int)(t % g.Wout); t /= g.Wout;
  const int ho = (int)(t % g.Hout); t /= g.Hout;
  u.n_off = t * (long)g.H * g.W;
  u.hb = ho * g.stride - g.pad;
  u.wb = wo * g.stride - g.pad;
  return u;
}

// per-column decode via reciprocal multiply (integer div is ~20 cyc and was
// executed per element per K-step before)
__device__ __forceinline__ void col_decode(int col, const StemGeom& g,
                                           float inv_cin, float inv_kw,
                                           int& kh, int& kw, int& ci) {
  const int tap = (int)((float)col * inv_cin + 1e-4f);
  ci = col - tap * g.Cin;
  kh = (int)((float)tap * inv_kw + 1e-4f);
  kw = tap - kh * g.KW;
}

// element gather using hoisted row coords (DGRAD: g.Cin holds the conv's
// Cout — the channel count of the gathered dY rows)
template <bool DGRAD = false>
__device__ __forceinline__ long stem_elem(const StemCoord& u, int col,
                                          const StemGeom& g, float inv_cin,
                                          float inv_kw) {
  if (col >= g.KH * g.KW * g.Cin) return -1;  // K padding
  int kh, kw, ci;
  col_decode(col, g, inv_cin, inv_kw, kh, kw, ci);
  if (DGRAD) {
    int ho2 = u.hb - kh, wo2 = u.wb - kw;
    if (g.stride > 1) {
      if (ho2 % g.stride || wo2 % g.stride) return -1;
      ho2 /= g.stride;
      wo2 /= g.stride;
    }
    if (ho2 < 0 || ho2 >= g.Hout || wo2 < 0 || wo2 >= g.Wout) return -1;
    return (u.n_off + (long)ho2 * g.Wout + wo2) * g.Cin + ci;
  }
  const int h = u.hb + kh, w = u.wb + kw;
  if (h < 0 || h >= g.H || w < 0 || w >= g.W) return -1;
  return (u.n_off + (long)h * g.W + w) * g.Cin + ci;
}

// stage a [128 m][32 col] im2col tile into LDS (per-element gather with
// hoisted per-row decode — uc[] is precomputed once per kernel)
template <bool DGRAD = false>
__device__ __forceinline__ void stage_im2col(
    const bf16* __restrict__ x, const StemCoord* uc, const bool* valid,
    int c0, const StemGeom& g, float inv_cin, float inv_kw, bf16* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    const int unit = rnd * GEMM_TPB + t;  // 4 units of 8 cols per m-row
    const int cc0 = c0 + (unit & 3) * 8;
    bf16 v[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      long idx = valid[rnd]
                     ? stem_elem<DGRAD>(uc[rnd], cc0 + j, g, inv_cin, inv_kw)
                     : -1;
      v[j] = idx < 0 ? bf16(0.f) : x[idx];
    }
    *(Pack<bf16, 8>*)(lds + unit * 8) = *(Pack<bf16, 8>*)v;
  }
}

__device__ __forceinline__ void stage_rows(
    const bf16* __restrict__ gsrc, long ld, long row0, long rows, long koff,
    bf16* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    int unit = rnd * GEMM_TPB + t;
    long row = row0 + (unit >> 2);
    if (row >= rows) row = rows - 1;
    const bf16* src = gsrc + row * ld + koff + (unit & 3) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(lds + unit * 8), 16,
        0, 0);
  }
}


// Cin==8 fast path (the channel-padded stem): one 16 B unit == one tap's
// 8 channels, so the A tile stages via global_load_lds with a per-unit
// tap-gathered source row — same structure as conv3x3.hip's stage, no
// per-element scalar gather (the r1 elem-gather fwd measured 1.2 ms/step
// at b512, VALU-bound).
__device__ __forceinline__ void stage_im2col8(
    const bf16* __restrict__ x, const StemCoord* uc, const bool* valid,
    int c0, const StemGeom& g, const bf16* __restrict__ zp, bf16* lds) {
  const int t = threadIdx.x;
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    const int unit = rnd * GEMM_TPB + t;  // 4 units (taps) per m-row
    const int tap = (c0 >> 3) + (unit & 3);
    const bf16* src = zp;
    if (valid[rnd] && tap < g.KH * g.KW) {
      const int kh = tap / g.KW, kw = tap - (tap / g.KW) * g.KW;
      const int h = uc[rnd].hb + kh, w = uc[rnd].wb + kw;
      if (h >= 0 && h < g.H && w >= 0 && w < g.W)
        src = x + (uc[rnd].n_off + (long)h * g.W + w) * 8;
    }
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(lds + unit * 8), 16,
        0, 0);
  }
}

// DGRAD instantiation computes dX[m, cin] from gathered dY rows and the
// permuted weight; operand roles mirror fwd exactly.
// DB: double-buffered K-loop (gemm.hip pattern — stage ks+1's tiles while
// ks computes, one barrier per chunk), AMDTRAIN_STEM_DB=0 reverts.
template <bool DGRAD = false, bool C8 = false, bool DB = false>
__global__ void __launch_bounds__(GEMM_TPB, 2)
conv_generic_fwd_kernel(const bf16* __restrict__ x,
                        const bf16* __restrict__ W2, bf16* __restrict__ Y,
                        long M, int Cout, StemGeom g, int nbm, int nbn,
                        const bf16* __restrict__ zp) {
  __shared__ bf16 SMEM[(DB ? 2 : 1) * 2 * 128 * BK];
  bf16* const As = SMEM;
  bf16* const Bs = SMEM + 128 * BK;
  const int bid = blockIdx.x;
  const int bm = bid / nbn, bn = bid % nbn;
  const long m0 = (long)bm * 128, n0 = (long)bn * 128;
  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE, lane = t % AMD_WAVE;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * 64;
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const float inv_cin = 1.f / g.Cin, inv_kw = 1.f / g.KW;
  StemCoord uc[2];
  bool valid[2];
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    long m = m0 + (((rnd * GEMM_TPB) + t) >> 2);
    valid[rnd] = m < M;
    uc[rnd] = stem_decode<DGRAD>(valid[rnd] ? m : M - 1, g);
  }

  const int ksteps = g.Kpad / BK;
  auto stage = [&](int ks, bf16* as) {
    if (C8)
      stage_im2col8(x, uc, valid, ks * BK, g, zp, as);
    else
      stage_im2col<DGRAD>(x, uc, valid, ks * BK, g, inv_cin, inv_kw, as);
    stage_rows(W2, g.Kpad, n0, Cout, ks * BK, as + 128 * BK);
  };
  auto compute = [&](const bf16* as) {
    const bf16* bs = as + 128 * BK;
    bf16x8 a[4], b[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = *(const bf16x8*)&as[(wm + i * 16 + fr) * BK + fq * 8];
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
    constexpr int HB = 2 * 128 * BK;
    if (ksteps > 0) stage(0, SMEM);
    for (int ks = 0; ks < ksteps; ++ks) {
      bf16* const as = SMEM + (ks & 1) * HB;
      __syncthreads();  // implicit vmcnt(0) drains this chunk's DMA
      if (ks + 1 < ksteps) stage(ks + 1, SMEM + ((ks + 1) & 1) * HB);
      compute(as);
    }
  } else {
    for (int ks = 0; ks < ksteps; ++ks) {
      __syncthreads();
      stage(ks, SMEM);
      __syncthreads();
      compute(SMEM);
    }
  }
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = m0 + wm + i * 16 + fq * 4 + r;
        long col = n0 + wn + j * 16 + fr;
        if (row < M && col < Cout)
          Y[row * Cout + col] = __float2bfloat16(acc[i][j][r]);
      }
}

// wgrad: transposed reg-staging like gemm.hip's gemm_tn, with the B chunk
// gathered from im2col
__device__ __forceinline__ int st_swz(int c, int m) {
  return c * 32 + (m ^ (((((c >> 3) & 3) ^ ((c >> 1) & 3))) << 3));
}

__global__ void __launch_bounds__(GEMM_TPB, 2)
conv_generic_wgrad_kernel(const bf16* __restrict__ dY,
                          const bf16* __restrict__ x, float* __restrict__ dW2,
                          long M, int Cout, StemGeom g, int nbn, int nbk,
                          int msplit) {
  __shared__ bf16 Ys[128 * 32];
  __shared__ bf16 Xs[128 * 32];
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

  // col -> (kh, kw, ci) is FIXED per thread across all m-chunks: hoist it
  const float inv_cin = 1.f / g.Cin, inv_kw = 1.f / g.KW;
  const int K = g.KH * g.KW * g.Cin;
  int ckh[2][8], ckw[2][8], cci[2][8];
  bool cok[2][8];
#pragma unroll
  for (int rnd = 0; rnd < 2; ++rnd) {
    const int unit = rnd * GEMM_TPB + t;
    const int c0 = (unit & 15) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int col = (int)(k0 + c0 + j);
      cok[rnd][j] = col < K;
      col_decode(cok[rnd][j] ? col : 0, g, inv_cin, inv_kw, ckh[rnd][j],
                 ckw[rnd][j], cci[rnd][j]);
    }
  }

  for (long mc = mc0; mc < mc1; ++mc) {
    const long m0 = mc * 32;
    __syncthreads();
#pragma unroll
    for (int rnd = 0; rnd < 2; ++rnd) {
      const int unit = rnd * GEMM_TPB + t;
      const long m = m0 + (unit >> 4);
      const int c0 = (unit & 15) * 8;
      const int mloc = unit >> 4;
      bf16 yv[8];
      if (m < M && n0 + c0 + 8 <= Cout) {
        uint4 raw = *(const uint4*)(dY + m * Cout + n0 + c0);
        __builtin_memcpy(yv, &raw, 16);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) yv[j] = bf16(0.f);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) Ys[st_swz(c0 + j, mloc)] = yv[j];
      bf16 xv[8];
      StemCoord u = stem_decode(m < M ? m : 0, g);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int h = u.hb + ckh[rnd][j], w = u.wb + ckw[rnd][j];
        const bool ok = (m < M) && cok[rnd][j] && h >= 0 && h < g.H &&
                        w >= 0 && w < g.W;
        long idx = (u.n_off + (long)h * g.W + w) * g.Cin + cci[rnd][j];
        xv[j] = ok ? x[idx] : bf16(0.f);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) Xs[st_swz(c0 + j, mloc)] = xv[j];
    }
    __syncthreads();

    bf16x8 a[4], b[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      a[i] = *(const bf16x8*)&Ys[st_swz(wn + i * 16 + fr, fq * 8)];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b[j] = *(const bf16x8*)&Xs[st_swz(wk + j * 16 + fr, fq * 8)];
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
        long n = n0 + wn + i * 16 + fq * 4 + r;
        long k = k0 + wk + j * 16 + fr;
        if (n < Cout && k < g.Kpad)
          atomicAdd(&dW2[n * (long)g.Kpad + k], acc[i][j][r]);
      }
}

static StemGeom make_geom(long H, long W, long KH, long KW, long Cin,
                          long stride, long pad) {
  StemGeom g;
  g.H = (int)H; g.W = (int)W;
  g.Hout = (int)((H + 2 * pad - KH) / stride + 1);
  g.Wout = (int)((W + 2 * pad - KW) / stride + 1);
  g.KH = (int)KH; g.KW = (int)KW; g.Cin = (int)Cin;
  g.stride = (int)stride; g.pad = (int)pad;
  int K = (int)(KH * KW * Cin);
  g.Kpad = (K + 31) / 32 * 32;
  return g;
}

static at::Tensor stem_zero_page(const at::Tensor& like) {
  static thread_local at::Tensor zp;
  if (!zp.defined() || zp.device() != like.device())
    zp = at::zeros({16}, like.options().dtype(at::kBFloat16));
  return zp;
}

}  // namespace


// AMDTRAIN_STEM_DB=0 reverts to the single-buffer K-loop
static bool stem_db_enabled() {
  static const bool v = []() {
    const char* e = std::getenv("AMDTRAIN_STEM_DB");
    return !(e && e[0] == '0');
  }();
  return v;
}

// x2d: [N*H*W, Cin] bf16 NHWC rows; w2: [Cout, Kpad] (host-padded).
// Cin == 8 (channel-padded stem) takes the vectorized tap-gather staging.
at::Tensor conv_generic_fwd(at::Tensor x2d, long Nn, long H, long W,
                            long KH, long KW, long stride, long pad,
                            at::Tensor w2) {
  TORCH_CHECK(x2d.is_cuda() && x2d.scalar_type() == at::kBFloat16);
  long Cin = x2d.size(1), Cout = w2.size(0);
  auto g = make_geom(H, W, KH, KW, Cin, stride, pad);
  TORCH_CHECK(w2.size(1) == g.Kpad, "weight must be K-padded to ", g.Kpad);
  long M = Nn * g.Hout * g.Wout;
  auto y = at::empty({M, Cout}, x2d.options());
  int nbm = (int)((M + 127) / 128), nbn = (int)((Cout + 127) / 128);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto zp = stem_zero_page(x2d);
  const bool db = stem_db_enabled();
  if (Cin == 8 && g.Kpad == (KH * KW * 8 + 31) / 32 * 32) {
    if (db)
      conv_generic_fwd_kernel<false, true, true>
          <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
              (const bf16*)x2d.const_data_ptr(),
              (const bf16*)w2.const_data_ptr(), (bf16*)y.data_ptr(), M,
              (int)Cout, g, nbm, nbn, (const bf16*)zp.const_data_ptr());
    else
      conv_generic_fwd_kernel<false, true>
          <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
              (const bf16*)x2d.const_data_ptr(),
              (const bf16*)w2.const_data_ptr(), (bf16*)y.data_ptr(), M,
              (int)Cout, g, nbm, nbn, (const bf16*)zp.const_data_ptr());
  } else if (db)
    conv_generic_fwd_kernel<false, false, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)x2d.const_data_ptr(),
            (const bf16*)w2.const_data_ptr(), (bf16*)y.data_ptr(), M,
            (int)Cout, g, nbm, nbn, (const bf16*)zp.const_data_ptr());
  else
    conv_generic_fwd_kernel<false><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)x2d.const_data_ptr(), (const bf16*)w2.const_data_ptr(),
        (bf16*)y.data_ptr(), M, (int)Cout, g, nbm, nbn,
        (const bf16*)zp.const_data_ptr());
  CHECK_CUDA_OK();
  return y;
}

// dy2d: [N*Hout*Wout, Cout]; w2p: [Cin, pad32(KH*KW*Cout)] (weight permuted
// to cin-major, host-padded).  Returns dx2d [N*H*W, Cin].
at::Tensor conv_generic_dgrad(at::Tensor dy2d, at::Tensor w2p, long Nn,
                              long H, long W, long KH, long KW, long stride,
                              long pad) {
  TORCH_CHECK(dy2d.is_cuda() && dy2d.scalar_type() == at::kBFloat16);
  long Cout = dy2d.size(1), Cin = w2p.size(0);
  // geometry carries the FORWARD conv dims; g.Cin = Cout (gathered rows)
  auto g = make_geom(H, W, KH, KW, Cout, stride, pad);
  TORCH_CHECK(w2p.size(1) == g.Kpad, "w2p must be K-padded to ", g.Kpad);
  TORCH_CHECK(dy2d.size(0) == Nn * g.Hout * g.Wout);
  long M = Nn * H * W;
  auto dx = at::empty({M, Cin}, dy2d.options());
  int nbm = (int)((M + 127) / 128), nbn = (int)((Cin + 127) / 128);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto zp = stem_zero_page(dy2d);
  if (stem_db_enabled())
    conv_generic_fwd_kernel<true, false, true>
        <<<nbm * nbn, GEMM_TPB, 0, stream>>>(
            (const bf16*)dy2d.const_data_ptr(),
            (const bf16*)w2p.const_data_ptr(), (bf16*)dx.data_ptr(), M,
            (int)Cin, g, nbm, nbn, (const bf16*)zp.const_data_ptr());
  else
    conv_generic_fwd_kernel<true><<<nbm * nbn, GEMM_TPB, 0, stream>>>(
        (const bf16*)dy2d.const_data_ptr(),
        (const bf16*)w2p.const_data_ptr(), (bf16*)dx.data_ptr(), M,
        (int)Cin, g, nbm, nbn, (const bf16*)zp.const_data_ptr());
  CHECK_CUDA_OK();
  return dx;
}

at::Tensor conv_generic_wgrad(at::Tensor dy2d, at::Tensor x2d, long Nn,
                              long H, long W, long KH, long KW, long stride,
                              long pad) {
  long Cin = x2d.size(1), Cout = dy2d.size(1);
  auto g = make_geom(H, W, KH, KW, Cin, stride, pad);
  long M = Nn * g.Hout * g.Wout;
  TORCH_CHECK(dy2d.size(0) == M);
  auto dW2 = at::zeros({Cout, g.Kpad}, dy2d.options().dtype(at::kFloat));
  int nbn = (int)((Cout + 127) / 128), nbk = (int)((g.Kpad + 127) / 128);
  long tiles = (long)nbn * nbk;
  const bool det = std::getenv("AMDTRAIN_DETERMINISTIC") != nullptr;
  int msplit = det ? 1
                   : (int)std::max<long>(
                         1, std::min<long>((M + 31) / 32, 512 / tiles));
  auto stream = at::cuda::getCurrentCUDAStream();
  conv_generic_wgrad_kernel<<<(int)(tiles * msplit), GEMM_TPB, 0, stream>>>(
      (const bf16*)dy2d.const_data_ptr(), (const bf16*)x2d.const_data_ptr(),
      dW2.data_ptr<float>(), M, (int)Cout, g, nbn, nbk, msplit);
  CHECK_CUDA_OK();
  return dW2;
}
