// TN ("wgrad") GEMM core v2 for gfx950 — dW[N, K] += sum_m dY[m,n] * Xg[m,k]
// covering conv1x1 wgrad (K = Cin), strided-1x1 wgrad (gathered X rows) and
// conv3x3 wgrad (K = 9*Cin, im2col column gather) in ONE kernel family
// (SURVEY §2c rows "Linear/GEMM (cuBLAS)" wgrad + "Conv2d 3x3" wgrad;
// reference hot path /root/reference/distributed.py:268 loss.backward()).
//
// Why v2 (round-1 profile: conv3x3_wgrad 19 ms/step + gemm_tn 10 ms/step):
//  * the round-1 TN kernels transposed through REGISTERS with per-element
//    swizzled ds_write_b16 (32 x 2B writes per thread per 32-m chunk) and
//    drained two barriers per chunk for only 16 MFMA/wave;
//  * conv3x3 wgrad launched 9 separate kernels re-reading dY and X per tap;
//  * split-M accumulated through fp32 atomicAdd — heavily contended on the
//    small early-layer dW tiles.
// v2 fixes all three:
//  * staging is __builtin_amdgcn_global_load_lds DIRECT into a subtiled
//    layout consumed by ds_read_b64_tr_b16 (gfx950 hardware transpose read,
//    guide T10): no register round-trip, vectorized 16 B stores, and the
//    9 taps are just columns of one [N, 9*Cin] output (dY staged ONCE per
//    m-chunk instead of 9 times);
//  * 2-phase double-buffer (guide §5.5 minimum-2-phase recipe): stage the
//    next m-chunk while MFMAs consume the current one, one vmcnt(0)+barrier
//    per chunk, 32 MFMA/wave between barriers (MC=64);
//  * split-M partial outputs go to a [msplit, N*K] fp32 buffer with plain
//    stores + a vectorized collapse kernel — no atomics, and the reduction
//    order is FIXED, so wgrad v2 is bitwise deterministic by construction
//    (no AMDTRAIN_DETERMINISTIC slow path needed).
//
// LDS layout ("tr-subtile", derived from the guide's m156 mapping): the
// [MC x NCOLS] operand tile is stored as 256-element (512 B) groups
// G = ((m>>5)*NCOLS/16 + (c>>4))*2 + ((m>>2)&1), element (m,c) at byte
//   G*512 + ((m>>3)&3)*128 + (m&3)*32 + (c&15)*2 .
// A group's four 128 B windows are exactly what one ds_read_b64_tr_b16
// returns per 16-lane quarter (lane l, elem j <- window byte
// (l&15)*2 + j*32), so two tr-reads (r = 0/1 groups, +512 B apart)
// assemble the full bf16x8 MFMA operand
//   frag[jj] = T[m = 8*(l>>4) + jj][c = l&15],   jj = 0..7.
// Bank behavior: a staging wave's 64 concurrent 16 B stores land on the 8
// 16 B positions of the 128 B bank cycle exactly 8x each (position =
// (2*(m&3) + ((c>>3)&1)) mod 8) — evenly distributed, i.e. the 8-cycle
// floor for 1 KiB of LDS writes with no aliasing penalty; each tr-read
// window is 128 B contiguous, so reads are conflict-free too.
#include "common.h"

namespace {

using bf16 = __hip_bfloat16;
using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using u32x2 = __attribute__((ext_vector_type(2))) unsigned int;

constexpr int TN2_TPB = 256;  // 4 waves
constexpr int GROUP_BYTES = 512;

__device__ __forceinline__ int xcd_swz_tn2(int bid, int nwg) {
  constexpr int NXCD = 8;
  if (nwg < NXCD) return bid;
  int xcd = bid % NXCD, idx = bid / NXCD;
  int q = nwg / NXCD, r = nwg % NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

struct TnGeom {
  int H, W, Hout, Wout, stride;
  int kh, kw, pad;  // filter geometry (3,3,1 for the bottleneck 3x3 convs)
};

// generic conv forward gather: output row m=(n,ho,wo) + tap -> X row or -1
__device__ __forceinline__ long tn2_gather_conv(long m, int tap,
                                                const TnGeom& g) {
  long t = m;
  const int wo = (int)(t % g.Wout); t /= g.Wout;
  const int ho = (int)(t % g.Hout); t /= g.Hout;
  const int h = ho * g.stride - g.pad + tap / g.kw;
  const int w = wo * g.stride - g.pad + tap % g.kw;
  if (h < 0 || h >= g.H || w < 0 || w >= g.W) return -1;
  return (t * g.H + h) * g.W + w;
}

// strided-1x1 gather: compact output row m=(n,ho,wo) -> input row
__device__ __forceinline__ long tn2_gather_stride(long m, const TnGeom& g) {
  long t = m;
  const int wo = (int)(t % g.Wout); t /= g.Wout;
  const int ho = (int)(t % g.Hout); t /= g.Hout;
  return (t * g.H + (long)ho * g.stride) * g.W + (long)wo * g.stride;
}

// byte offset of element (m, c) of a [MC x NCOLS] tile in tr-subtile layout
template <int NCOLS>
__device__ __forceinline__ int st_byte(int m, int c) {
  const int sub = (m >> 5) * (NCOLS / 16) + (c >> 4);
  const int G = sub * 2 + ((m >> 2) & 1);
  return G * GROUP_BYTES + ((m >> 3) & 3) * 128 + (m & 3) * 32 + (c & 15) * 2;
}

template <int NCOLS, int MC>
constexpr int tile_bytes() {
  return (MC / 32) * (NCOLS / 16) * 2 * GROUP_BYTES;
}

// Stage one [MC x NCOLS] operand tile into tr-subtile LDS via
// global_load_lds.  MODE 0: rows are m (plain; also the dY operand),
// MODE 1: strided-1x1 gather, MODE 2: conv3x3 tap gather (cols span taps).
// Out-of-range rows/cols pull from the 16 B zero page.
//
// global_load_lds is a WAVE-UNIFORM-BASE DMA: the hardware writes lane i's
// 16 B at base + i*16 regardless of per-lane LDS addresses (guide: "CDNA's
// is wave-uniform-base not per-lane scatter").  So each WAVE stages one
// complete 1 KiB subtile (32 m x 16 c), whose tr-layout bytes are exactly
// subtile_base + lane*16 under the lane -> (m, c) decode
//   m' = 8*((lane>>3)&3) + 4*(lane>>5) + ((lane>>1)&3),
//   c  = (lane&1)*8 .. +8
// (inverse of st_byte: lane*16 = r*512 + q*128 + j*32 + c_half*2).
template <int MODE, int NCOLS, int MC, int NWAVES = 4>
__device__ __forceinline__ void stage_tn2(
    const bf16* __restrict__ g, int ld, long m0, long M, int col0,
    int total_cols, int cin, const TnGeom& geo, const bf16* __restrict__ zp,
    bf16* lds) {
  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE, lane = t % AMD_WAVE;
  constexpr int SUBTILES = (MC / 32) * (NCOLS / 16);
  const int r = lane >> 5, q = (lane >> 3) & 3, j = (lane >> 1) & 3;
  const int c_half = (lane & 1) * 8;
  const int mfrag = q * 8 + r * 4 + j;  // m offset within the 32-row block
#pragma unroll
  for (int ss = 0; ss < SUBTILES / NWAVES; ++ss) {
    const int s = ss * NWAVES + wave;
    const int mb = s / (NCOLS / 16);         // 32-row block
    const int cs = (s % (NCOLS / 16)) * 16;  // subtile column base
    const int mloc = mb * 32 + mfrag;
    const long m = m0 + mloc;
    const int col = col0 + cs + c_half;
    const bf16* src = zp;
    if (m < M && col + 8 <= total_cols) {
      if (MODE == 0) {
        src = g + m * (long)ld + col;
      } else if (MODE == 1) {
        src = g + tn2_gather_stride(m, geo) * (long)ld + col;
      } else {
        const int tap = col / cin;  // unit never spans taps (cin % 8 == 0)
        const long row = tn2_gather_conv(m, tap, geo);
        if (row >= 0) src = g + row * (long)ld + (col - tap * cin);
      }
    }
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)(
            (char*)lds + s * 1024 + lane * 16),
        16, 0, 0);
  }
}

template <int N>
__device__ __forceinline__ void vmcnt_wait() {
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 2) asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 5) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
  else if constexpr (N == 6) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else if constexpr (N == 8) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else if constexpr (N == 10)
    asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
  else if constexpr (N == 12)
    asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
  else if constexpr (N == 16)
    asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
  else static_assert(N == 0, "unsupported vmcnt");
}

// Two hardware transpose reads -> one bf16x8 MFMA operand.
// addr = LDS tile base + lane*8 + (runtime wave column offset); OFF is the
// compile-time byte offset of the (sub, r=0) group; r=1 is the next group.
template <int OFF>
__device__ __forceinline__ bf16x8 tr_frag(unsigned addr) {
  u32x2 v0, v1;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2 offset:%3\n\t"
      "ds_read_b64_tr_b16 %1, %2 offset:%4"
      : "=&v"(v0), "=&v"(v1)
      : "v"(addr), "i"(OFF), "i"(OFF + GROUP_BYTES));
  union {
    unsigned int u[4];
    bf16x8 f;
  } pack;
  pack.u[0] = v0.x;
  pack.u[1] = v0.y;
  pack.u[2] = v1.x;
  pack.u[3] = v1.y;
  return pack.f;
}

// NW x KW = wave grid over the (n, k) output tile (NW*KW == 4); per-wave
// sub-tile is 64x64 (4x4 fragments).  MC = m rows staged per chunk.
// NBUF = LDS pipeline depth: 2 = classic double buffer with a full
// vmcnt(0) drain per chunk; 3 = stage TWO chunks ahead and only wait for
// the NEWEST chunk's loads at each barrier (counted vmcnt, guide T4) —
// each wave drains to <= its own newest-chunk loads before the barrier,
// so the chunk consumed next iteration is complete for every wave.
// BAND: group-diagonal wgrad (grouped conv, Cin == N, both % 128): each
// n-tile needs only ONE k-tile per tap (its own 128-channel window), so
// tiles = nbn*9 and the output is the COMPACT [N, 9*NCX] band.
template <int GMODE, int NW, int KW, int MC, int NBUF = 2, bool BAND = false>
__global__ void __launch_bounds__(NW * KW * 64, (NW * KW == 4 ? 2 : 1))
tn2_kernel(const bf16* __restrict__ dY, const bf16* __restrict__ X,
           float* __restrict__ parts, long M, int N, int K9, int Cin,
           TnGeom geo, int nbn, int nbk, int msplit,
           const bf16* __restrict__ zp) {
  constexpr int NWAVES = NW * KW;  // 4 (256 thr, 2 blk/CU) or 8 (512, 1)
  constexpr int NCY = NW * 64;   // dY tile cols
  constexpr int NCX = KW * 64;   // X tile cols
  constexpr int KSTEPS = MC / 32;
  // per-thread global_load_lds issues per chunk (both operands):
  // subtiles = (MC/32)*(cols/16) per operand, one wave-round each
  constexpr int LC = (MC / 32) * (NCY / 16 + NCX / 16) * 64 / (NW * KW * 64);
  __shared__ __align__(16) char Ys[NBUF][tile_bytes<NCY, MC>()];
  __shared__ __align__(16) char Xs[NBUF][tile_bytes<NCX, MC>()];

  const int tiles = nbn * nbk;
  const int bid = xcd_swz_tn2(blockIdx.x, tiles * msplit);
  const int tile = bid % tiles;
  const int mpart = bid / tiles;
  const int bn = tile / nbk, bk = tile % nbk;
  const int n0 = bn * NCY;
  // BAND: bk is the TAP index; the k window is this n-tile's own channels
  const int k0 = BAND ? bk * Cin + n0 : bk * NCX;

  const long mchunks = (M + MC - 1) / MC;
  const long cpp = (mchunks + msplit - 1) / msplit;
  const long mc0 = (long)mpart * cpp;
  const long mc1 = min(mc0 + cpp, mchunks);

  const int t = threadIdx.x;
  const int wave = t / AMD_WAVE, lane = t % AMD_WAVE;
  const int wn = (wave / KW) * 64;  // n offset of this wave's sub-tile
  const int wk = (wave % KW) * 64;  // k offset
  const int fr = lane & 15, fq = lane >> 4;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  if (mc0 < mc1) {
    // per-wave LDS base addresses (bytes; +lane*8 is the tr-read's own slot)
    const unsigned ybase0 =
        (unsigned)(unsigned long long)(&Ys[0][0]) + lane * 8 +
        (wn / 16) * 2 * GROUP_BYTES;
    const unsigned xbase0 =
        (unsigned)(unsigned long long)(&Xs[0][0]) + lane * 8 +
        (wk / 16) * 2 * GROUP_BYTES;
    constexpr unsigned YB = tile_bytes<NCY, MC>();
    constexpr unsigned XB = tile_bytes<NCX, MC>();

    stage_tn2<0, NCY, MC, NWAVES>(dY, N, mc0 * MC, M, n0, N, Cin, geo, zp,
                                  (bf16*)Ys[0]);
    stage_tn2<GMODE, NCX, MC, NWAVES>(X, Cin, mc0 * MC, M, k0, K9, Cin, geo,
                                      zp, (bf16*)Xs[0]);
    if (NBUF >= 3 && mc0 + 1 < mc1) {
      stage_tn2<0, NCY, MC, NWAVES>(dY, N, (mc0 + 1) * MC, M, n0, N, Cin,
                                    geo, zp, (bf16*)Ys[1]);
      stage_tn2<GMODE, NCX, MC, NWAVES>(X, Cin, (mc0 + 1) * MC, M, k0, K9,
                                        Cin, geo, zp, (bf16*)Xs[1]);
      vmcnt_wait<LC>();  // chunk 0 complete; chunk 1 may stay in flight
    } else {
      vmcnt_wait<0>();
    }
    __builtin_amdgcn_s_barrier();

    for (long mc = mc0; mc < mc1; ++mc) {
      const int cur = (int)((mc - mc0) % NBUF);
      const long pre = NBUF >= 3 ? mc + 2 : mc + 1;
      if (pre < mc1) {
        const int nxt = (int)((pre - mc0) % NBUF);
        stage_tn2<0, NCY, MC, NWAVES>(dY, N, pre * MC, M, n0, N, Cin, geo,
                                      zp, (bf16*)Ys[nxt]);
        stage_tn2<GMODE, NCX, MC, NWAVES>(X, Cin, pre * MC, M, k0, K9, Cin,
                                          geo, zp, (bf16*)Xs[nxt]);
      }
      const unsigned yb = ybase0 + cur * YB;
      const unsigned xb = xbase0 + cur * XB;
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        bf16x8 a[4], b[4];
        // each 16-col subtile is 2 groups; the m-half (ks) advances by a
        // full row of subtiles
#pragma unroll
        for (int i = 0; i < 4; ++i)
          a[i] = tr_frag<0>(yb + (ks * (NCY / 16) + i) * 2 * GROUP_BYTES);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          b[j] = tr_frag<0>(xb + (ks * (NCX / 16) + j) * 2 * GROUP_BYTES);
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_sched_barrier(0);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a[i], b[j], acc[i][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
      if (NBUF >= 3 && mc + 2 < mc1)
        vmcnt_wait<LC>();  // only the newest chunk's loads may remain
      else
        vmcnt_wait<0>();
      __builtin_amdgcn_s_barrier();
    }
  }

  // plain (non-atomic) per-mpart partial tile; BAND writes the compact
  // [N, 9*NCX] layout (col = tap*NCX + in-window offset)
  const int kout_cols = BAND ? 9 * NCX : K9;
  float* out = parts + (long)mpart * N * (long)kout_cols;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int n = n0 + wn + i * 16 + fq * 4 + r;
        const int klocal = wk + j * 16 + fr;
        const int k = BAND ? bk * NCX + klocal : k0 + klocal;
        if (n < N && (BAND || k < K9))
          out[(long)n * kout_cols + k] = acc[i][j][r];
      }
}

// out[i] = sum over p in {slot, slot+step, ...} of parts[p][i] — fixed-order,
// vectorized.  Used two-stage for large msplit (a single-stage collapse of
// e.g. 170 partials x 36 K elems leaves only ~36 blocks = latency-bound;
// stage 1 folds to TN2_FOLD rows in parallel, stage 2 sums those).
constexpr int TN2_FOLD = 16;

__global__ void __launch_bounds__(AMD_TPB)
tn2_collapse_kernel(const float* __restrict__ parts, float* __restrict__ out,
                    int nparts, int step, long NK) {
  const long nv = NK / 4;
  const int slot = blockIdx.x % step;
  for (long i = (long)(blockIdx.x / step) * blockDim.x + threadIdx.x; i < nv;
       i += (long)(gridDim.x / step) * blockDim.x) {
    f32x4 s = ((const f32x4*)(parts + (long)slot * NK))[i];
    for (int p = slot + step; p < nparts; p += step) {
      f32x4 v = ((const f32x4*)(parts + (long)p * NK))[i];
      s.x += v.x; s.y += v.y; s.z += v.z; s.w += v.w;
    }
    ((f32x4*)(out + (long)slot * NK))[i] = s;
  }
}

// semantics probe for ds_read_b64_tr_b16 (GPU test asserts the m156
// mapping: lane l, elem j  <-  lds[(l&15) + j*16 + (l>>4)*64])
__global__ void tr16_probe_kernel(const short* __restrict__ in,
                                  short* __restrict__ out) {
  __shared__ short lds[256];
  for (int i = threadIdx.x; i < 256; i += blockDim.x) lds[i] = in[i];
  __syncthreads();
  if (threadIdx.x < 64) {
    const unsigned addr =
        (unsigned)(unsigned long long)(&lds[0]) + threadIdx.x * 8;
    u32x2 v;
    asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
                 : "=&v"(v) : "v"(addr));
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    short4 s;
    __builtin_memcpy(&s, &v, 8);
    out[threadIdx.x * 4 + 0] = s.x;
    out[threadIdx.x * 4 + 1] = s.y;
    out[threadIdx.x * 4 + 2] = s.z;
    out[threadIdx.x * 4 + 3] = s.w;
  }
}

static at::Tensor tn2_zero_page(const at::Tensor& like) {
  static thread_local at::Tensor zp;
  if (!zp.defined() || zp.device() != like.device())
    zp = at::zeros({16}, like.options().dtype(at::kBFloat16));
  return zp;
}

template <int GMODE, int NW, int KW, int MC, int NBUF = 2>
void tn2_launch(const at::Tensor& dY, const at::Tensor& X, at::Tensor& out,
                long M, int N, int K9, int Cin, TnGeom geo, int target_blocks,
                hipStream_t stream) {
  const int nbn = (N + NW * 64 - 1) / (NW * 64);
  const int nbk = (K9 + KW * 64 - 1) / (KW * 64);
  const long tiles = (long)nbn * nbk;
  const long mchunks = (M + MC - 1) / MC;
  int msplit = (int)std::max<long>(
      1, std::min<long>(mchunks, target_blocks / tiles));
  auto zp = tn2_zero_page(dY);
  at::Tensor parts = out;
  if (msplit > 1)
    parts = at::empty({msplit, (long)N * K9}, out.options());
  tn2_kernel<GMODE, NW, KW, MC, NBUF>
      <<<(int)(tiles * msplit), NW * KW * 64, 0, stream>>>(
          (const bf16*)dY.const_data_ptr(), (const bf16*)X.const_data_ptr(),
          parts.data_ptr<float>(), M, N, K9, Cin, geo, nbn, nbk, msplit,
          (const bf16*)zp.const_data_ptr());
  CHECK_CUDA_OK();
  if (msplit > 1) {
    const long NK = (long)N * K9;
    TORCH_CHECK(NK % 4 == 0);
    const int nkgrid = amd_grid(NK / 4);
    if (msplit > 2 * TN2_FOLD) {
      auto folded = at::empty({TN2_FOLD, NK}, out.options());
      tn2_collapse_kernel<<<nkgrid * TN2_FOLD, AMD_TPB, 0, stream>>>(
          (const float*)parts.const_data_ptr(), folded.data_ptr<float>(),
          msplit, TN2_FOLD, NK);
      CHECK_CUDA_OK();
      tn2_collapse_kernel<<<nkgrid, AMD_TPB, 0, stream>>>(
          (const float*)folded.const_data_ptr(), out.data_ptr<float>(),
          TN2_FOLD, 1, NK);
      CHECK_CUDA_OK();
    } else {
      tn2_collapse_kernel<<<nkgrid, AMD_TPB, 0, stream>>>(
          (const float*)parts.const_data_ptr(), out.data_ptr<float>(), msplit,
          1, NK);
      CHECK_CUDA_OK();
    }
  }
}

}  // namespace

// Unified TN wgrad v2.  gmode: 0 = plain rows (1x1 s1 / linear), 1 =
// strided-1x1 row gather, 2 = conv3x3 tap gather (taps*Cin columns).
// Returns fp32 [N, taps*Cin]; bitwise deterministic (fixed split + ordered
// collapse).
at::Tensor tn2_wgrad(at::Tensor dY, at::Tensor X, long taps, long Nn, long H,
                     long W, long stride, long gmode, long kh, long kw,
                     long pad) {
  TORCH_CHECK(dY.is_cuda() && dY.scalar_type() == at::kBFloat16 &&
              X.scalar_type() == at::kBFloat16);
  auto Yc = dY.contiguous();
  auto Xc = X.contiguous();
  const long M = Yc.size(0);
  const int N = (int)Yc.size(1);
  const int Cin = (int)Xc.size(1);
  const int K9 = (int)(taps * Cin);
  TORCH_CHECK(Cin % 8 == 0 && N % 8 == 0);
  TnGeom geo{0, 0, 0, 0, 1, 3, 3, 1};
  if (gmode == 1) {
    long Hout = (H + stride - 1) / stride, Wout = (W + stride - 1) / stride;
    TORCH_CHECK(M == Nn * Hout * Wout);
    geo = TnGeom{(int)H, (int)W, (int)Hout, (int)Wout, (int)stride, 1, 1, 0};
  } else if (gmode == 2) {
    long Hout = (H + 2 * pad - kh) / stride + 1;
    long Wout = (W + 2 * pad - kw) / stride + 1;
    TORCH_CHECK(M == Nn * Hout * Wout && taps == kh * kw);
    geo = TnGeom{(int)H, (int)W, (int)Hout, (int)Wout, (int)stride,
                 (int)kh, (int)kw, (int)pad};
  } else {
    TORCH_CHECK(M == Xc.size(0));
  }
  auto out = at::empty({(long)N, (long)K9}, Yc.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const char* tb = std::getenv("AMDTRAIN_TN2_BLOCKS");
  const int target = tb ? atoi(tb) : 512;

  // config: (2,2) 128x128 tile MC=64 default; skinny-N with wide K -> (1,4);
  // skinny-K with tall N -> (4,1) (both MC=32 to keep 2 blocks/CU in LDS)
  if (N < 128 && K9 >= 256) {
    // skinny-N conv wgrad: the (2,2,64) tile zero-pads the N half but its
    // MC=64 pipeline beats the fully-dense (1,4,32) config (which is
    // issue-bound at 16 MFMA/barrier) for layer-1-class shapes; the
    // channel-padded STEM (Cin=8) measured the other way (1.55 vs 1.78 ms
    // at b512) — its 16x-redundant tap gather prefers the narrower k-tile.
    // A/B via AMDTRAIN_TN2_14
    if (gmode == 2 && Cin >= 32 && std::getenv("AMDTRAIN_TN2_14") == nullptr)
      tn2_launch<2, 2, 2, 64>(Yc, Xc, out, M, N, K9, Cin, geo, target, stream);
    else if (gmode == 2)
      tn2_launch<2, 1, 4, 32>(Yc, Xc, out, M, N, K9, Cin, geo, target, stream);
    else if (gmode == 1)
      tn2_launch<1, 2, 2, 64>(Yc, Xc, out, M, N, K9, Cin, geo, target, stream);
    else
      tn2_launch<0, 1, 4, 32>(Yc, Xc, out, M, N, K9, Cin, geo, target, stream);
  } else if (K9 < 128 && N >= 256 && gmode == 0) {
    tn2_launch<0, 4, 1, 32>(Yc, Xc, out, M, N, K9, Cin, geo, target, stream);
  } else {
    static const bool pipe3 = []() {
      const char* v = std::getenv("AMDTRAIN_TN2_PIPE3");
      return v && v[0] == '1';
    }();
    static const bool wide8 = []() {  // A/B override
      const char* v = std::getenv("AMDTRAIN_TN2_W8");
      return v && v[0] == '1';
    }();
    // 8-wave (2m x 4k) block halves dY re-reads (two k-tiles per staged
    // chunk); measured +11% at deep-K (K9=4608 layer4 3x3), flat at
    // smaller K (L3 already absorbs those re-reads) -> auto for K>=4096
    if ((wide8 || (gmode == 2 && K9 >= 4096)) && K9 % 256 == 0
        && gmode != 1) {
      if (gmode == 2)
        tn2_launch<2, 2, 4, 64>(Yc, Xc, out, M, N, K9, Cin, geo, target,
                                stream);
      else
        tn2_launch<0, 2, 4, 64>(Yc, Xc, out, M, N, K9, Cin, geo, target,
                                stream);
    } else if (pipe3) {
      // A/B: 3-buffer counted-vmcnt pipeline at MC=32 (48 KB LDS -> 3
      // blocks/CU).  MEASURED -10..-15% vs the MC=64 double buffer
      // (docs/KERNELS.md negative-results) — kept off-by-default
      if (gmode == 2)
        tn2_launch<2, 2, 2, 32, 3>(Yc, Xc, out, M, N, K9, Cin, geo, target,
                                   stream);
      else if (gmode == 1)
        tn2_launch<1, 2, 2, 32, 3>(Yc, Xc, out, M, N, K9, Cin, geo, target,
                                   stream);
      else
        tn2_launch<0, 2, 2, 32, 3>(Yc, Xc, out, M, N, K9, Cin, geo, target,
                                   stream);
    } else if (gmode == 2) {
      tn2_launch<2, 2, 2, 64>(Yc, Xc, out, M, N, K9, Cin, geo, target, stream);
    } else if (gmode == 1) {
      tn2_launch<1, 2, 2, 64>(Yc, Xc, out, M, N, K9, Cin, geo, target, stream);
    } else {
      tn2_launch<0, 2, 2, 64>(Yc, Xc, out, M, N, K9, Cin, geo, target, stream);
    }
  }
  return out;
}

// Group-diagonal conv3x3 wgrad: returns the COMPACT band [Cout, 9*128]
// (column = tap*128 + channel offset within the cout-row's own 128-wide
// group window).  Requires Cin == Cout, both % 128.
at::Tensor tn2_wgrad_banded(at::Tensor dY, at::Tensor X, long Nn, long H,
                            long W, long stride) {
  TORCH_CHECK(dY.is_cuda() && dY.scalar_type() == at::kBFloat16 &&
              X.scalar_type() == at::kBFloat16);
  auto Yc = dY.contiguous();
  auto Xc = X.contiguous();
  const long M = Yc.size(0);
  const int N = (int)Yc.size(1);
  const int Cin = (int)Xc.size(1);
  TORCH_CHECK(N == Cin && N % 128 == 0);
  long Hout = (H + 2 - 3) / stride + 1, Wout = (W + 2 - 3) / stride + 1;
  TORCH_CHECK(M == Nn * Hout * Wout);
  TnGeom geo{(int)H, (int)W, (int)Hout, (int)Wout, (int)stride, 3, 3, 1};
  auto out = at::empty({(long)N, (long)9 * 128},
                       Yc.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  const char* tb = std::getenv("AMDTRAIN_TN2_BLOCKS");
  const int target = tb ? atoi(tb) : 512;
  const int nbn = N / 128, nbk = 9;
  const long tiles = (long)nbn * nbk;
  const long mchunks = (M + 63) / 64;
  int msplit = (int)std::max<long>(
      1, std::min<long>(mchunks, target / tiles));
  auto zp = tn2_zero_page(Yc);
  at::Tensor parts = out;
  if (msplit > 1)
    parts = at::empty({msplit, (long)N * 9 * 128}, out.options());
  tn2_kernel<2, 2, 2, 64, 2, true>
      <<<(int)(tiles * msplit), TN2_TPB, 0, stream>>>(
          (const bf16*)Yc.const_data_ptr(), (const bf16*)Xc.const_data_ptr(),
          parts.data_ptr<float>(), M, N, 9 * Cin, Cin, geo, nbn, nbk, msplit,
          (const bf16*)zp.const_data_ptr());
  CHECK_CUDA_OK();
  if (msplit > 1) {
    const long NK = (long)N * 9 * 128;
    const int nkgrid = amd_grid(NK / 4);
    if (msplit > 2 * TN2_FOLD) {
      auto folded = at::empty({TN2_FOLD, NK}, out.options());
      tn2_collapse_kernel<<<nkgrid * TN2_FOLD, AMD_TPB, 0, stream>>>(
          (const float*)parts.const_data_ptr(), folded.data_ptr<float>(),
          msplit, TN2_FOLD, NK);
      CHECK_CUDA_OK();
      tn2_collapse_kernel<<<nkgrid, AMD_TPB, 0, stream>>>(
          (const float*)folded.const_data_ptr(), out.data_ptr<float>(),
          TN2_FOLD, 1, NK);
      CHECK_CUDA_OK();
    } else {
      tn2_collapse_kernel<<<nkgrid, AMD_TPB, 0, stream>>>(
          (const float*)parts.const_data_ptr(), out.data_ptr<float>(),
          msplit, 1, NK);
      CHECK_CUDA_OK();
    }
  }
  return out;
}

at::Tensor tr16_probe(at::Tensor in) {
  TORCH_CHECK(in.is_cuda() && in.scalar_type() == at::kShort &&
              in.numel() == 256);
  auto inc = in.contiguous();
  auto out = at::zeros({64, 4}, inc.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  tr16_probe_kernel<<<1, 64, 0, stream>>>(
      (const short*)inc.const_data_ptr(), out.data_ptr<short>());
  CHECK_CUDA_OK();
  return out;
}
