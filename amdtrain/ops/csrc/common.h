// Shared helpers for the amdtrain gfx950 (CDNA4) HIP kernels.
//
// Conventions (per the CDNA4 programming model):
//  * wavefront = 64 lanes (hard-coded; NOT 32)
//  * block size = 256 (4 waves) unless stated otherwise
//  * memory-bound kernels vectorize loads/stores to 16 B per lane
//    (Pack<T,N>), grid-stride, grid capped ~2048 blocks
//  * NHWC ("channels_last") layout: a logical [N,C,H,W] tensor is stored as
//    rows of C contiguous channels, R = N*H*W rows
#pragma once

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define AMD_WAVE 64
#define AMD_TPB 256
#define AMD_MAX_BLOCKS 2048

#define CHECK_CUDA_OK()                                                     \
  do {                                                                      \
    hipError_t e = hipGetLastError();                                       \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",              \
                hipGetErrorString(e));                                      \
  } while (0)

static inline int amd_grid(long total, int tpb = AMD_TPB,
                           int cap = AMD_MAX_BLOCKS) {
  long g = (total + tpb - 1) / tpb;
  return (int)std::min<long>(g, cap);
}

// ---- device dtype mapping: at:: scalar types -> HIP device types ----------

template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = __half; };

template <typename T> __device__ __forceinline__ float to_f32(T v);
template <> __device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <> __device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <> __device__ __forceinline__ float to_f32<__half>(__half v) {
  return __half2float(v);
}
template <> __device__ __forceinline__ float to_f32<double>(double v) {
  return (float)v;  // double path exists only to satisfy AT_DISPATCH
}

template <typename T> __device__ __forceinline__ T from_f32(float v);
template <> __device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <> __device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <> __device__ __forceinline__ __half from_f32<__half>(float v) {
  return __float2half(v);
}
template <> __device__ __forceinline__ double from_f32<double>(float v) {
  return (double)v;
}

// 16-byte packed vector for coalesced loads (8x bf16/fp16 or 4x fp32)
template <typename T, int N> struct alignas(sizeof(T) * N) Pack {
  T v[N];
};
template <typename T> struct VecWidth {
  static constexpr int value = 16 / sizeof(T);
};

// ---- wave / block reductions ----------------------------------------------

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = AMD_WAVE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, AMD_WAVE);
  return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = AMD_WAVE / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, AMD_WAVE));
  return v;
}

// ---- multi-tensor-apply metadata (apex amp_C style, struct by value) -------

constexpr int MT_TENSORS = 24;
constexpr int MT_BLOCKS = 320;
constexpr long MT_CHUNK = 1 << 16;  // elements per block-chunk

struct MTMeta {
  const void* a[MT_TENSORS];
  void* b[MT_TENSORS];
  void* c[MT_TENSORS];
  long sizes[MT_TENSORS];
  short t_for_block[MT_BLOCKS];
  short chunk_for_block[MT_BLOCKS];
};

// Host-side chunking driver: fills MTMeta over (up to) three parallel tensor
// lists and invokes `launch(meta, nblocks, ntensors)` for each full batch.
template <typename LaunchFn>
static void mt_apply(const std::vector<at::Tensor>& A,
                     const std::vector<at::Tensor>* B,
                     const std::vector<at::Tensor>* C, LaunchFn launch) {
  MTMeta meta;
  int t = 0, blk = 0;
  for (size_t i = 0; i < A.size(); ++i) {
    long n = A[i].numel();
    meta.a[t] = A[i].const_data_ptr();
    meta.b[t] = B ? (*B)[i].data_ptr() : nullptr;
    meta.c[t] = C ? (*C)[i].data_ptr() : nullptr;
    meta.sizes[t] = n;
    long nchunks = (n + MT_CHUNK - 1) / MT_CHUNK;
    for (long ch = 0; ch < nchunks; ++ch) {
      meta.t_for_block[blk] = (short)t;
      meta.chunk_for_block[blk] = (short)ch;
      ++blk;
      if (blk == MT_BLOCKS) {
        launch(meta, blk, t + 1);
        // re-seed current tensor into slot 0 for remaining chunks
        meta.a[0] = meta.a[t];
        meta.b[0] = meta.b[t];
        meta.c[0] = meta.c[t];
        meta.sizes[0] = meta.sizes[t];
        t = 0;
        blk = 0;
      }
    }
    ++t;
    if (t == MT_TENSORS && i + 1 < A.size()) {
      if (blk) launch(meta, blk, t);
      t = 0;
      blk = 0;
    }
  }
  if (blk) launch(meta, blk, t);
}
