// Elementwise / reduction utility kernels for gfx950:
//  * normalize_u8      — fused uint8->float cast + (x-mean)/std (the apex
//                        prefetcher's GPU-side normalize,
//                        apex_distributed.py:119-122,157-158)
//  * topk_ranks        — rank of the label among [C] logits (accuracy(),
//                        distributed.py:381-395, without materializing topk)
//  * multi_tensor_scale_check — in-place scale + inf/nan detect (apex amp_C
//                        unscale, apex_distributed.py:328-329)
//  * multi_tensor_cast — fused dtype conversion (fp16/bf16 compression and
//                        O2 master-weight copies, horovod_distributed.py:159)
#include "common.h"

namespace {

// ---- normalize_u8: NHWC uint8 [N,H,W,3] -> float/bf16 ---------------------
// 16 pixels (48 B) per thread: 12 aligned u32 loads, vectorized 16 B stores
// (48 outputs = 6x uint4 for bf16, 12x uint4 for fp32).

template <typename T>
__global__ void __launch_bounds__(AMD_TPB)
normalize_u8_kernel(const unsigned char* __restrict__ x, T* __restrict__ y,
                    float m0, float m1, float m2, float i0, float i1, float i2,
                    long npix) {
  const float mean[3] = {m0, m1, m2};
  const float inv[3] = {i0, i1, i2};
  const long ngrp = npix / 16;  // 16 pixels = 48 bytes per group
  for (long q = (long)blockIdx.x * blockDim.x + threadIdx.x; q < ngrp;
       q += (long)gridDim.x * blockDim.x) {
    const long byte0 = q * 48;
    unsigned char b[48];
    const uint32_t* xu = (const uint32_t*)(x + byte0);
#pragma unroll
    for (int k = 0; k < 12; ++k) *(uint32_t*)(b + 4 * k) = xu[k];
    T out[48];
#pragma unroll
    for (int k = 0; k < 48; ++k) {
      int c = k % 3;
      out[k] = from_f32<T>(((float)b[k] - mean[c]) * inv[c]);
    }
    // 48 * sizeof(T) bytes, 16B-aligned (48*2=96 / 48*4=192)
    const uint4* src = (const uint4*)out;
    uint4* dst = (uint4*)(y + byte0);
#pragma unroll
    for (int k = 0; k < (int)(48 * sizeof(T) / 16); ++k) dst[k] = src[k];
  }
  // tail pixels (npix % 16)
  for (long pix = ngrp * 16 + (long)blockIdx.x * blockDim.x + threadIdx.x;
       pix < npix; pix += (long)gridDim.x * blockDim.x) {
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      long idx = pix * 3 + c;
      y[idx] = from_f32<T>(((float)x[idx] - mean[c]) * inv[c]);
    }
  }
}

// ---- topk_ranks: one wave per row --------------------------------------

template <typename T>
__global__ void __launch_bounds__(AMD_TPB)
topk_ranks_kernel(const T* __restrict__ logits, const long* __restrict__ target,
                  int* __restrict__ ranks, long B, long C) {
  const int wave = threadIdx.x / AMD_WAVE;
  const int lane = threadIdx.x % AMD_WAVE;
  const long row = (long)blockIdx.x * (AMD_TPB / AMD_WAVE) + wave;
  if (row >= B) return;
  const T* x = logits + row * C;
  const long tidx = target[row];
  const float tv = to_f32(x[tidx]);
  float cnt = 0.f;
  for (long c = lane; c < C; c += AMD_WAVE) {
    float v = to_f32(x[c]);
    // rank rule: higher value first; ties broken by lower index
    cnt += (v > tv || (v == tv && c < tidx)) ? 1.f : 0.f;
  }
  cnt = wave_reduce_sum(cnt);
  if (lane == 0) ranks[row] = (int)cnt;
}

// ---- multi-tensor scale + inf/nan check --------------------------------

template <typename T>
__global__ void __launch_bounds__(AMD_TPB)
mt_scale_kernel(MTMeta meta, float scale, float* __restrict__ found_inf) {
  const int t = meta.t_for_block[blockIdx.x];
  const long base = (long)meta.chunk_for_block[blockIdx.x] * MT_CHUNK;
  const long end = min(base + MT_CHUNK, meta.sizes[t]);
  T* __restrict__ p = (T*)meta.b[t];
  bool bad = false;
  for (long i = base + threadIdx.x; i < end; i += blockDim.x) {
    float v = to_f32(p[i]) * scale;
    bad |= !isfinite(v);
    p[i] = from_f32<T>(v);
  }
  if (__any(bad) && (threadIdx.x % AMD_WAVE) == 0) *found_inf = 1.f;
}

// ---- multi-tensor cast --------------------------------------------------

template <typename S, typename D>
__global__ void __launch_bounds__(AMD_TPB)
mt_cast_kernel(MTMeta meta) {
  const int t = meta.t_for_block[blockIdx.x];
  const long base = (long)meta.chunk_for_block[blockIdx.x] * MT_CHUNK;
  const long end = min(base + MT_CHUNK, meta.sizes[t]);
  const S* __restrict__ src = (const S*)meta.a[t];
  D* __restrict__ dst = (D*)meta.b[t];
  for (long i = base + threadIdx.x; i < end; i += blockDim.x) {
    dst[i] = from_f32<D>(to_f32(src[i]));
  }
}

// scatter_rows_x2: dx[n,h,w,:] = (h,w both even) ? src[n,h/2,w/2,:] : 0
// (stride-2 1x1-conv dgrad epilogue — one write pass instead of a zero fill
// plus a strided copy)
template <typename T, int VEC>
__global__ void __launch_bounds__(AMD_TPB)
scatter_rows_x2_kernel(const T* __restrict__ src, T* __restrict__ dst,
                       long N, int H, int W, int Hs, int Ws, int C) {
  const int gpr = C / VEC;
  const long total = N * H * W * gpr;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long t = i;
    const int gc = (int)(t % gpr); t /= gpr;
    const int w = (int)(t % W); t /= W;
    const int h = (int)(t % H); t /= H;
    const long n = t;
    Pack<T, VEC> v;
    if (((h | w) & 1) == 0 && h / 2 < Hs && w / 2 < Ws) {
      v = *(const Pack<T, VEC>*)(
          src + (((n * Hs + h / 2) * Ws) + w / 2) * (long)C + gc * VEC);
    } else {
#pragma unroll
      for (int k = 0; k < VEC; ++k) v.v[k] = from_f32<T>(0.f);
    }
    *(Pack<T, VEC>*)(dst + i * VEC) = v;
  }
}

}  // namespace

at::Tensor scatter_rows_x2(at::Tensor src2d, long Nn, long H, long W,
                           long Hs, long Ws) {
  // src2d: [Nn*Hs*Ws, C] -> [Nn, C, H, W] channels_last with zeros at odd
  TORCH_CHECK(src2d.is_cuda() && src2d.dim() == 2);
  long C = src2d.size(1);
  auto out = at::empty({Nn, C, H, W},
                       src2d.options().memory_format(
                           at::MemoryFormat::ChannelsLast));
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, src2d.scalar_type(),
      "scatter_rows_x2", [&] {
        using devT = typename DevT<scalar_t>::type;
        constexpr int VEC = 16 / sizeof(devT);
        TORCH_CHECK(C % VEC == 0);
        long total = Nn * H * W * (C / VEC);
        scatter_rows_x2_kernel<devT, VEC>
            <<<amd_grid(total), AMD_TPB, 0, stream>>>(
                (const devT*)src2d.const_data_ptr(), (devT*)out.data_ptr(),
                Nn, (int)H, (int)W, (int)Hs, (int)Ws, (int)C);
        CHECK_CUDA_OK();
      });
  return out;
}

at::Tensor normalize_u8(at::Tensor x, std::vector<double> mean,
                        std::vector<double> std_, long dtype_code) {
  // dtype_code: 0 = fp32, 1 = bf16, 2 = fp16 (the O2-fp16 prefetcher path)
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kByte);
  TORCH_CHECK(x.dim() == 4 && x.size(1) == 3, "expects NCHW with C=3");
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  long npix = x.size(0) * x.size(2) * x.size(3);
  at::ScalarType st = dtype_code == 1   ? at::kBFloat16
                      : dtype_code == 2 ? at::kHalf
                                        : at::kFloat;
  auto opts = x.options().dtype(st);
  auto y = at::empty(xc.sizes(),
                     opts.memory_format(at::MemoryFormat::ChannelsLast));
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = amd_grid(npix / 16 + 1);
  float m[3], iv[3];
  for (int c = 0; c < 3; ++c) {
    m[c] = (float)mean[c];
    iv[c] = 1.f / (float)std_[c];
  }
  if (dtype_code == 1) {
    normalize_u8_kernel<__hip_bfloat16><<<grid, AMD_TPB, 0, stream>>>(
        (const unsigned char*)xc.const_data_ptr(),
        (__hip_bfloat16*)y.data_ptr(), m[0], m[1], m[2], iv[0], iv[1], iv[2],
        npix);
  } else if (dtype_code == 2) {
    normalize_u8_kernel<__half><<<grid, AMD_TPB, 0, stream>>>(
        (const unsigned char*)xc.const_data_ptr(), (__half*)y.data_ptr(),
        m[0], m[1], m[2], iv[0], iv[1], iv[2], npix);
  } else {
    normalize_u8_kernel<float><<<grid, AMD_TPB, 0, stream>>>(
        (const unsigned char*)xc.const_data_ptr(), y.data_ptr<float>(), m[0],
        m[1], m[2], iv[0], iv[1], iv[2], npix);
  }
  CHECK_CUDA_OK();
  return y;
}

at::Tensor topk_ranks(at::Tensor logits, at::Tensor target) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2);
  auto lc = logits.contiguous();
  long B = lc.size(0), C = lc.size(1);
  auto tl = target.to(at::kLong).contiguous();
  auto ranks = at::empty({B}, lc.options().dtype(at::kInt));
  auto stream = at::cuda::getCurrentCUDAStream();
  int rows_per_block = AMD_TPB / AMD_WAVE;
  int grid = (int)((B + rows_per_block - 1) / rows_per_block);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, lc.scalar_type(),
      "topk_ranks", [&] {
        using devT = typename DevT<scalar_t>::type;
        topk_ranks_kernel<devT><<<grid, AMD_TPB, 0, stream>>>(
            (const devT*)lc.const_data_ptr(), tl.data_ptr<long>(),
            ranks.data_ptr<int>(), B, C);
        CHECK_CUDA_OK();
      });
  return ranks;
}

void multi_tensor_scale_check(std::vector<at::Tensor> tensors, double scale,
                              at::Tensor found_inf) {
  if (tensors.empty()) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  // group by dtype (BN params stay fp32 in the O2 path -> mixed lists)
  std::map<at::ScalarType, std::vector<at::Tensor>> groups;
  for (auto& t : tensors) groups[t.scalar_type()].push_back(t);
  for (auto& [st, group] : groups) {
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, st, "mt_scale", [&] {
          using devT = typename DevT<scalar_t>::type;
          mt_apply(group, &group, nullptr,
                   [&](const MTMeta& meta, int blocks, int) {
                     mt_scale_kernel<devT><<<blocks, AMD_TPB, 0, stream>>>(
                         meta, (float)scale, found_inf.data_ptr<float>());
                     CHECK_CUDA_OK();
                   });
        });
  }
}

void multi_tensor_cast(std::vector<at::Tensor> src,
                       std::vector<at::Tensor> dst) {
  TORCH_CHECK(src.size() == dst.size());
  if (src.empty()) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  std::map<std::pair<at::ScalarType, at::ScalarType>,
           std::pair<std::vector<at::Tensor>, std::vector<at::Tensor>>>
      groups;
  for (size_t i = 0; i < src.size(); ++i) {
    auto key = std::make_pair(src[i].scalar_type(), dst[i].scalar_type());
    groups[key].first.push_back(src[i]);
    groups[key].second.push_back(dst[i]);
  }
  for (auto& [key, pair] : groups) {
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, key.first, "mt_cast_s",
        [&] {
          using srcT = typename DevT<scalar_t>::type;
          AT_DISPATCH_FLOATING_TYPES_AND2(
              at::ScalarType::BFloat16, at::ScalarType::Half, key.second,
              "mt_cast_d", [&] {
                using dstT = typename DevT<scalar_t>::type;
                mt_apply(pair.first, &pair.second, nullptr,
                         [&](const MTMeta& meta, int blocks, int) {
                           mt_cast_kernel<srcT, dstT>
                               <<<blocks, AMD_TPB, 0, stream>>>(meta);
                           CHECK_CUDA_OK();
                         });
              });
        });
  }
}
