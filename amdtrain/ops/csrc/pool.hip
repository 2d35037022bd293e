// NHWC pooling kernels for gfx950:
//  * 3x3 stride-2 pad-1 max pool (ResNet stem) with u8 argmax
//    indices; backward is a deterministic gather (no atomics)
//  * global average pool (AdaptiveAvgPool2d((1,1)) parity) fwd/bwd
#include "common.h"

namespace {

template <typename T, int VEC>
__global__ void __launch_bounds__(AMD_TPB)
maxpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                   unsigned char* __restrict__ idx, long N, int H, int W,
                   int C, int Ho, int Wo) {
  const int gpr = C / VEC;
  const long total = N * Ho * Wo * gpr;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long t = i;
    const int gc = (int)(t % gpr); t /= gpr;
    const int wo = (int)(t % Wo); t /= Wo;
    const int ho = (int)(t % Ho); t /= Ho;
    const long n = t;
    const int c0 = gc * VEC;

    float best[VEC];
    unsigned char bidx[VEC];
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      best[k] = -INFINITY;
      bidx[k] = 255;
    }
    const int h0 = ho * 2 - 1, w0 = wo * 2 - 1;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int h = h0 + kh;
      if (h < 0 || h >= H) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int w = w0 + kw;
        if (w < 0 || w >= W) continue;
        const long base = ((n * H + h) * W + w) * C + c0;
        Pack<T, VEC> xv = *(const Pack<T, VEC>*)(x + base);
#pragma unroll
        for (int k = 0; k < VEC; ++k) {
          float v = to_f32(xv.v[k]);
          if (v > best[k]) {
            best[k] = v;
            bidx[k] = (unsigned char)(kh * 3 + kw);
          }
        }
      }
    }
    const long obase = ((n * Ho + ho) * Wo + wo) * C + c0;
    Pack<T, VEC> yv;
    Pack<unsigned char, VEC> iv;
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      yv.v[k] = from_f32<T>(best[k]);
      iv.v[k] = bidx[k];
    }
    *(Pack<T, VEC>*)(y + obase) = yv;
    *(Pack<unsigned char, VEC>*)(idx + obase) = iv;
  }
}

template <typename T, int VEC>
__global__ void __launch_bounds__(AMD_TPB)
maxpool_bwd_kernel(const T* __restrict__ gy, const unsigned char* __restrict__ idx,
                   T* __restrict__ gx, long N, int H, int W, int C, int Ho,
                   int Wo) {
  const int gpr = C / VEC;
  const long total = N * H * W * gpr;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long t = i;
    const int gc = (int)(t % gpr); t /= gpr;
    const int w = (int)(t % W); t /= W;
    const int h = (int)(t % H); t /= H;
    const long n = t;
    const int c0 = gc * VEC;

    float acc[VEC];
#pragma unroll
    for (int k = 0; k < VEC; ++k) acc[k] = 0.f;
    // windows (ho,wo) containing (h,w): ho*2-1 <= h <= ho*2+1
    const int ho_lo = max(0, (h - 1 + 1) / 2);   // ceil((h-1)/2)
    const int ho_hi = min(Ho - 1, (h + 1) / 2);
    const int wo_lo = max(0, (w - 1 + 1) / 2);
    const int wo_hi = min(Wo - 1, (w + 1) / 2);
    for (int ho = ho_lo; ho <= ho_hi; ++ho) {
      const int kh = h - (ho * 2 - 1);
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        const int kw = w - (wo * 2 - 1);
        const unsigned char code = (unsigned char)(kh * 3 + kw);
        const long obase = ((n * Ho + ho) * Wo + wo) * C + c0;
        Pack<unsigned char, VEC> iv =
            *(const Pack<unsigned char, VEC>*)(idx + obase);
        Pack<T, VEC> gv = *(const Pack<T, VEC>*)(gy + obase);
#pragma unroll
        for (int k = 0; k < VEC; ++k)
          if (iv.v[k] == code) acc[k] += to_f32(gv.v[k]);
      }
    }
    const long base = ((n * H + h) * W + w) * C + c0;
    Pack<T, VEC> out;
#pragma unroll
    for (int k = 0; k < VEC; ++k) out.v[k] = from_f32<T>(acc[k]);
    *(Pack<T, VEC>*)(gx + base) = out;
  }
}

template <typename T, int VEC>
__global__ void __launch_bounds__(AMD_TPB)
gap_fwd_kernel(const T* __restrict__ x, T* __restrict__ y, long N, int HW,
               int C) {
  const int gpr = C / VEC;
  const long total = N * gpr;
  const float inv = 1.f / HW;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long n = i / gpr;
    const int c0 = (int)(i % gpr) * VEC;
    float acc[VEC];
#pragma unroll
    for (int k = 0; k < VEC; ++k) acc[k] = 0.f;
    const T* xp = x + n * (long)HW * C + c0;
    for (int r = 0; r < HW; ++r) {
      Pack<T, VEC> xv = *(const Pack<T, VEC>*)(xp + (long)r * C);
#pragma unroll
      for (int k = 0; k < VEC; ++k) acc[k] += to_f32(xv.v[k]);
    }
    Pack<T, VEC> yv;
#pragma unroll
    for (int k = 0; k < VEC; ++k) yv.v[k] = from_f32<T>(acc[k] * inv);
    *(Pack<T, VEC>*)(y + n * C + c0) = yv;
  }
}

template <typename T, int VEC>
__global__ void __launch_bounds__(AMD_TPB)
gap_bwd_kernel(const T* __restrict__ gy, T* __restrict__ gx, long N, int HW,
               int C) {
  const int gpr = C / VEC;
  const long total = N * (long)HW * gpr;
  const float inv = 1.f / HW;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / gpr;       // n*HW + r
    const long n = row / HW;
    const int c0 = (int)(i % gpr) * VEC;
    Pack<T, VEC> gv = *(const Pack<T, VEC>*)(gy + n * C + c0);
    Pack<T, VEC> out;
#pragma unroll
    for (int k = 0; k < VEC; ++k)
      out.v[k] = from_f32<T>(to_f32(gv.v[k]) * inv);
    *(Pack<T, VEC>*)(gx + row * C + c0) = out;
  }
}

static void check_nhwc4(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
                  x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "expected NHWC CUDA tensor");
}

}  // namespace

std::vector<at::Tensor> max_pool_3x3_s2_fwd(at::Tensor x) {
  check_nhwc4(x);
  const long N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const long Ho = (H + 2 - 3) / 2 + 1, Wo = (W + 2 - 3) / 2 + 1;
  auto y = at::empty({N, C, Ho, Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto idx = at::empty({N, C, Ho, Wo}, x.options()
                                           .dtype(at::kByte)
                                           .memory_format(
                                               at::MemoryFormat::ChannelsLast));
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "maxpool_fwd", [&] {
        using devT = typename DevT<scalar_t>::type;
        constexpr int VEC = 16 / sizeof(devT);
        TORCH_CHECK(C % VEC == 0);
        long total = N * Ho * Wo * (C / VEC);
        maxpool_fwd_kernel<devT, VEC>
            <<<amd_grid(total), AMD_TPB, 0, stream>>>(
                (const devT*)x.const_data_ptr(), (devT*)y.data_ptr(),
                (unsigned char*)idx.data_ptr(), N, (int)H, (int)W, (int)C,
                (int)Ho, (int)Wo);
        CHECK_CUDA_OK();
      });
  return {y, idx};
}

at::Tensor max_pool_3x3_s2_bwd(at::Tensor grad_y, at::Tensor idx, long H,
                               long W) {
  check_nhwc4(grad_y);
  const long N = grad_y.size(0), C = grad_y.size(1), Ho = grad_y.size(2),
             Wo = grad_y.size(3);
  auto gx = at::empty({N, C, H, W},
                      grad_y.options().memory_format(
                          at::MemoryFormat::ChannelsLast));
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, grad_y.scalar_type(),
      "maxpool_bwd", [&] {
        using devT = typename DevT<scalar_t>::type;
        constexpr int VEC = 16 / sizeof(devT);
        long total = N * H * W * (C / VEC);
        maxpool_bwd_kernel<devT, VEC>
            <<<amd_grid(total), AMD_TPB, 0, stream>>>(
                (const devT*)grad_y.const_data_ptr(),
                (const unsigned char*)idx.const_data_ptr(),
                (devT*)gx.data_ptr(), N, (int)H, (int)W, (int)C, (int)Ho,
                (int)Wo);
        CHECK_CUDA_OK();
      });
  return gx;
}

at::Tensor global_avg_pool_fwd(at::Tensor x) {
  check_nhwc4(x);
  const long N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  auto y = at::empty({N, C, 1, 1}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "gap_fwd", [&] {
        using devT = typename DevT<scalar_t>::type;
        constexpr int VEC = 16 / sizeof(devT);
        long total = N * (C / VEC);
        gap_fwd_kernel<devT, VEC><<<amd_grid(total), AMD_TPB, 0, stream>>>(
            (const devT*)x.const_data_ptr(), (devT*)y.data_ptr(), N,
            (int)(H * W), (int)C);
        CHECK_CUDA_OK();
      });
  return y;
}

at::Tensor global_avg_pool_bwd(at::Tensor grad_y, long H, long W) {
  TORCH_CHECK(grad_y.is_cuda());
  auto g = grad_y.contiguous();
  const long N = g.size(0), C = g.size(1);
  auto gx = at::empty({N, C, H, W},
                      g.options().memory_format(
                          at::MemoryFormat::ChannelsLast));
  auto stream = at::cuda::getCurrentCUDAStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, g.scalar_type(),
      "gap_bwd", [&] {
        using devT = typename DevT<scalar_t>::type;
        constexpr int VEC = 16 / sizeof(devT);
        long total = N * H * W * (C / VEC);
        gap_bwd_kernel<devT, VEC><<<amd_grid(total), AMD_TPB, 0, stream>>>(
            (const devT*)g.const_data_ptr(), (devT*)gx.data_ptr(), N,
            (int)(H * W), (int)C);
        CHECK_CUDA_OK();
      });
  return gx;
}
