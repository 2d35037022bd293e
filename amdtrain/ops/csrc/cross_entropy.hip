// Fused cross-entropy (log-softmax + NLL) forward/backward for gfx950.
//
// Reference launches separate softmax/NLL kernels via nn.CrossEntropyLoss
// (distributed.py:151,251).  Here: one wave per [C]-row doing an
// online max/sum-exp in fp32 (works for bf16 or fp32 logits), and an
// elementwise backward computing (softmax - onehot) * grad_scale from the
// saved per-row logsumexp.
#include "common.h"

namespace {

template <typename T>
__global__ void __launch_bounds__(AMD_TPB)
ce_fwd_kernel(const T* __restrict__ logits, const long* __restrict__ target,
              float* __restrict__ losses, float* __restrict__ lse, long B,
              long C) {
  const int wave = threadIdx.x / AMD_WAVE;
  const int lane = threadIdx.x % AMD_WAVE;
  const long row = (long)blockIdx.x * (AMD_TPB / AMD_WAVE) + wave;
  if (row >= B) return;
  const T* x = logits + row * C;

  // per-lane online max + scaled sum
  float m = -INFINITY, s = 0.f;
  for (long c = lane; c < C; c += AMD_WAVE) {
    float v = to_f32(x[c]);
    float m2 = fmaxf(m, v);
    s = s * __expf(m - m2) + __expf(v - m2);
    m = m2;
  }
  // wave reduce (max, then rescaled sums).  Lanes beyond C keep m = -inf;
  // exp(-inf - (-inf)) is NaN, so -inf partials contribute exactly 0.
#pragma unroll
  for (int off = AMD_WAVE / 2; off > 0; off >>= 1) {
    float mo = __shfl_down(m, off, AMD_WAVE);
    float so = __shfl_down(s, off, AMD_WAVE);
    float m2 = fmaxf(m, mo);
    float fa = (m == -INFINITY) ? 0.f : __expf(m - m2);
    float fb = (mo == -INFINITY) ? 0.f : __expf(mo - m2);
    s = s * fa + so * fb;
    m = m2;
  }
  if (lane == 0) {
    float l = m + __logf(s);
    lse[row] = l;
    losses[row] = l - to_f32(x[target[row]]);
  }
}

template <typename T>
__global__ void __launch_bounds__(AMD_TPB)
ce_bwd_kernel(const T* __restrict__ logits, const long* __restrict__ target,
              const float* __restrict__ lse, T* __restrict__ grad,
              const float* __restrict__ upstream, float inv_B, long B,
              long C) {
  // upstream grad is read on-device (no host .item() sync; graph-capturable)
  const float gscale = upstream[0] * inv_B;
  const long total = B * C;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long r = i / C, c = i - r * C;
    float p = __expf(to_f32(logits[i]) - lse[r]);
    float g = (p - (c == target[r] ? 1.f : 0.f)) * gscale;
    grad[i] = from_f32<T>(g);
  }
}

}  // namespace

std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits,
                                          at::Tensor target) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2);
  TORCH_CHECK(target.scalar_type() == at::kLong);
  auto lc = logits.contiguous();
  long B = lc.size(0), C = lc.size(1);
  auto losses = at::empty({B}, lc.options().dtype(at::kFloat));
  auto lse = at::empty({B}, lc.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentCUDAStream();
  int rows_per_block = AMD_TPB / AMD_WAVE;
  int grid = (int)((B + rows_per_block - 1) / rows_per_block);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, lc.scalar_type(),
      "ce_fwd", [&] {
        using devT = typename DevT<scalar_t>::type;
        ce_fwd_kernel<devT><<<grid, AMD_TPB, 0, stream>>>(
            (const devT*)lc.const_data_ptr(), target.data_ptr<long>(),
            losses.data_ptr<float>(), lse.data_ptr<float>(), B, C);
        CHECK_CUDA_OK();
      });
  return {losses, lse};
}

at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor target,
                             at::Tensor lse, at::Tensor upstream) {
  // upstream: scalar fp32 tensor on device (the grad flowing into the mean
  // loss); the kernel folds in the 1/B of the mean reduction itself
  auto lc = logits.contiguous();
  long B = lc.size(0), C = lc.size(1);
  auto grad = at::empty_like(lc);
  auto up = upstream.to(at::kFloat).contiguous();
  auto stream = at::cuda::getCurrentCUDAStream();
  int grid = amd_grid(B * C);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, lc.scalar_type(),
      "ce_bwd", [&] {
        using devT = typename DevT<scalar_t>::type;
        ce_bwd_kernel<devT><<<grid, AMD_TPB, 0, stream>>>(
            (const devT*)lc.const_data_ptr(), target.data_ptr<long>(),
            lse.data_ptr<float>(), (devT*)grad.data_ptr(),
            up.data_ptr<float>(), 1.f / (float)B, B, C);
        CHECK_CUDA_OK();
      });
  return grad;
}
