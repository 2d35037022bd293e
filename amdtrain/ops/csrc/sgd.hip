// Fused multi-tensor SGD (momentum + weight decay + optional nesterov +
// folded grad-zeroing) for gfx950.
//
// Replaces the reference's 161 per-tensor SGD kernels + 161 zero_grad fills
// per step (torch.optim.SGD, reference distributed.py:153-156,267-269;
// SURVEY §2c "fuse into one multi-tensor HIP kernel") with ~2 launches over
// chunked tensor lists.  Memory-bound: fp32, float4-vectorized, one
// read+write of params/grads/momenta per step.
#include "common.h"

namespace {

struct SGDArgs {
  float lr, momentum, weight_decay;
  bool nesterov, first, zero_grad;
};

template <bool HAS_MU>
__global__ void __launch_bounds__(AMD_TPB)
mt_sgd_kernel(MTMeta meta, SGDArgs args) {
  const int t = meta.t_for_block[blockIdx.x];
  const long base = (long)meta.chunk_for_block[blockIdx.x] * MT_CHUNK;
  const long n = meta.sizes[t];
  const long end = min(base + MT_CHUNK, n);
  float* __restrict__ p = (float*)meta.b[t];
  float* __restrict__ g = (float*)meta.c[t];
  float* __restrict__ m = (float*)const_cast<void*>(meta.a[t]);

  // grads can be views into bucket flats at arbitrary offsets — vectorize
  // only when all three pointers are 16B-aligned (uniform branch per block)
  bool vec_ok = ((((uintptr_t)p | (uintptr_t)g |
                   (uintptr_t)(HAS_MU ? m : p)) & 15) == 0);
  if (!vec_ok) {
    for (long i = base + threadIdx.x; i < end; i += blockDim.x) {
      float pe = p[i];
      float ge = g[i] + args.weight_decay * pe;
      float u;
      if (HAS_MU) {
        float me = args.first ? ge : args.momentum * m[i] + ge;
        m[i] = me;
        u = args.nesterov ? ge + args.momentum * me : me;
      } else {
        u = ge;
      }
      p[i] = pe - args.lr * u;
      if (args.zero_grad) g[i] = 0.f;
    }
    return;
  }

  // float4 main body over the 16B-aligned interior of the chunk
  long i0 = base + threadIdx.x * 4;
  for (long i = i0; i + 3 < end; i += (long)blockDim.x * 4) {
    float4 pv = *(const float4*)(p + i);
    float4 gv = *(const float4*)(g + i);
    float4 mv;
    if (HAS_MU) mv = *(const float4*)(m + i);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float pe = (&pv.x)[k];
      float ge = (&gv.x)[k] + args.weight_decay * pe;
      float u;
      if (HAS_MU) {
        float me = args.first ? ge : args.momentum * (&mv.x)[k] + ge;
        (&mv.x)[k] = me;
        u = args.nesterov ? ge + args.momentum * me : me;
      } else {
        u = ge;
      }
      (&pv.x)[k] = pe - args.lr * u;
      if (args.zero_grad) (&gv.x)[k] = 0.f;
    }
    *(float4*)(p + i) = pv;
    if (HAS_MU) *(float4*)(m + i) = mv;
    if (args.zero_grad) *(float4*)(g + i) = gv;
  }
  // scalar tail (chunk length not a multiple of 4)
  long tail_start = base + ((end - base) / 4) * 4;
  for (long i = tail_start + threadIdx.x; i < end; i += blockDim.x) {
    float pe = p[i];
    float ge = g[i] + args.weight_decay * pe;
    float u;
    if (HAS_MU) {
      float me = args.first ? ge : args.momentum * m[i] + ge;
      m[i] = me;
      u = args.nesterov ? ge + args.momentum * me : me;
    } else {
      u = ge;
    }
    p[i] = pe - args.lr * u;
    if (args.zero_grad) g[i] = 0.f;
  }
}

}  // namespace

void multi_tensor_sgd(std::vector<at::Tensor> params,
                      std::vector<at::Tensor> grads,
                      std::vector<at::Tensor> momenta, double lr,
                      double momentum, double weight_decay, bool nesterov,
                      bool first, bool zero_grad) {
  TORCH_CHECK(params.size() == grads.size() &&
              params.size() == momenta.size());
  if (params.empty()) return;
  for (auto& p : params)
    TORCH_CHECK(p.is_cuda() && p.scalar_type() == at::kFloat,
                "multi_tensor_sgd expects fp32 CUDA tensors");
  SGDArgs args{(float)lr, (float)momentum, (float)weight_decay, nesterov,
               first, zero_grad};
  auto stream = at::cuda::getCurrentCUDAStream();
  bool has_mu = momentum != 0.0;
  mt_apply(momenta, &params, &grads, [&](const MTMeta& meta, int blocks, int) {
    if (has_mu)
      mt_sgd_kernel<true><<<blocks, AMD_TPB, 0, stream>>>(meta, args);
    else
      mt_sgd_kernel<false><<<blocks, AMD_TPB, 0, stream>>>(meta, args);
    CHECK_CUDA_OK();
  });
}
