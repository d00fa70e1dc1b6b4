#include "hip/hip_runtime.h"
// Fused masked-moments + whitening (trl masked_whiten,
// GRPO/grpo_trainer.py:607,619).  fp32 advantage tensors [N] with a 0/1
// mask: pass 1 reduces per-block (sum, sumsq, count) partials; the host
// combines the tiny partial array; pass 2 applies (x-mean)*invstd (+mean).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

template <int BLOCK>
__global__ void masked_moments_kernel(const float* __restrict__ v,
                                      const float* __restrict__ mask,
                                      float* __restrict__ partials,  // [nblk, 3]
                                      long n) {
  __shared__ float scratch[BLOCK / 64];
  float s = 0.f, ss = 0.f, c = 0.f;
  for (long i = (long)blockIdx.x * BLOCK + threadIdx.x; i < n;
       i += (long)gridDim.x * BLOCK) {
    const float m = mask[i];
    const float x = v[i] * m;
    s += x;
    ss += x * x;
    c += m;
  }
  s = block_sum<BLOCK>(s, scratch);
  ss = block_sum<BLOCK>(ss, scratch);
  c = block_sum<BLOCK>(c, scratch);
  if (threadIdx.x == 0) {
    partials[blockIdx.x * 3 + 0] = s;
    partials[blockIdx.x * 3 + 1] = ss;
    partials[blockIdx.x * 3 + 2] = c;
  }
}

__global__ void whiten_apply_kernel(const float* __restrict__ v,
                                    float* __restrict__ out,
                                    float mean, float invstd, float shift,
                                    long n) {
  // unmasked apply (trl semantics): callers re-apply the mask themselves
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  out[i] = (v[i] - mean) * invstd + shift;
}

std::vector<torch::Tensor> masked_moments(torch::Tensor v, torch::Tensor mask) {
  TORCH_CHECK(v.scalar_type() == torch::kFloat32 && v.is_contiguous());
  TORCH_CHECK(mask.scalar_type() == torch::kFloat32 && mask.is_contiguous());
  const long n = v.numel();
  const int nblk = (int)std::min<long>((n + 255) / 256, 1024);
  auto partials = torch::zeros({std::max(nblk, 1), 3}, v.options());
  if (n == 0) return {partials};
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((masked_moments_kernel<256>), dim3(nblk), dim3(256), 0, stream,
                     v.data_ptr<float>(), mask.data_ptr<float>(),
                     partials.data_ptr<float>(), n);
  HIP_CHECK_LAST();
  return {partials};
}

torch::Tensor whiten_apply(torch::Tensor v, double mean, double invstd, double shift) {
  TORCH_CHECK(v.scalar_type() == torch::kFloat32 && v.is_contiguous());
  const long n = v.numel();
  auto out = torch::empty_like(v);
  if (n == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(whiten_apply_kernel, dim3((n + 255) / 256), dim3(256), 0, stream,
                     v.data_ptr<float>(), out.data_ptr<float>(),
                     (float)mean, (float)invstd, (float)shift, n);
  HIP_CHECK_LAST();
  return out;
}
