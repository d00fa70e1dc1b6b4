// Fused AdamW step: one pass over (p, g, m, v) with fp32 state, decoupled
// weight decay and bias correction; bf16 or fp32 params.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

template <bool BF16>
__global__ void adamw_kernel(void* __restrict__ p_,
                             const void* __restrict__ g_,
                             float* __restrict__ m,
                             float* __restrict__ v,
                             long n, float lr, float b1, float b2, float eps,
                             float wd, float bc1, float bc2) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float p, g;
    if (BF16) {
      p = bf2f(((const short*)p_)[i]);
      g = bf2f(((const short*)g_)[i]);
    } else {
      p = ((const float*)p_)[i];
      g = ((const float*)g_)[i];
    }
    float mi = m[i] = b1 * m[i] + (1.f - b1) * g;
    float vi = v[i] = b2 * v[i] + (1.f - b2) * g * g;
    float denom = sqrtf(vi / bc2) + eps;
    p = p * (1.f - lr * wd) - lr * (mi / bc1) / denom;
    if (BF16) ((short*)p_)[i] = f2bf(p);
    else ((float*)p_)[i] = p;
  }
}

void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m, torch::Tensor v,
                double lr, double b1, double b2, double eps, double wd, long step) {
  TORCH_CHECK(p.is_contiguous() && m.is_contiguous() && v.is_contiguous());
  TORCH_CHECK(m.scalar_type() == torch::kFloat32 && v.scalar_type() == torch::kFloat32);
  auto gc = g.contiguous();
  const long n = p.numel();
  const float bc1 = 1.f - powf((float)b1, (float)step);
  const float bc2 = 1.f - powf((float)b2, (float)step);
  const int block = 256;
  const long grid = std::min<long>((n + block - 1) / block, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  if (p.scalar_type() == torch::kBFloat16) {
    TORCH_CHECK(gc.scalar_type() == torch::kBFloat16);
    hipLaunchKernelGGL((adamw_kernel<true>), dim3(grid), dim3(block), 0, stream,
                       p.data_ptr(), gc.data_ptr(), m.data_ptr<float>(), v.data_ptr<float>(),
                       n, (float)lr, (float)b1, (float)b2, (float)eps, (float)wd, bc1, bc2);
  } else {
    TORCH_CHECK(p.scalar_type() == torch::kFloat32 && gc.scalar_type() == torch::kFloat32);
    hipLaunchKernelGGL((adamw_kernel<false>), dim3(grid), dim3(block), 0, stream,
                       p.data_ptr(), gc.data_ptr(), m.data_ptr<float>(), v.data_ptr<float>(),
                       n, (float)lr, (float)b1, (float)b2, (float)eps, (float)wd, bc1, bc2);
  }
  HIP_CHECK_LAST();
}
