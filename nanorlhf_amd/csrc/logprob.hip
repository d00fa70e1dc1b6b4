// Fused token-logprob + entropy row statistics over the vocab (V ~ 151k),
// and the matching backward dlogits transform.  fp32 accumulation (bf16
// logits over 151k need fp32 lse — SURVEY §7 hard part (c)).  Never
// materializes log-softmax (the reference's memory pain point,
// grpo_trainer.py:548-549,653-656,678-679).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// one 256-thread block per row; online (m, s=sum exp, t=sum exp*l) merge
template <int BLOCK>
__global__ void ce_rowstats_kernel(const short* __restrict__ logits,
                                   const long* __restrict__ labels,
                                   float inv_temp, int V,
                                   float* __restrict__ lp,
                                   float* __restrict__ ent,
                                   float* __restrict__ lse_out) {
  __shared__ float sm[BLOCK / 64], ss[BLOCK / 64], st[BLOCK / 64];
  const long row = blockIdx.x;
  const short* lr = logits + row * (long)V;
  float m = -INFINITY, s = 0.f, t = 0.f;
  const int nvec = V / 8;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 v = *reinterpret_cast<const s16x8*>(lr + i * 8);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float l = bf2f(v[j]) * inv_temp;
      if (l > m) {
        float c = __expf(m - l);
        s = s * c + 1.f;
        t = t * c + l;
        m = l;
      } else {
        float e = __expf(l - m);
        s += e;
        t += e * l;
      }
    }
  }
  // tail (V not divisible by 8)
  for (int i = nvec * 8 + threadIdx.x; i < V; i += BLOCK) {
    float l = bf2f(lr[i]) * inv_temp;
    if (l > m) { float c = __expf(m - l); s = s * c + 1.f; t = t * c + l; m = l; }
    else { float e = __expf(l - m); s += e; t += e * l; }
  }
  // merge across the wave then block: state (m, s, t)
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float m2 = __shfl_xor(m, off), s2 = __shfl_xor(s, off), t2 = __shfl_xor(t, off);
    float mn = fmaxf(m, m2);
    // -inf guard: threads with no elements (V < 8*BLOCK) would otherwise
    // poison the merge with exp(-inf - -inf) = NaN
    float c1 = (m == -INFINITY) ? 0.f : __expf(m - mn);
    float c2 = (m2 == -INFINITY) ? 0.f : __expf(m2 - mn);
    s = s * c1 + s2 * c2;
    t = t * c1 + t2 * c2;
    m = mn;
  }
  if (lane == 0) { sm[wid] = m; ss[wid] = s; st[wid] = t; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = sm[0], S = ss[0], T = st[0];
    for (int w = 1; w < BLOCK / 64; w++) {
      float mn = fmaxf(M, sm[w]);
      float c1 = (M == -INFINITY) ? 0.f : __expf(M - mn);
      float c2 = (sm[w] == -INFINITY) ? 0.f : __expf(sm[w] - mn);
      S = S * c1 + ss[w] * c2;
      T = T * c1 + st[w] * c2;
      M = mn;
    }
    const float lse = M + __logf(S);
    const float label_logit = bf2f(lr[labels[row]]) * inv_temp;
    lp[row] = label_logit - lse;
    // H = lse - E[l] ; E[l] = (T/S) because T = sum exp(l-M)*l
    ent[row] = lse - T / S;
    lse_out[row] = lse;
  }
}

// dlogits (in place over logits, bf16):
//   d_j = g * (onehot_j - softmax_j) * inv_temp   (d logprob/d logit)
__global__ void ce_backward_kernel(short* __restrict__ logits,
                                   const long* __restrict__ labels,
                                   const float* __restrict__ lse,
                                   const float* __restrict__ g,
                                   float inv_temp, long n, int V) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;  // units of 8
  const long total = n * (long)(V / 8);
  if (idx >= total) return;
  const long row = idx / (V / 8);
  const int col0 = (int)(idx % (V / 8)) * 8;
  short* lr = logits + row * (long)V;
  const float L = lse[row];
  const float gr = g[row];
  const long lab = labels[row];
  s16x8 v = *reinterpret_cast<const s16x8*>(lr + col0);
  s16x8 o;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    float l = bf2f(v[j]) * inv_temp;
    float p = __expf(l - L);
    float d = gr * (((col0 + j) == lab ? 1.f : 0.f) - p) * inv_temp;
    o[j] = f2bf(d);
  }
  *reinterpret_cast<s16x8*>(lr + col0) = o;
}

__global__ void ce_backward_tail_kernel(short* __restrict__ logits,
                                        const long* __restrict__ labels,
                                        const float* __restrict__ lse,
                                        const float* __restrict__ g,
                                        float inv_temp, long n, int V, int tail0) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const int ntail = V - tail0;
  if (idx >= n * ntail) return;
  const long row = idx / ntail;
  const int col = tail0 + (int)(idx % ntail);
  short* lr = logits + row * (long)V;
  float l = bf2f(lr[col]) * inv_temp;
  float p = __expf(l - lse[row]);
  lr[col] = f2bf(g[row] * ((col == labels[row] ? 1.f : 0.f) - p) * inv_temp);
}

// ======================================================== host wrappers ==
void ce_rowstats(torch::Tensor logits, torch::Tensor labels, double inv_temp,
                 torch::Tensor lp, torch::Tensor ent, torch::Tensor lse) {
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16 && logits.is_contiguous());
  TORCH_CHECK(labels.scalar_type() == torch::kLong);
  const long n = logits.size(0);
  const int V = logits.size(1);
  if (n == 0) return;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((ce_rowstats_kernel<256>), dim3(n), dim3(256), 0, stream,
                     (const short*)logits.data_ptr(), labels.data_ptr<long>(),
                     (float)inv_temp, V, lp.data_ptr<float>(), ent.data_ptr<float>(),
                     lse.data_ptr<float>());
  HIP_CHECK_LAST();
}

void ce_backward_dlogits(torch::Tensor logits, torch::Tensor labels,
                         torch::Tensor lse, torch::Tensor g, double inv_temp) {
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16 && logits.is_contiguous());
  const long n = logits.size(0);
  const int V = logits.size(1);
  if (n == 0) return;
  const int V8 = (V / 8) * 8;
  auto stream = at::hip::getCurrentHIPStream();
  const long total = n * (long)(V / 8);
  hipLaunchKernelGGL(ce_backward_kernel, dim3((total + 255) / 256), dim3(256), 0, stream,
                     (short*)logits.data_ptr(), labels.data_ptr<long>(),
                     lse.data_ptr<float>(), g.data_ptr<float>(), (float)inv_temp, n, V);
  if (V8 != V) {
    const long ttotal = n * (V - V8);
    hipLaunchKernelGGL(ce_backward_tail_kernel, dim3((ttotal + 255) / 256), dim3(256),
                       0, stream, (short*)logits.data_ptr(), labels.data_ptr<long>(),
                       lse.data_ptr<float>(), g.data_ptr<float>(), (float)inv_temp,
                       n, V, V8);
  }
  HIP_CHECK_LAST();
}
