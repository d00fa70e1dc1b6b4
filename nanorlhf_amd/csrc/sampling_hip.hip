#include "hip/hip_runtime.h"
// Temperature / top-p token sampling over [B, V] bf16 logits.
// One 256-thread workgroup per row:
//   pass 1: online max + exp-sum (+ argmax for greedy)
//   pass 2: LDS histogram over u = (l - m)/T to locate the top-p threshold
//           (coarse 1024 bins + one refinement level — no 151k sort)
//   pass 3: inverse-CDF selection over the kept set with a counter-hash
//           uniform (deterministic in (seed, step, row)).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

#define SBLOCK 256
#define NBINS 1024
#define URANGE 32.0f  // histogram covers u in [-URANGE, 0]

__global__ void sample_topp_kernel(const short* __restrict__ logits,
                                   long* __restrict__ out,
                                   int V, float temperature, float top_p,
                                   unsigned long long seed,
                                   unsigned long long step_imm,
                                   const long* __restrict__ step_ptr) {
  const unsigned long long step = step_ptr ? (unsigned long long)*step_ptr : step_imm;
  __shared__ float red[SBLOCK / 64];
  __shared__ float hist[NBINS];
  __shared__ int argmax_sh;
  __shared__ float cdf_carry;
  __shared__ int found_sh;
  const long row = blockIdx.x;
  const short* lr = logits + row * (long)V;

  // ---- pass 1: max (+argmax) --------------------------------------------
  float m = -INFINITY;
  int am = 0;
  for (int i = threadIdx.x; i < V; i += SBLOCK) {
    float l = bf2f(lr[i]);
    if (l > m) { m = l; am = i; }
  }
  {  // block arg-max
    float mv = m;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float m2 = __shfl_xor(mv, off);
      int a2 = __shfl_xor(am, off);
      if (m2 > mv || (m2 == mv && a2 < am)) { mv = m2; am = a2; }
    }
    const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
    __shared__ float wm[SBLOCK / 64];
    __shared__ int wa[SBLOCK / 64];
    if (lane == 0) { wm[wid] = mv; wa[wid] = am; }
    __syncthreads();
    if (threadIdx.x == 0) {
      float M = wm[0]; int A = wa[0];
      for (int w = 1; w < SBLOCK / 64; w++)
        if (wm[w] > M || (wm[w] == M && wa[w] < A)) { M = wm[w]; A = wa[w]; }
      red[0] = M; argmax_sh = A;
    }
    __syncthreads();
    m = red[0];
  }
  if (temperature == 0.f) {
    if (threadIdx.x == 0) out[row] = argmax_sh;
    return;
  }
  const float invT = 1.f / temperature;

  // ---- exp-sum -----------------------------------------------------------
  float s = 0.f;
  for (int i = threadIdx.x; i < V; i += SBLOCK)
    s += __expf((bf2f(lr[i]) - m) * invT);
  s = block_sum<SBLOCK>(s, red);
  const float invS = 1.f / s;

  // ---- pass 2: histogram of prob mass by u ------------------------------
  // bin(u) = clamp((u + URANGE) / URANGE * NBINS)
  for (int i = threadIdx.x; i < NBINS; i += SBLOCK) hist[i] = 0.f;
  __syncthreads();
  for (int i = threadIdx.x; i < V; i += SBLOCK) {
    float u = (bf2f(lr[i]) - m) * invT;
    int b = (int)((u + URANGE) * (NBINS / URANGE));
    b = max(0, min(NBINS - 1, b));
    atomicAdd(&hist[b], __expf(u) * invS);
  }
  __syncthreads();
  // serial scan from the top bin (1024 iterations by thread 0 — small)
  __shared__ float u_thresh_sh;
  __shared__ float mass_above_sh;
  __shared__ int bin_star_sh;
  if (threadIdx.x == 0) {
    float acc = 0.f;
    int bstar = 0;
    for (int b = NBINS - 1; b >= 0; b--) {
      float nacc = acc + hist[b];
      if (nacc >= top_p || b == 0) { bstar = b; mass_above_sh = acc; break; }
      acc = nacc;
    }
    bin_star_sh = bstar;
  }
  __syncthreads();
  const int bstar = bin_star_sh;
  const float bin_lo = (float)bstar * (URANGE / NBINS) - URANGE;
  const float bin_hi = bin_lo + (URANGE / NBINS);
  // ---- refinement: sub-histogram inside bin* (membership decided by the
  // SAME coarse binning as pass 2, so boundary values like u == 0 land in
  // the clamped top bin consistently) -----------------------------------
  for (int i = threadIdx.x; i < NBINS; i += SBLOCK) hist[i] = 0.f;
  __syncthreads();
  const float sub_scale = NBINS / (bin_hi - bin_lo);
  for (int i = threadIdx.x; i < V; i += SBLOCK) {
    float u = (bf2f(lr[i]) - m) * invT;
    int cb = (int)((u + URANGE) * (NBINS / URANGE));
    cb = max(0, min(NBINS - 1, cb));
    if (cb == bstar) {
      int b = (int)((u - bin_lo) * sub_scale);
      b = max(0, min(NBINS - 1, b));
      atomicAdd(&hist[b], __expf(u) * invS);
    }
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float acc = mass_above_sh;
    float thresh = bin_lo;
    for (int b = NBINS - 1; b >= 0; b--) {
      acc += hist[b];
      if (acc >= top_p || b == 0) {
        thresh = bin_lo + (float)b * (bin_hi - bin_lo) / NBINS;
        mass_above_sh = acc;  // kept mass (>= top_p)
        break;
      }
    }
    u_thresh_sh = thresh;
  }
  __syncthreads();
  float u_thresh = u_thresh_sh;
  float kept_mass = mass_above_sh;
  if (kept_mass < 1e-9f) {  // degenerate: keep everything
    u_thresh = -2.f * URANGE;
    kept_mass = 1.f;
  }

  // ---- pass 3: inverse-CDF over kept tokens (index order) ---------------
  const float r = hash_uniform(seed, step, (unsigned long long)row);
  const float target = r * kept_mass;
  if (threadIdx.x == 0) { cdf_carry = 0.f; found_sh = -1; }
  __syncthreads();
  const int TILE = SBLOCK * 8;
  for (int base = 0; base < V && found_sh < 0; base += TILE) {
    // each thread accumulates 8 strided-contig elems: [base + tid*8, +8)
    float local[8];
    float lsum = 0.f;
    const int i0 = base + threadIdx.x * 8;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const int i = i0 + j;
      float p = 0.f;
      if (i < V) {
        float u = (bf2f(lr[i]) - m) * invT;
        if (u >= u_thresh) p = __expf(u) * invS;
      }
      local[j] = p;
      lsum += p;
    }
    // exclusive prefix of lsum across the block
    float wpre = lsum;
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      float x = __shfl_up(wpre, off);
      if ((threadIdx.x & 63) >= off) wpre += x;
    }
    const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
    __shared__ float wsum[SBLOCK / 64];
    if (lane == 63) wsum[wid] = wpre;
    __syncthreads();
    float wbase = 0.f;
    for (int w = 0; w < wid; w++) wbase += wsum[w];
    float excl = cdf_carry + wbase + wpre - lsum;  // exclusive prefix for this thread
    // does the target fall inside this thread's 8 elements?
    if (target >= excl && target < excl + lsum) {
      float acc = excl;
#pragma unroll
      for (int j = 0; j < 8; j++) {
        acc += local[j];
        if (target < acc) { atomicCAS(&found_sh, -1, i0 + j); break; }
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      float tile_total = 0.f;
      for (int w = 0; w < SBLOCK / 64; w++) tile_total += wsum[w];
      cdf_carry += tile_total;
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    int f = found_sh;
    out[row] = (f >= 0) ? f : argmax_sh;  // numeric fallback: argmax
  }
}

torch::Tensor sample_topp(torch::Tensor logits, double temperature, double top_p,
                          long seed, long step) {
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16, "sample_topp expects bf16 logits");
  const long B = logits.size(0);
  const int V = logits.size(1);
  auto out = torch::empty({B}, logits.options().dtype(torch::kLong));
  if (B == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sample_topp_kernel, dim3(B), dim3(SBLOCK), 0, stream,
                     (const short*)logits.data_ptr(), out.data_ptr<long>(), V,
                     (float)temperature, (float)top_p,
                     (unsigned long long)seed, (unsigned long long)step, nullptr);
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor sample_topp_dev(torch::Tensor logits, double temperature, double top_p,
                              long seed, torch::Tensor step) {
  // step: int64 [1] device tensor, read inside the kernel — safe under
  // hipGraph capture/replay (the immediate-arg form would freeze the step)
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(step.scalar_type() == torch::kLong && step.is_cuda());
  const long B = logits.size(0);
  const int V = logits.size(1);
  auto out = torch::empty({B}, logits.options().dtype(torch::kLong));
  if (B == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sample_topp_kernel, dim3(B), dim3(SBLOCK), 0, stream,
                     (const short*)logits.data_ptr(), out.data_ptr<long>(), V,
                     (float)temperature, (float)top_p,
                     (unsigned long long)seed, 0ull, step.data_ptr<long>());
  HIP_CHECK_LAST();
  return out;
}
