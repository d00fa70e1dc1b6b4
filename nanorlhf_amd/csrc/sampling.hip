// Temperature / top-p token sampling over [B, V] bf16 logits.
// One 256-thread workgroup per row, four vectorized passes (s16x8 16-B
// loads — the scalar version measured 2.7 ms/row-batch, ~6x off BW):
//   1. fused online (max, exp-sum) + argmax
//   2. coarse LDS histogram of prob mass over u = (l - m)/T (1024 bins)
//   3. one refinement histogram inside the threshold bin (no 151k sort)
//   4. per-thread kept-mass + one block scan + inverse-CDF walk by the
//      owning thread (counter-hash uniform: deterministic in (seed, step,
//      row), replay-safe via a device-side step counter for hipGraphs).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

#define SBLOCK 256
#define NBINS 1024
#define URANGE 32.0f  // histogram covers u in [-URANGE, 0]

DEVINL float warp_incl_scan(float v, int lane) {
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) {
    float x = __shfl_up(v, off);
    if (lane >= off) v += x;
  }
  return v;
}

__global__ void sample_topp_kernel(const short* __restrict__ logits,
                                   long* __restrict__ out,
                                   float* __restrict__ out_lp,  // nullable
                                   int V, float temperature, float top_p,
                                   unsigned long long seed,
                                   unsigned long long step_imm,
                                   const long* __restrict__ step_ptr) {
  const unsigned long long step = step_ptr ? (unsigned long long)*step_ptr : step_imm;
  __shared__ float sm[SBLOCK / 64];
  __shared__ float wm_[SBLOCK / 64];
  __shared__ int wa_[SBLOCK / 64];
  __shared__ int argmax_sh;
  __shared__ int found_sh;
  __shared__ float wsum[SBLOCK / 64];
  const long row = blockIdx.x;
  const short* lr = logits + row * (long)V;
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int nvec = V / 8;

  // ---- pass 1: plain row max + argmax ----------------------------------
  // (the round-1 online (m,s) update carried m,s ACROSS loop iterations —
  // a serial exp/fma dependency chain per wave; plain fmax has none, and
  // the exp-sum folds into the histogram pass below, which also drops one
  // full read of the row)
  float lmax = -INFINITY;
  int am = 0;
  const float invT = temperature > 0.f ? 1.f / temperature : 1.f;
  for (int i = threadIdx.x; i < nvec; i += SBLOCK) {
    s16x8 v = *reinterpret_cast<const s16x8*>(lr + i * 8);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const float l = bf2f(v[j]);
      if (l > lmax) { lmax = l; am = i * 8 + j; }
    }
  }
  for (int i = nvec * 8 + threadIdx.x; i < V; i += SBLOCK) {
    const float l = bf2f(lr[i]);
    if (l > lmax) { lmax = l; am = i; }
  }
  {  // merge argmax across the block
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float mx2 = __shfl_xor(lmax, off);
      int a2 = __shfl_xor(am, off);
      if (mx2 > lmax || (mx2 == lmax && a2 < am)) { lmax = mx2; am = a2; }
    }
    if (lane == 0) { wm_[wid] = lmax; wa_[wid] = am; }
    __syncthreads();
    if (threadIdx.x == 0) {
      float MX = wm_[0];
      int A = wa_[0];
      for (int w = 1; w < SBLOCK / 64; w++)
        if (wm_[w] > MX || (wm_[w] == MX && wa_[w] < A)) { MX = wm_[w]; A = wa_[w]; }
      wm_[0] = MX; argmax_sh = A;
    }
    __syncthreads();
  }
  const float M = wm_[0] * invT;        // max of temperature-scaled logits
  if (temperature == 0.f) {
    // greedy: one extra pass only to report the T=1-softmax logprob
    if (out_lp) {
      float se = 0.f;
      const float M1 = wm_[0];
      for (int i = threadIdx.x; i < nvec; i += SBLOCK) {
        s16x8 v = *reinterpret_cast<const s16x8*>(lr + i * 8);
#pragma unroll
        for (int j = 0; j < 8; j++) se += __expf(bf2f(v[j]) - M1);
      }
      for (int i = nvec * 8 + threadIdx.x; i < V; i += SBLOCK)
        se += __expf(bf2f(lr[i]) - M1);
      se = block_sum<SBLOCK>(se, sm);
      if (threadIdx.x == 0) {
        out[row] = argmax_sh;
        out_lp[row] = bf2f(lr[argmax_sh]) - (M1 + __logf(se));
      }
    } else if (threadIdx.x == 0) {
      out[row] = argmax_sh;
    }
    return;
  }

  // ---- pass 2: coarse histogram of prob mass (wave-privatized: 4 waves
  // hammering one 1024-bin histogram serialize on LDS atomics; a private
  // histogram per wave merged once removes the cross-wave conflicts) ----
  // Threshold search WITHOUT LDS atomics.  Ablation (tools/topp_ablate.hip)
  // showed the 1024-bin atomicAdd histogram was 75% of the kernel — LDS
  // f32 atomics are throughput-bound regardless of conflict spreading
  // (lane-keyed slots measured no better).  Instead: a 32-bin REGISTER
  // histogram per thread (no atomics), block-reduced through LDS once,
  // then the boundary bin is refined by re-binning only its interior on
  // additional streaming passes (each pass is pure reads + VALU at ~6 TB/s).
  __shared__ float Hist32[32 * (SBLOCK / 64)];
  __shared__ float S_sh, lo_sh, span_sh, above_sh;
  __shared__ int done_sh;
  float sacc = 0.f;
  {
    float hb[32];
#pragma unroll
    for (int b = 0; b < 32; b++) hb[b] = 0.f;
    for (int i = threadIdx.x; i < nvec; i += SBLOCK) {
      s16x8 v = *reinterpret_cast<const s16x8*>(lr + i * 8);
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const float u = bf2f(v[j]) * invT - M;
        const float e = __expf(u);
        sacc += e;
        int b = (int)(u + URANGE);
        b = max(0, min(31, b));
        hb[b] += e;
      }
    }
    for (int i = nvec * 8 + threadIdx.x; i < V; i += SBLOCK) {
      const float u = bf2f(lr[i]) * invT - M;
      const float e = __expf(u);
      sacc += e;
      int b = (int)(u + URANGE);
      b = max(0, min(31, b));
      hb[b] += e;
    }
    // wave-reduce each bin, then one slot per (bin, wave) in LDS
#pragma unroll
    for (int b = 0; b < 32; b++) {
      float x = hb[b];
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off);
      if (lane == 0) Hist32[b * (SBLOCK / 64) + wid] = x;
    }
    sacc = block_sum<SBLOCK>(sacc, sm);
    if (threadIdx.x == 0) S_sh = sacc;
    __syncthreads();
  }
  const float S = S_sh;
  const float ptarget = top_p * S;
  // coarse bin select (bins are 1 nat wide, u in [-URANGE, 0])
  if (threadIdx.x == 0) {
    float acc = 0.f;
    int bstar = 0;
    float above = 0.f;
    for (int b = 31; b >= 0; b--) {
      float h = 0.f;
      for (int w = 0; w < SBLOCK / 64; w++) h += Hist32[b * (SBLOCK / 64) + w];
      const float nacc = acc + h;
      if (nacc >= ptarget || b == 0) { bstar = b; above = acc; break; }
      acc = nacc;
    }
    lo_sh = (float)bstar - URANGE;
    span_sh = 1.0f;
    above_sh = above;
    done_sh = 0;
    found_sh = -1;
  }
  __syncthreads();
  // refine the boundary bin: re-bin its interior into 32 sub-bins per pass
  // until the bin is narrow (1/1024 nat) or holds negligible mass
  for (int it = 0; it < 2 && !done_sh; it++) {
    const float lo = lo_sh, span = span_sh;
    float hb[32];
#pragma unroll
    for (int b = 0; b < 32; b++) hb[b] = 0.f;
    const float sub = 32.0f / span;
    for (int i = threadIdx.x; i < nvec; i += SBLOCK) {
      s16x8 v = *reinterpret_cast<const s16x8*>(lr + i * 8);
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const float u = bf2f(v[j]) * invT - M;
        if (u >= lo && u < lo + span) {
          int b = (int)((u - lo) * sub);
          b = max(0, min(31, b));
          hb[b] += __expf(u);
        }
      }
    }
    for (int i = nvec * 8 + threadIdx.x; i < V; i += SBLOCK) {
      const float u = bf2f(lr[i]) * invT - M;
      if (u >= lo && u < lo + span) {
        int b = (int)((u - lo) * sub);
        b = max(0, min(31, b));
        hb[b] += __expf(u);
      }
    }
#pragma unroll
    for (int b = 0; b < 32; b++) {
      float x = hb[b];
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off);
      if (lane == 0) Hist32[b * (SBLOCK / 64) + wid] = x;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      float acc = above_sh;
      int bstar = 0;
      float above = above_sh;
      for (int b = 31; b >= 0; b--) {
        float h = 0.f;
        for (int w = 0; w < SBLOCK / 64; w++) h += Hist32[b * (SBLOCK / 64) + w];
        const float nacc = acc + h;
        if (nacc >= ptarget || b == 0) { bstar = b; above = acc; break; }
        acc = nacc;
      }
      lo_sh = lo + (float)bstar * (span / 32.0f);
      span_sh = span / 32.0f;
      above_sh = above;
      // stop when the boundary sliver is negligible (over-keeping the
      // whole sliver changes the nucleus by < 2% of S)
      float bmass = 0.f;
      for (int w = 0; w < SBLOCK / 64; w++)
        bmass += Hist32[bstar * (SBLOCK / 64) + w];
      if (bmass <= 0.02f * S) done_sh = 1;
    }
    __syncthreads();
  }
  const float u_thresh = lo_sh;

  // ---- pass 4: per-thread kept mass + block scan + owner walk ----------
  float own = 0.f;
  for (int i = threadIdx.x; i < nvec; i += SBLOCK) {
    s16x8 v = *reinterpret_cast<const s16x8*>(lr + i * 8);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const float u = bf2f(v[j]) * invT - M;
      if (u >= u_thresh) own += __expf(u);
    }
  }
  for (int i = nvec * 8 + threadIdx.x; i < V; i += SBLOCK) {
    const float u = bf2f(lr[i]) * invT - M;
    if (u >= u_thresh) own += __expf(u);
  }
  float incl = warp_incl_scan(own, lane);
  if (lane == 63) wsum[wid] = incl;
  __syncthreads();
  float wbase = 0.f;
  for (int w = 0; w < wid; w++) wbase += wsum[w];
  float total = 0.f;
  for (int w = 0; w < SBLOCK / 64; w++) total += wsum[w];
  const float excl = wbase + incl - own;
  const float r = hash_uniform(seed, step, (unsigned long long)row);
  const float target = r * total;
  if (own > 0.f && target >= excl && target < excl + own) {
    // this thread owns the crossing: re-walk its strided chunks (the
    // sampled category order is thread-strided — a fixed permutation,
    // which leaves the sampled distribution exactly the kept softmax)
    // scan ALL chunks with no data-dependent loop exit: an early-out on
    // `found` makes every iteration's branch wait on the previous load
    // (~594 serial HBM latencies); unconditional scan lets the loads
    // pipeline and costs only the owner thread's single stride
    float acc = excl;
    int found = -1;
    for (int i = threadIdx.x; i < nvec; i += SBLOCK) {
      s16x8 v = *reinterpret_cast<const s16x8*>(lr + i * 8);
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const float u = bf2f(v[j]) * invT - M;
        if (u >= u_thresh) {
          acc += __expf(u);
          if (target < acc && found < 0) found = i * 8 + j;
        }
      }
    }
    for (int i = nvec * 8 + threadIdx.x; i < V; i += SBLOCK) {
      const float u = bf2f(lr[i]) * invT - M;
      if (u >= u_thresh) {
        acc += __expf(u);
        if (target < acc && found < 0) found = i;
      }
    }
    if (found >= 0) atomicCAS(&found_sh, -1, found);
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    const int f = (found_sh >= 0) ? found_sh : argmax_sh;  // rounding fallback
    out[row] = f;
    // logprob of the chosen token under the temperature-scaled softmax
    // (matches the scoring pass's logits/temperature quirk, grpo_trainer.py:547)
    if (out_lp) out_lp[row] = bf2f(lr[f]) * invT - (M + __logf(S));
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
                     (const short*)logits.data_ptr(), out.data_ptr<long>(), nullptr,
                     V, (float)temperature, (float)top_p,
                     (unsigned long long)seed, (unsigned long long)step, nullptr);
  HIP_CHECK_LAST();
  return out;
}

std::vector<torch::Tensor> sample_topp_dev(torch::Tensor logits, double temperature,
                                           double top_p, long seed, torch::Tensor step) {
  // step: int64 [1] device tensor, read inside the kernel — safe under
  // hipGraph capture/replay (the immediate-arg form would freeze the step).
  // Also returns the chosen token's logprob (vLLM logprobs parity).
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(step.scalar_type() == torch::kLong && step.is_cuda());
  const long B = logits.size(0);
  const int V = logits.size(1);
  auto out = torch::empty({B}, logits.options().dtype(torch::kLong));
  auto lp = torch::empty({B}, logits.options().dtype(torch::kFloat32));
  if (B == 0) return {out, lp};
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sample_topp_kernel, dim3(B), dim3(SBLOCK), 0, stream,
                     (const short*)logits.data_ptr(), out.data_ptr<long>(),
                     lp.data_ptr<float>(), V,
                     (float)temperature, (float)top_p,
                     (unsigned long long)seed, 0ull, step.data_ptr<long>());
  HIP_CHECK_LAST();
  return {out, lp};
}
