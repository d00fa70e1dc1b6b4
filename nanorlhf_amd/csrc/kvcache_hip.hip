#include "hip/hip_runtime.h"
// Paged KV-cache ops: append + single-token decode attention.
//
// paged_attn_decode: one 256-thread workgroup (4 waves) per (sequence,
// kv-head).  The G = Hq/Hkv grouped query heads are kept in registers
// (GQA); each wave walks a strided slice of the sequence's tokens, 16-lane
// groups own one token each (a K/V row at D=128 is 256 B = 16 lanes x 16 B
// vector loads), online softmax in fp32.  Token-slot states are merged
// in-register via cross-lane shuffles, then across waves through LDS.
// Memory-bound by design: the KV stream is read exactly once; grid =
// B * Hkv >> 256 CUs keeps the chip full at decode batch sizes.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// ------------------------------------------------------------- kv append
__global__ void kv_append_kernel(const short* __restrict__ k,
                                 const short* __restrict__ v,
                                 const long* __restrict__ slots,
                                 short* __restrict__ kc,
                                 short* __restrict__ vc,
                                 long T, int row_elems /* Hkv*D */) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;  // units of 8 elems
  const int nvec = row_elems / 8;
  if (idx >= T * (long)nvec) return;
  const long t = idx / nvec;
  const int i = (int)(idx % nvec) * 8;
  const long dst = slots[t] * (long)row_elems + i;
  *reinterpret_cast<s16x8*>(kc + dst) =
      *reinterpret_cast<const s16x8*>(k + t * (long)row_elems + i);
  *reinterpret_cast<s16x8*>(vc + dst) =
      *reinterpret_cast<const s16x8*>(v + t * (long)row_elems + i);
}

void kv_append(torch::Tensor k, torch::Tensor v, torch::Tensor slots,
               torch::Tensor k_cache, torch::Tensor v_cache) {
  TORCH_CHECK(k.scalar_type() == torch::kBFloat16 && k.is_contiguous());
  TORCH_CHECK(slots.scalar_type() == torch::kLong);
  const long T = k.size(0);
  if (T == 0) return;
  const int row_elems = k.size(1) * k.size(2);
  TORCH_CHECK(row_elems % 8 == 0);
  const long total = T * (row_elems / 8);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(kv_append_kernel, dim3((total + 255) / 256), dim3(256), 0, stream,
                     (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                     slots.data_ptr<long>(), (short*)k_cache.data_ptr(),
                     (short*)v_cache.data_ptr(), T, row_elems);
  HIP_CHECK_LAST();
}

// ------------------------------------------------------- decode attention
template <int VEC>
DEVINL void load_bf16_slice(const short* p, float* out) {
  if constexpr (VEC == 8) {
    s16x8 x = *reinterpret_cast<const s16x8*>(p);
#pragma unroll
    for (int j = 0; j < 8; j++) out[j] = bf2f(x[j]);
  } else if constexpr (VEC == 4) {
    s16x4 x = *reinterpret_cast<const s16x4*>(p);
#pragma unroll
    for (int j = 0; j < 4; j++) out[j] = bf2f(x[j]);
  } else {
#pragma unroll
    for (int j = 0; j < VEC; j++) out[j] = bf2f(p[j]);
  }
}

template <int D, int MAXG>
__global__ void paged_decode_kernel(const short* __restrict__ q,    // [B, Hq, D]
                                    const short* __restrict__ kc,   // [P, ps, Hkv, D]
                                    const short* __restrict__ vc,
                                    const int* __restrict__ bt,     // [B, maxP]
                                    const int* __restrict__ seq_lens,
                                    short* __restrict__ out,        // [B, Hq, D]
                                    int Hq, int Hkv, int page_size, int maxP,
                                    int G, float scale) {
  const int b = blockIdx.x / Hkv;
  const int kvh = blockIdx.x % Hkv;
  const int L = seq_lens[b];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;            // 4 waves
  const int tok_in_wave = lane >> 4;           // 4 token slots per wave
  const int sub = lane & 15;                   // 16 lanes per token
  constexpr int VEC = D / 16;                  // bf16 per lane slice

  float qreg[MAXG][VEC];
  const int h0 = kvh * G;
  for (int g = 0; g < G; g++)
    load_bf16_slice<VEC>(q + ((long)b * Hq + h0 + g) * D + sub * VEC, qreg[g]);

  float m[MAXG], s[MAXG], acc[MAXG][VEC];
  for (int g = 0; g < G; g++) {
    m[g] = -INFINITY; s[g] = 0.f;
#pragma unroll
    for (int j = 0; j < VEC; j++) acc[g][j] = 0.f;
  }

  const long kv_row = (long)Hkv * D;
  for (int t = wid * 4 + tok_in_wave; t < L; t += 16) {
    const int page = bt[b * maxP + t / page_size];
    const long base = ((long)page * page_size + (t % page_size)) * kv_row
                      + (long)kvh * D + sub * VEC;
    float kf[VEC], vf[VEC];
    load_bf16_slice<VEC>(kc + base, kf);
    load_bf16_slice<VEC>(vc + base, vf);
    for (int g = 0; g < G; g++) {
      float dot = 0.f;
#pragma unroll
      for (int j = 0; j < VEC; j++) dot += qreg[g][j] * kf[j];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) dot += __shfl_xor(dot, off);
      const float score = dot * scale;
      if (score > m[g]) {
        const float c = __expf(m[g] - score);
        s[g] = s[g] * c + 1.f;
#pragma unroll
        for (int j = 0; j < VEC; j++) acc[g][j] = acc[g][j] * c + vf[j];
        m[g] = score;
      } else {
        const float e = __expf(score - m[g]);
        s[g] += e;
#pragma unroll
        for (int j = 0; j < VEC; j++) acc[g][j] += e * vf[j];
      }
    }
  }

  // merge the wave's 4 token-slot states in-register: lanes with equal `sub`
  // hold the same d-slice, so shfl_xor(16/32) pairs matching slices.
#pragma unroll
  for (int off = 16; off <= 32; off <<= 1) {
    for (int g = 0; g < G; g++) {
      const float m2 = __shfl_xor(m[g], off);
      const float s2 = __shfl_xor(s[g], off);
      float a2[VEC];
#pragma unroll
      for (int j = 0; j < VEC; j++) a2[j] = __shfl_xor(acc[g][j], off);
      const float mn = fmaxf(m[g], m2);
      const float c1 = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - mn);
      const float c2 = (m2 == -INFINITY) ? 0.f : __expf(m2 - mn);
      s[g] = s[g] * c1 + s2 * c2;
#pragma unroll
      for (int j = 0; j < VEC; j++) acc[g][j] = acc[g][j] * c1 + a2[j] * c2;
      m[g] = mn;
    }
  }

  // cross-wave merge through LDS (4 states, one per wave)
  __shared__ float lds_m[4][MAXG];
  __shared__ float lds_s[4][MAXG];
  __shared__ float lds_acc[4][MAXG][D];
  if (lane < 16) {
    for (int g = 0; g < G; g++) {
      if (sub == 0) { lds_m[wid][g] = m[g]; lds_s[wid][g] = s[g]; }
#pragma unroll
      for (int j = 0; j < VEC; j++) lds_acc[wid][g][sub * VEC + j] = acc[g][j];
    }
  }
  __syncthreads();
  if (wid == 0) {
    constexpr int OV = (D >= 64) ? D / 64 : 1;
    for (int g = 0; g < G; g++) {
      float M = lds_m[0][g];
      for (int st = 1; st < 4; st++) M = fmaxf(M, lds_m[st][g]);
      float S = 0.f;
      float o[OV];
#pragma unroll
      for (int j = 0; j < OV; j++) o[j] = 0.f;
      for (int st = 0; st < 4; st++) {
        const float c = (lds_m[st][g] == -INFINITY) ? 0.f : __expf(lds_m[st][g] - M);
        S += lds_s[st][g] * c;
        if (lane * OV < D) {
#pragma unroll
          for (int j = 0; j < OV; j++) o[j] += lds_acc[st][g][lane * OV + j] * c;
        }
      }
      const float invS = S > 0.f ? 1.f / S : 0.f;
      if (lane * OV < D) {
        short* op = out + ((long)b * Hq + h0 + g) * D + lane * OV;
#pragma unroll
        for (int j = 0; j < OV; j++) op[j] = f2bf(o[j] * invS);
      }
    }
  }
}

torch::Tensor paged_attn_decode(torch::Tensor q, torch::Tensor k_cache,
                                torch::Tensor v_cache, torch::Tensor block_tables,
                                torch::Tensor seq_lens, double scale) {
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt && seq_lens.scalar_type() == torch::kInt);
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int page_size = k_cache.size(1);
  const int Hkv = k_cache.size(2);
  const int maxP = block_tables.size(1);
  const int G = Hq / Hkv;
  TORCH_CHECK(Hq % Hkv == 0 && G <= 8, "GQA group size must be <= 8");
  auto out = torch::empty_like(q);
  if (B == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(B * Hkv), block(256);
#define LAUNCH_D(DD)                                                            \
  hipLaunchKernelGGL((paged_decode_kernel<DD, 8>), grid, block, 0, stream,      \
                     (const short*)q.data_ptr(), (const short*)k_cache.data_ptr(), \
                     (const short*)v_cache.data_ptr(), block_tables.data_ptr<int>(), \
                     seq_lens.data_ptr<int>(), (short*)out.data_ptr(),          \
                     Hq, Hkv, page_size, maxP, G, (float)scale)
  if (D == 128) { LAUNCH_D(128); }
  else if (D == 64) { LAUNCH_D(64); }
  else if (D == 32) { LAUNCH_D(32); }
  else { TORCH_CHECK(false, "unsupported head_dim ", D); }
#undef LAUNCH_D
  HIP_CHECK_LAST();
  return out;
}
