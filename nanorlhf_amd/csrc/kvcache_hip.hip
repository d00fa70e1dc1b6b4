#include "hip/hip_runtime.h"
// Paged KV-cache ops: append + single-token decode attention (MFMA).
//
// Layouts: K cache [P, ps, Hkv, D] (token-major rows); V cache d-major per
// page, [P, Hkv, D, ps], so the decode PV MFMA B-fragment (contraction over
// tokens) is a contiguous 16-B load.  page_size MUST be 16 (one MFMA
// half-K per page half).
//
// paged_attn_decode: one 256-thread workgroup (4 waves) per (sequence,
// kv-head).  The G <= 16 grouped query heads form the 16-row M dimension of
// v_mfma_f32_16x16x32_bf16 tiles; each wave walks 32-token KV tiles with
// stride 4, doing QK^T and PV on matrix cores with fp32 online softmax
// (the first rocprof profile showed the scalar-dot version VALU-bound at
// ~0.3 TB/s — MFMA moves the arithmetic off the VALU so the kernel can run
// at the KV stream's memory bound).  Cross-wave states merge through LDS.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8k __attribute__((ext_vector_type(8)));

DEVINL f32x4 mfma16k(bf16x8k a, bf16x8k b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// ------------------------------------------------------------- kv append
__global__ void kv_append_kernel(const short* __restrict__ k,
                                 const short* __restrict__ v,
                                 const long* __restrict__ slots,
                                 short* __restrict__ kc,
                                 short* __restrict__ vc,
                                 long T, int Hkv, int D, int ps) {
  const int row_elems = Hkv * D;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;  // units of 8 elems
  const int nvec = row_elems / 8;
  if (idx >= T * (long)nvec) return;
  const long t = idx / nvec;
  const int e0 = (int)(idx % nvec) * 8;
  const long slot = slots[t];
  // K: token-major row copy
  *reinterpret_cast<s16x8*>(kc + slot * (long)row_elems + e0) =
      *reinterpret_cast<const s16x8*>(k + t * (long)row_elems + e0);
  // V: d-major scatter — vc[((page*Hkv + h)*D + d)*ps + off]
  const long page = slot / ps;
  const int off = (int)(slot % ps);
  s16x8 vv = *reinterpret_cast<const s16x8*>(v + t * (long)row_elems + e0);
  const int h = e0 / D;
  const int d0 = e0 % D;
#pragma unroll
  for (int j = 0; j < 8; j++)
    vc[((page * Hkv + h) * (long)D + d0 + j) * ps + off] = vv[j];
}

void kv_append(torch::Tensor k, torch::Tensor v, torch::Tensor slots,
               torch::Tensor k_cache, torch::Tensor v_cache) {
  TORCH_CHECK(k.scalar_type() == torch::kBFloat16 && k.is_contiguous());
  TORCH_CHECK(slots.scalar_type() == torch::kLong);
  const long T = k.size(0);
  if (T == 0) return;
  const int Hkv = k.size(1), D = k.size(2);
  const int ps = k_cache.size(1);
  TORCH_CHECK(D % 8 == 0);
  const long total = T * (Hkv * D / 8);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(kv_append_kernel, dim3((total + 255) / 256), dim3(256), 0, stream,
                     (const short*)k.data_ptr(), (const short*)v.data_ptr(),
                     slots.data_ptr<long>(), (short*)k_cache.data_ptr(),
                     (short*)v_cache.data_ptr(), T, Hkv, D, ps);
  HIP_CHECK_LAST();
}

// ------------------------------------------------------- decode attention
template <int D>
__global__ void paged_decode_kernel(const short* __restrict__ q,    // [B, Hq, D]
                                    const short* __restrict__ kc,   // [P, 16, Hkv, D]
                                    const short* __restrict__ vc,   // [P, Hkv, D, 16]
                                    const int* __restrict__ bt,     // [B, maxP]
                                    const int* __restrict__ seq_lens,
                                    short* __restrict__ out,        // [B, Hq, D]
                                    int Hq, int Hkv, int maxP, int G, float scale) {
  constexpr int NC = D / 32;
  constexpr int PS = 16;
  const int b = blockIdx.x / Hkv;
  const int kvh = blockIdx.x % Hkv;
  const int L = seq_lens[b];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int hi = lane >> 4, lo = lane & 15;
  const int h0 = kvh * G;

  __shared__ short Pb[4][16 * 32];          // per-wave P [g][tok]
  __shared__ float Lm[4][16], Ls[4][16];    // per-wave (m, s) per row
  __shared__ float Lacc[4][16][D];          // per-wave O accumulators

  // Q fragment: lane holds Q[g = lo][32c + 8hi + j]; zero for g >= G
  bf16x8k qf[NC];
  {
    const int g = lo;
    const long base = ((long)b * Hq + h0 + min(g, G - 1)) * D;
#pragma unroll
    for (int c = 0; c < NC; c++) {
      s16x8 raw{};
      if (g < G)
        raw = *reinterpret_cast<const s16x8*>(q + base + 32 * c + 8 * hi);
      qf[c] = *reinterpret_cast<bf16x8k*>(&raw);
    }
  }

  float m[4], s_[4];
  f32x4 acc_o[D / 16];
#pragma unroll
  for (int r = 0; r < 4; r++) { m[r] = -INFINITY; s_[r] = 0.f; }
#pragma unroll
  for (int t = 0; t < D / 16; t++) acc_o[t] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int ntiles = (L + 31) / 32;
  const int* btb = bt + (long)b * maxP;
  for (int tile = wid; tile < ntiles; tile += 4) {
    const int t0 = tile * 32;
    // ---- scores: S[g][tok] over two 16-token subtiles ------------------
    f32x4 sc[2];
#pragma unroll
    for (int n = 0; n < 2; n++) {
      sc[n] = f32x4{0.f, 0.f, 0.f, 0.f};
      const int page = btb[min(2 * tile + n, maxP - 1)];
      const long krow = (((long)page * PS + lo) * Hkv + kvh) * D;
#pragma unroll
      for (int c = 0; c < NC; c++) {
        s16x8 raw = *reinterpret_cast<const s16x8*>(kc + krow + 32 * c + 8 * hi);
        sc[n] = mfma16k(qf[c], *reinterpret_cast<bf16x8k*>(&raw), sc[n]);
      }
    }
    // ---- mask + online softmax (rows = q heads g = 4*hi + r) ----------
    float pm[2][4];
#pragma unroll
    for (int n = 0; n < 2; n++) {
      const int ki = t0 + 16 * n + lo;
#pragma unroll
      for (int r = 0; r < 4; r++)
        pm[n][r] = (ki < L) ? sc[n][r] * scale : -INFINITY;
    }
    float rowmax[4], fac[4];
#pragma unroll
    for (int r = 0; r < 4; r++) {
      float x = fmaxf(pm[0][r], pm[1][r]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off));
      const float mn = fmaxf(m[r], x);
      fac[r] = (m[r] == -INFINITY) ? 0.f : __expf(m[r] - mn);
      m[r] = mn;
    }
    float rowsum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 2; n++) {
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const float p = (pm[n][r] == -INFINITY) ? 0.f : __expf(pm[n][r] - m[r]);
        rowsum[r] += p;
        Pb[wid][(4 * hi + r) * 32 + 16 * n + lo] = f2bf(p);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; r++) {
      float x = rowsum[r];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) x += __shfl_xor(x, off);
      s_[r] = s_[r] * fac[r] + x;
    }
#pragma unroll
    for (int t = 0; t < D / 16; t++) {
#pragma unroll
      for (int r = 0; r < 4; r++) acc_o[t][r] *= fac[r];
    }
    lds_fence_wave_kv();
    // ---- PV: A = P [g][tok], B = V^T (d-major pages) -------------------
    bf16x8k pf;
    {
      s16x8 raw = *reinterpret_cast<const s16x8*>(&Pb[wid][lo * 32 + 8 * hi]);
      pf = *reinterpret_cast<bf16x8k*>(&raw);
    }
    const int pv_page = btb[min(2 * tile + (hi >> 1), maxP - 1)];
    const int tokoff = 8 * (hi & 1);
#pragma unroll
    for (int t = 0; t < D / 16; t++) {
      const long vaddr = (((long)pv_page * Hkv + kvh) * D + 16 * t + lo) * PS + tokoff;
      s16x8 raw = *reinterpret_cast<const s16x8*>(vc + vaddr);
      acc_o[t] = mfma16k(pf, *reinterpret_cast<bf16x8k*>(&raw), acc_o[t]);
    }
  }

  // ---- cross-wave merge through LDS -----------------------------------
  if (lo < 16) {
    // lane (hi, lo): writes rows 4*hi..4*hi+3 at cols d = 16t + lo
#pragma unroll
    for (int r = 0; r < 4; r++) {
      if (lo == 0) { Lm[wid][4 * hi + r] = m[r]; Ls[wid][4 * hi + r] = s_[r]; }
#pragma unroll
      for (int t = 0; t < D / 16; t++)
        Lacc[wid][4 * hi + r][16 * t + lo] = acc_o[t][r];
    }
  }
  __syncthreads();
  if (wid == 0) {
    constexpr int OV = D / 64;
    for (int g = 0; g < G; g++) {
      float M = Lm[0][g];
      for (int w = 1; w < 4; w++) M = fmaxf(M, Lm[w][g]);
      if (M == -INFINITY) M = 0.f;
      float S = 0.f;
      float o[OV > 0 ? OV : 1];
#pragma unroll
      for (int j = 0; j < (OV > 0 ? OV : 1); j++) o[j] = 0.f;
      for (int w = 0; w < 4; w++) {
        const float c = (Lm[w][g] == -INFINITY) ? 0.f : __expf(Lm[w][g] - M);
        S += Ls[w][g] * c;
        if (OV > 0) {
#pragma unroll
          for (int j = 0; j < (OV > 0 ? OV : 1); j++)
            o[j] += Lacc[w][g][lane * OV + j] * c;
        } else if (lane < D) {
          o[0] += Lacc[w][g][lane] * c;
        }
      }
      const float invS = S > 0.f ? 1.f / S : 0.f;
      short* op = out + ((long)b * Hq + h0 + g) * D;
      if (OV > 0) {
#pragma unroll
        for (int j = 0; j < (OV > 0 ? OV : 1); j++)
          op[lane * OV + j] = f2bf(o[j] * invS);
      } else if (lane < D) {
        op[lane] = f2bf(o[0] * invS);
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
  TORCH_CHECK(page_size == 16, "decode kernel requires page_size == 16");
  TORCH_CHECK(Hq % Hkv == 0 && G <= 16, "GQA group size must be <= 16");
  auto out = torch::empty_like(q);
  if (B == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(B * Hkv), block(256);
#define LAUNCH_D(DD)                                                            \
  hipLaunchKernelGGL((paged_decode_kernel<DD>), grid, block, 0, stream,         \
                     (const short*)q.data_ptr(), (const short*)k_cache.data_ptr(), \
                     (const short*)v_cache.data_ptr(), block_tables.data_ptr<int>(), \
                     seq_lens.data_ptr<int>(), (short*)out.data_ptr(),          \
                     Hq, Hkv, maxP, G, (float)scale)
  if (D == 128) { LAUNCH_D(128); }
  else if (D == 64) { LAUNCH_D(64); }
  else if (D == 32) { LAUNCH_D(32); }
  else { TORCH_CHECK(false, "unsupported head_dim ", D); }
#undef LAUNCH_D
  HIP_CHECK_LAST();
  return out;
}
