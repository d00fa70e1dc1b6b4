// Fused LoRA GEMM for gfx950 — y = x·Wᵀ [+ u·Bᵀ] [+ bias].
//
// The LoRA adapter matmul the reference delegates to peft/cuBLAS
// (GRPO/grpo.py:226-243 call site).  MI355X-native
// design: instead of the 3-launch chain  y = F.linear(x,W) +
// s·F.linear(F.linear(x,A),B), the adapter contribution is folded into the
// base GEMM as ONE extra K-tile — the kernel consumes a "K-concat" operand
// pair (x  with K=H  from the frozen weight) + (u = s·x·Aᵀ with K=r=64 from
// the adapter B matrix), so the whole thing is a single MFMA GEMM of depth
// H+64 (≈4 % more FLOPs than the base GEMM, one launch, no y round trip).
// Backward reuses the SAME kernel: dx = dy·W + du·A is the identical shape
// with (dy, Wᵀ) and (du = s·dy·B, Aᵀ) as the operand pairs; W is frozen
// under LoRA so Wᵀ is cached once.  dA/dB are skinny r-rank GEMMs left to
// hipBLASLt (r/N of the FLOPs).
//
// Kernel structure (cdna_hip_programming.md §5 "step-3" ladder structure,
// ~874 TF at 4096³ bf16): 128×128 output tile, BK=64, 4 waves (2×2 of
// 64×64), v_mfma_f32_16x16x32_bf16, double-buffered LDS filled by
// global_load_lds dwordx4 (16 B/lane), one __syncthreads per K-step,
// XCD-aware bijective block swizzle (8 XCDs, each with its own L2).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef float f32x4_l __attribute__((ext_vector_type(4)));
typedef short bf16x8_l __attribute__((ext_vector_type(8)));

DEVINL f32x4_l mfma16(bf16x8_l a, bf16x8_l b, f32x4_l c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

#define GLDS16(gsrc, ldst)                                                     \
  __builtin_amdgcn_global_load_lds(                                            \
      (const __attribute__((address_space(1))) void*)(gsrc),                   \
      (__attribute__((address_space(3))) void*)(ldst), 16, 0, 0)

// 128x128x(K1[+64]) bf16 GEMM tile: y[M,N] = x·Wᵀ (+ u·Bᵀ) (+ bias).
//   x:[M,K1] row-major  W:[N,K1] row-major (nn.Linear layout)
//   u:[M,64]            B:[N,64]
// Requirements: K1 % 64 == 0; all pointers 16-B aligned rows (K1 % 8 == 0).
// M/N edges are handled by clamping staging rows and masking stores.
template <bool HAS_LORA, bool HAS_BIAS>
__launch_bounds__(256, 2)
__global__ void lora_gemm_kernel(const short* __restrict__ x,
                                 const short* __restrict__ w,
                                 const short* __restrict__ u,
                                 const short* __restrict__ b,
                                 const float* __restrict__ bias,
                                 short* __restrict__ y,
                                 int M, int N, int K1, int grid_n) {
  // one __shared__ object only (§5 trap 4a): [buf][A/B][128][64]
  __shared__ short lds[2][2][128 * 64];

  // XCD-aware bijective swizzle (8 XCDs round-robin workgroup dispatch)
  const int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    const int q = nwg >> 3, r = nwg & 7;
    const int xcd = wg & 7, i = wg >> 3;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  const int bm = wg / grid_n, bn = wg % grid_n;
  const int m0 = bm * 128, n0 = bn * 128;

  const int tid = threadIdx.x;           // 256 threads = 4 waves
  const int lane = tid & 63, wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1; // wave grid 2x2 -> 64x64 per wave

  const int n_k1 = K1 >> 6;
  const int n_kt = n_k1 + (HAS_LORA ? 1 : 0);

  // staging geometry: tile 128x64 bf16 = 16 KB = 4 glds rounds x 256 lanes
  // x 16 B; chunk c = round*256 + tid covers row c/8, col (c%8)*8.
  // LDS image is lane-linear ([row][col] row-major) as glds requires.
  auto stage_tile = [&](int kt, int buf) {
    const bool lora_t = HAS_LORA && (kt == n_k1);
    const int k0 = kt << 6;
#pragma unroll
    for (int rnd = 0; rnd < 4; ++rnd) {
      const int c = rnd * 256 + tid;
      const int row = c >> 3, col = (c & 7) << 3;
      // A operand: x rows (or u rows on the adapter K-tile)
      const int ar = min(m0 + row, M - 1);
      const short* asrc = lora_t ? (u + (long)ar * 64 + col)
                                 : (x + (long)ar * K1 + k0 + col);
      GLDS16(asrc, &lds[buf][0][c << 3]);
      // B operand: W rows (or B rows)
      const int brn = min(n0 + row, N - 1);
      const short* bsrc = lora_t ? (b + (long)brn * 64 + col)
                                 : (w + (long)brn * K1 + k0 + col);
      GLDS16(bsrc, &lds[buf][1][c << 3]);
    }
  };

  f32x4_l acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4_l{0.f, 0.f, 0.f, 0.f};

  stage_tile(0, 0);
  __syncthreads();  // drains the glds (vmcnt(0) folded into the barrier)

  const int fr = lane & 15;          // fragment row (A) / col (B)
  const int fk = (lane >> 4) << 3;   // fragment k-offset (8 bf16)

  for (int kt = 0; kt < n_kt; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < n_kt) stage_tile(kt + 1, buf ^ 1);
    const short* At = lds[buf][0];
    const short* Bt = lds[buf][1];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8_l af[4], bf[4];
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
        af[mf] = *reinterpret_cast<const bf16x8_l*>(
            At + ((wm * 64 + mf * 16 + fr) << 6) + kk * 32 + fk);
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        bf[nf] = *reinterpret_cast<const bf16x8_l*>(
            Bt + ((wn * 64 + nf * 16 + fr) << 6) + kk * 32 + fk);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mf][nf] = mfma16(af[mf], bf[nf], acc[mf][nf]);
    }
    __syncthreads();  // waves done reading buf; glds into buf^1 drained
  }

  // epilogue: C/D map col=lane&15, row=(lane>>4)*4+reg
  const int crow0 = (lane >> 4) << 2;
#pragma unroll
  for (int nf = 0; nf < 4; ++nf) {
    const int gn = n0 + wn * 64 + nf * 16 + fr;
    if (gn >= N) continue;
    const float bv = HAS_BIAS ? bias[gn] : 0.f;
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int gm = m0 + wm * 64 + mf * 16 + crow0 + reg;
        if (gm < M) y[(long)gm * N + gn] = f2bf(acc[mf][nf][reg] + bv);
      }
    }
  }
}

torch::Tensor lora_gemm(torch::Tensor x, torch::Tensor w,
                        c10::optional<torch::Tensor> u,
                        c10::optional<torch::Tensor> b,
                        c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 2);
  TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kBFloat16 && w.dim() == 2);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  const long M = x.size(0), K1 = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K1, "W inner dim mismatch");
  TORCH_CHECK(K1 % 64 == 0, "K must be a multiple of 64 (got ", K1, ")");
  TORCH_CHECK(M > 0 && N > 0);
  const bool has_lora = u.has_value();
  if (has_lora) {
    TORCH_CHECK(b.has_value(), "u given without B");
    TORCH_CHECK(u->is_contiguous() && b->is_contiguous());
    TORCH_CHECK(u->size(0) == M && u->size(1) == 64,
                "adapter rank must be 64 for the fused kernel");
    TORCH_CHECK(b->size(0) == N && b->size(1) == 64);
    TORCH_CHECK(u->dtype() == torch::kBFloat16 && b->dtype() == torch::kBFloat16);
  }
  torch::Tensor bias_f;
  const bool has_bias = bias.has_value();
  if (has_bias) {
    bias_f = bias->to(torch::kFloat32).contiguous();
    TORCH_CHECK(bias_f.size(0) == N);
  }
  auto y = torch::empty({M, N}, x.options());
  const int grid_m = (int)((M + 127) / 128), grid_n = (int)((N + 127) / 128);
  dim3 grid(grid_m * grid_n), block(256);
  auto stream = at::hip::getCurrentHIPStream();
  const short* up = has_lora ? (const short*)u->data_ptr() : nullptr;
  const short* bp = has_lora ? (const short*)b->data_ptr() : nullptr;
  const float* biasp = has_bias ? bias_f.data_ptr<float>() : nullptr;

#define LAUNCH(HL, HB)                                                        \
  hipLaunchKernelGGL((lora_gemm_kernel<HL, HB>), grid, block, 0, stream,      \
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),  \
                     up, bp, biasp, (short*)y.data_ptr(), (int)M, (int)N,     \
                     (int)K1, grid_n)
  if (has_lora && has_bias) LAUNCH(true, true);
  else if (has_lora) LAUNCH(true, false);
  else if (has_bias) LAUNCH(false, true);
  else LAUNCH(false, false);
#undef LAUNCH
  HIP_CHECK_LAST();
  return y;
}

// ---------------------------------------------------------------------------
// Rank-64 adapter epilogue: y += u·Bᵀ   (y:[M,N] bf16 in place, u:[M,64],
// B:[N,64]).  Used when hipBLASLt wins the base GEMM (deep-K shapes): the
// adapter contribution is ONE custom MFMA pass over y instead of two skinny
// library GEMMs + an eager add (3 launches + an extra y round trip).
// ---------------------------------------------------------------------------
__launch_bounds__(256, 4)
__global__ void lora_add_kernel(short* __restrict__ y,
                                const short* __restrict__ u,
                                const short* __restrict__ b,
                                int M, int N, int grid_n) {
  __shared__ short lds[2][128 * 64];
  const int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    const int q = nwg >> 3, r = nwg & 7;
    const int xcd = wg & 7, i = wg >> 3;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  }
  const int m0 = (wg / grid_n) * 128, n0 = (wg % grid_n) * 128;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;

#pragma unroll
  for (int rnd = 0; rnd < 4; ++rnd) {
    const int c = rnd * 256 + tid;
    const int row = c >> 3, col = (c & 7) << 3;
    GLDS16(u + (long)min(m0 + row, M - 1) * 64 + col, &lds[0][c << 3]);
    GLDS16(b + (long)min(n0 + row, N - 1) * 64 + col, &lds[1][c << 3]);
  }
  __syncthreads();

  const int fr = lane & 15, fk = (lane >> 4) << 3;
  f32x4_l acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4_l{0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    bf16x8_l af[4], bf[4];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
      af[mf] = *reinterpret_cast<const bf16x8_l*>(
          lds[0] + ((wm * 64 + mf * 16 + fr) << 6) + kk * 32 + fk);
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
      bf[nf] = *reinterpret_cast<const bf16x8_l*>(
          lds[1] + ((wn * 64 + nf * 16 + fr) << 6) + kk * 32 + fk);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf)
        acc[mf][nf] = mfma16(af[mf], bf[nf], acc[mf][nf]);
  }

  const int crow0 = (lane >> 4) << 2;
#pragma unroll
  for (int nf = 0; nf < 4; ++nf) {
    const int gn = n0 + wn * 64 + nf * 16 + fr;
    if (gn >= N) continue;
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int gm = m0 + wm * 64 + mf * 16 + crow0 + reg;
        if (gm < M) {
          const long idx = (long)gm * N + gn;
          y[idx] = f2bf(bf2f(y[idx]) + acc[mf][nf][reg]);
        }
      }
  }
}

void lora_add_(torch::Tensor y, torch::Tensor u, torch::Tensor b) {
  TORCH_CHECK(y.is_cuda() && y.dtype() == torch::kBFloat16 && y.dim() == 2);
  TORCH_CHECK(y.is_contiguous() && u.is_contiguous() && b.is_contiguous());
  const long M = y.size(0), N = y.size(1);
  TORCH_CHECK(u.size(0) == M && u.size(1) == 64);
  TORCH_CHECK(b.size(0) == N && b.size(1) == 64);
  const int grid_m = (int)((M + 127) / 128), grid_n = (int)((N + 127) / 128);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(lora_add_kernel, dim3(grid_m * grid_n), dim3(256), 0, stream,
                     (short*)y.data_ptr(), (const short*)u.data_ptr(),
                     (const short*)b.data_ptr(), (int)M, (int)N, grid_n);
  HIP_CHECK_LAST();
}
