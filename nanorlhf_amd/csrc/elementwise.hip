// RMSNorm fwd/bwd, RoPE apply, SwiGLU fwd/bwd — fused memory-bound kernels
// for gfx950.  All bf16 I/O with fp32 accumulation, bf16x8 (16 B) vector
// loads per lane (cdna_hip_programming.md G13), one workgroup per row for
// the norm reductions.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

// ------------------------------------------------------------- RMSNorm fwd
// x [N, H] bf16 -> y [N, H] bf16, invrms [N] f32.  One 256-thread block/row.
template <int BLOCK>
__global__ void rmsnorm_fwd_kernel(const short* __restrict__ x,
                                   const short* __restrict__ w,
                                   short* __restrict__ y,
                                   float* __restrict__ invrms,
                                   int H, float eps) {
  __shared__ float scratch[BLOCK / 64];
  const long row = blockIdx.x;
  const short* xr = x + row * (long)H;
  short* yr = y + row * (long)H;
  float ss = 0.f;
  const int nvec = H / 8;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 v = *reinterpret_cast<const s16x8*>(xr + i * 8);
#pragma unroll
    for (int j = 0; j < 8; j++) { float f = bf2f(v[j]); ss += f * f; }
  }
  ss = block_sum<BLOCK>(ss, scratch);
  const float inv = rsqrtf(ss / (float)H + eps);
  if (threadIdx.x == 0) invrms[row] = inv;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 v = *reinterpret_cast<const s16x8*>(xr + i * 8);
    s16x8 wv = *reinterpret_cast<const s16x8*>(w + i * 8);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      // match the reference numerics: bf16(x*inv) * w
      float t = bf2f(f2bf(bf2f(v[j]) * inv));
      o[j] = f2bf(t * bf2f(wv[j]));
    }
    *reinterpret_cast<s16x8*>(yr + i * 8) = o;
  }
}

// ------------------------------------------------------------- RMSNorm bwd
// dx = inv * (g - x_hat * mean(g * x_hat)),  g = dy*w,  x_hat = x*inv
// Each block processes ROWS_PER_BLOCK rows; every thread owns fixed column
// slices, accumulating its dw partial in REGISTERS across the block's rows
// (no per-element atomics — the atomic version measured 1.0 ms/call from
// contention), then one global atomicAdd per element per block.
template <int BLOCK>
__global__ void rmsnorm_bwd_kernel(const short* __restrict__ dy,
                                   const short* __restrict__ x,
                                   const short* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   short* __restrict__ dx,
                                   float* __restrict__ dw,  // [H]
                                   long N, int H, int rows_per_block) {
  __shared__ float scratch[BLOCK / 64];
  constexpr int MAX_SLICES = 4;  // supports H up to BLOCK*8*4 = 8192
  const int nvec = H / 8;
  float dwacc[MAX_SLICES][8];
#pragma unroll
  for (int sl = 0; sl < MAX_SLICES; sl++)
#pragma unroll
    for (int j = 0; j < 8; j++) dwacc[sl][j] = 0.f;

  const long row0 = (long)blockIdx.x * rows_per_block;
  const long row1 = min(row0 + rows_per_block, N);
  for (long row = row0; row < row1; row++) {
    const short* dyr = dy + row * (long)H;
    const short* xr = x + row * (long)H;
    short* dxr = dx + row * (long)H;
    const float inv = invrms[row];
    float dot = 0.f;
    for (int i = threadIdx.x; i < nvec; i += BLOCK) {
      s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i * 8);
      s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i * 8);
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i * 8);
#pragma unroll
      for (int j = 0; j < 8; j++)
        dot += bf2f(dv[j]) * bf2f(wv[j]) * bf2f(xv[j]) * inv;
    }
    dot = block_sum<BLOCK>(dot, scratch) / (float)H;
    int sl = 0;
    for (int i = threadIdx.x; i < nvec; i += BLOCK, sl++) {
      s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i * 8);
      s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i * 8);
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i * 8);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const float xh = bf2f(xv[j]) * inv;
        const float g = bf2f(dv[j]) * bf2f(wv[j]);
        o[j] = f2bf((g - xh * dot) * inv);
        if (sl < MAX_SLICES) dwacc[sl][j] += bf2f(dv[j]) * bf2f(f2bf(xh));
      }
      *reinterpret_cast<s16x8*>(dxr + i * 8) = o;
    }
  }
  // drain register partials: one atomic per element per block
  int sl = 0;
  for (int i = threadIdx.x; i < nvec && sl < MAX_SLICES; i += BLOCK, sl++) {
#pragma unroll
    for (int j = 0; j < 8; j++) atomicAdd(&dw[i * 8 + j], dwacc[sl][j]);
  }
}

// ----------------------------------------------------------------- RoPE
// x [T, Hh, D] bf16 in-place; table [maxpos, D] f32 (cos | sin halves);
// positions [T] i64.  sign=-1 applies the inverse rotation (backward).
__global__ void rope_kernel(const short* __restrict__ x,
                            short* __restrict__ y,
                            const float* __restrict__ table,
                            const long* __restrict__ pos,
                            int T, int Hh, int D, float sign) {
  // one thread per (t, h, 4 pairs)
  const int pairs4 = D / 8;  // groups of 4 rotation pairs
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)T * Hh * pairs4;
  if (idx >= total) return;
  const int g = idx % pairs4;
  const int h = (idx / pairs4) % Hh;
  const long t = idx / ((long)pairs4 * Hh);
  const short* base = x + (t * Hh + h) * (long)D;
  short* obase = y + (t * Hh + h) * (long)D;
  const float* tb = table + pos[t] * (long)D;
  s16x4 x1 = *reinterpret_cast<const s16x4*>(base + g * 4);
  s16x4 x2 = *reinterpret_cast<const s16x4*>(base + D / 2 + g * 4);
  f32x4 c = *reinterpret_cast<const f32x4*>(tb + g * 4);
  f32x4 s = *reinterpret_cast<const f32x4*>(tb + D / 2 + g * 4);
  s16x4 o1, o2;
#pragma unroll
  for (int j = 0; j < 4; j++) {
    float a = bf2f(x1[j]), b = bf2f(x2[j]);
    float cs = c[j], sn = s[j] * sign;
    o1[j] = f2bf(a * cs - b * sn);
    o2[j] = f2bf(b * cs + a * sn);
  }
  *reinterpret_cast<s16x4*>(obase + g * 4) = o1;
  *reinterpret_cast<s16x4*>(obase + D / 2 + g * 4) = o2;
}

// ----------------------------------------------------------------- SwiGLU
// gate_up [N, 2I] -> y [N, I]; y = silu(g) * u
__global__ void swiglu_fwd_kernel(const short* __restrict__ gu,
                                  short* __restrict__ y,
                                  long N, int I) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;  // in units of 8
  const long total = N * (long)(I / 8);
  if (idx >= total) return;
  const long n = idx / (I / 8);
  const int i = (idx % (I / 8)) * 8;
  const short* row = gu + n * 2L * I;
  s16x8 g = *reinterpret_cast<const s16x8*>(row + i);
  s16x8 u = *reinterpret_cast<const s16x8*>(row + I + i);
  s16x8 o;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    float gf = bf2f(g[j]);
    float sig = 1.f / (1.f + __expf(-gf));
    o[j] = f2bf(gf * sig * bf2f(u[j]));
  }
  *reinterpret_cast<s16x8*>(y + n * (long)I + i) = o;
}

__global__ void swiglu_bwd_kernel(const short* __restrict__ dy,
                                  const short* __restrict__ gu,
                                  short* __restrict__ dgu,
                                  long N, int I) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = N * (long)(I / 8);
  if (idx >= total) return;
  const long n = idx / (I / 8);
  const int i = (idx % (I / 8)) * 8;
  const short* row = gu + n * 2L * I;
  s16x8 g = *reinterpret_cast<const s16x8*>(row + i);
  s16x8 u = *reinterpret_cast<const s16x8*>(row + I + i);
  s16x8 d = *reinterpret_cast<const s16x8*>(dy + n * (long)I + i);
  s16x8 dg, du;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    float gf = bf2f(g[j]), uf = bf2f(u[j]), df = bf2f(d[j]);
    float sig = 1.f / (1.f + __expf(-gf));
    float silu = gf * sig;
    float dsilu = sig * (1.f + gf * (1.f - sig));
    dg[j] = f2bf(df * uf * dsilu);
    du[j] = f2bf(df * silu);
  }
  short* drow = dgu + n * 2L * I;
  *reinterpret_cast<s16x8*>(drow + i) = dg;
  *reinterpret_cast<s16x8*>(drow + I + i) = du;
}

// ======================================================== host wrappers ==
static inline void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16 on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  check_bf16(x, "x"); check_bf16(w, "w");
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "hidden size must be divisible by 8");
  const long N = x.numel() / H;
  auto y = torch::empty_like(x);
  auto invrms = torch::empty({N}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((rmsnorm_fwd_kernel<256>), dim3(N), dim3(256), 0, stream,
                     (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                     (short*)y.data_ptr(), invrms.data_ptr<float>(), H, (float)eps);
  HIP_CHECK_LAST();
  return {y, invrms};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor invrms) {
  check_bf16(dy, "dy"); check_bf16(x, "x"); check_bf16(w, "w");
  const int H = x.size(-1);
  const long N = x.numel() / H;
  TORCH_CHECK(H <= 256 * 8 * 4, "hidden too large for rmsnorm_bwd slices");
  auto dx = torch::empty_like(x);
  auto dwf = torch::zeros({H}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  // ~4 blocks/CU worth of parallelism; more rows per block = fewer atomics
  const int rows_per_block = (int)std::max<long>(1, (N + 1023) / 1024);
  const long grid = (N + rows_per_block - 1) / rows_per_block;
  hipLaunchKernelGGL((rmsnorm_bwd_kernel<256>), dim3(grid), dim3(256), 0, stream,
                     (const short*)dy.data_ptr(), (const short*)x.data_ptr(),
                     (const short*)w.data_ptr(), invrms.data_ptr<float>(),
                     (short*)dx.data_ptr(), dwf.data_ptr<float>(), N, H,
                     rows_per_block);
  HIP_CHECK_LAST();
  return {dx, dwf.to(torch::kBFloat16)};
}

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor table, torch::Tensor positions,
                       double sign) {
  check_bf16(x, "x");
  TORCH_CHECK(table.scalar_type() == torch::kFloat32, "rope table must be fp32");
  TORCH_CHECK(positions.scalar_type() == torch::kLong, "positions must be int64");
  const int D = x.size(-1);
  const int Hh = x.size(1);
  const int T = x.size(0);
  TORCH_CHECK(D % 8 == 0, "head_dim must be divisible by 8");
  const long total = (long)T * Hh * (D / 8);
  const int block = 256;
  const long grid = (total + block - 1) / block;
  auto y = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(block), 0, stream,
                     (const short*)x.data_ptr(), (short*)y.data_ptr(),
                     table.data_ptr<float>(),
                     positions.data_ptr<long>(), T, Hh, D, (float)sign);
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor swiglu_fwd(torch::Tensor gu) {
  check_bf16(gu, "gate_up");
  const int I2 = gu.size(-1);
  TORCH_CHECK(I2 % 16 == 0, "2*intermediate must be divisible by 16");
  const int I = I2 / 2;
  const long N = gu.numel() / I2;
  auto y = torch::empty({gu.size(0), I}, gu.options());
  const long total = N * (I / 8);
  const int block = 256;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3((total + block - 1) / block), dim3(block),
                     0, stream, (const short*)gu.data_ptr(), (short*)y.data_ptr(), N, I);
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor gu) {
  check_bf16(dy, "dy"); check_bf16(gu, "gate_up");
  const int I2 = gu.size(-1);
  const int I = I2 / 2;
  const long N = gu.numel() / I2;
  auto dgu = torch::empty_like(gu);
  const long total = N * (I / 8);
  const int block = 256;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3((total + block - 1) / block), dim3(block),
                     0, stream, (const short*)dy.data_ptr(), (const short*)gu.data_ptr(),
                     (short*)dgu.data_ptr(), N, I);
  HIP_CHECK_LAST();
  return dgu;
}

// ----------------------------------------------- fused residual + RMSNorm
// h = x + res;  y = rmsnorm(h) * w.  For H <= BLOCK*8 each thread keeps its
// h slice in registers across both phases (read x,res; write h,y — the
// unfused pair costs an extra h read plus a launch).  invrms saved for the
// backward, which reuses rmsnorm_bwd on h (d h = d x = d res).
template <int BLOCK>
__global__ void rmsnorm_addres_fwd_kernel(const short* __restrict__ x,
                                          const short* __restrict__ res,
                                          const short* __restrict__ w,
                                          short* __restrict__ h,
                                          short* __restrict__ y,
                                          float* __restrict__ invrms,
                                          int H, float eps) {
  __shared__ float scratch[BLOCK / 64];
  const long row = blockIdx.x;
  const short* xr = x + row * (long)H;
  const short* rr = res + row * (long)H;
  short* hr = h + row * (long)H;
  short* yr = y + row * (long)H;
  const int nvec = H / 8;
  float ss = 0.f;
  if (nvec <= BLOCK) {
    // register-resident path (H <= BLOCK*8)
    s16x8 hv{};
    const int i = threadIdx.x;
    if (i < nvec) {
      s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i * 8);
      s16x8 rv = *reinterpret_cast<const s16x8*>(rr + i * 8);
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const float f = bf2f(xv[j]) + bf2f(rv[j]);
        hv[j] = f2bf(f);
        const float fb = bf2f(hv[j]);
        ss += fb * fb;
      }
    }
    ss = block_sum<BLOCK>(ss, scratch);
    const float inv = rsqrtf(ss / (float)H + eps);
    if (threadIdx.x == 0) invrms[row] = inv;
    if (i < nvec) {
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i * 8);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const float t = bf2f(f2bf(bf2f(hv[j]) * inv));
        o[j] = f2bf(t * bf2f(wv[j]));
      }
      *reinterpret_cast<s16x8*>(hr + i * 8) = hv;
      *reinterpret_cast<s16x8*>(yr + i * 8) = o;
    }
    return;
  }
  // general path: write h in pass 1, re-read (L2-hot) in pass 2
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i * 8);
    s16x8 rv = *reinterpret_cast<const s16x8*>(rr + i * 8);
    s16x8 hv;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      hv[j] = f2bf(bf2f(xv[j]) + bf2f(rv[j]));
      const float f = bf2f(hv[j]);
      ss += f * f;
    }
    *reinterpret_cast<s16x8*>(hr + i * 8) = hv;
  }
  ss = block_sum<BLOCK>(ss, scratch);
  const float inv = rsqrtf(ss / (float)H + eps);
  if (threadIdx.x == 0) invrms[row] = inv;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    s16x8 hv = *reinterpret_cast<const s16x8*>(hr + i * 8);
    s16x8 wv = *reinterpret_cast<const s16x8*>(w + i * 8);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const float t = bf2f(f2bf(bf2f(hv[j]) * inv));
      o[j] = f2bf(t * bf2f(wv[j]));
    }
    *reinterpret_cast<s16x8*>(yr + i * 8) = o;
  }
}

std::vector<torch::Tensor> rmsnorm_addres_fwd(torch::Tensor x, torch::Tensor res,
                                              torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.is_contiguous() && res.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.sizes() == res.sizes());
  const long N = x.numel() / x.size(-1);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0);
  auto h = torch::empty_like(x);
  auto y = torch::empty_like(x);
  auto invrms = torch::empty({N}, x.options().dtype(torch::kFloat32));
  if (N == 0) return {y, h, invrms};
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_addres_fwd_kernel<256>, dim3(N), dim3(256), 0, stream,
                     (const short*)x.data_ptr(), (const short*)res.data_ptr(),
                     (const short*)w.data_ptr(), (short*)h.data_ptr(),
                     (short*)y.data_ptr(), invrms.data_ptr<float>(), H, (float)eps);
  HIP_CHECK_LAST();
  return {y, h, invrms};
}
