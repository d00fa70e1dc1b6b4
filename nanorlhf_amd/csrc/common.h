// Shared device helpers for the nanorlhf_amd gfx950 kernels.
// CDNA4: wavefront = 64 lanes, 4x SIMD-32 per CU, LDS 160 KiB/CU.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEVINL __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;

// vector types for 8/16-byte loads (guide G13: always vectorize bf16 loads)
typedef short s16x4 __attribute__((ext_vector_type(4)));   // 4 bf16 = 8 B
typedef short s16x8 __attribute__((ext_vector_type(8)));   // 8 bf16 = 16 B
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short bf16x8_frag __attribute__((ext_vector_type(8)));  // MFMA A/B frag (4 VGPR)

DEVINL float bf2f(short x) {
  union { float f; unsigned int u; } c;
  c.u = ((unsigned int)(unsigned short)x) << 16;
  return c.f;
}

DEVINL short f2bf(float f) {
  union { float f; unsigned int u; } c;
  c.f = f;
  unsigned int lsb = (c.u >> 16) & 1;
  c.u += 0x7fff + lsb;  // round-to-nearest-even
  return (short)(c.u >> 16);
}

// ---- wave (64-lane) reductions ------------------------------------------
DEVINL float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

DEVINL float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// block reduction via LDS (block size multiple of 64)
template <int BLOCK>
DEVINL float block_sum(float v, float* lds_scratch /* BLOCK/64 floats */) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  v = wave_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float out = 0.f;
  if (wid == 0) {
    float x = (lane < BLOCK / 64) ? lds_scratch[lane] : 0.f;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off);
    out = x;
    if (lane == 0) lds_scratch[0] = out;
  }
  __syncthreads();
  out = lds_scratch[0];
  __syncthreads();
  return out;
}

template <int BLOCK>
DEVINL float block_max(float v, float* lds_scratch) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  v = wave_max(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  if (wid == 0) {
    float x = (lane < BLOCK / 64) ? lds_scratch[lane] : -INFINITY;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off));
    if (lane == 0) lds_scratch[0] = x;
  }
  __syncthreads();
  float out = lds_scratch[0];
  __syncthreads();
  return out;
}

// ---- counter-based RNG (deterministic replay) ----------------------------
DEVINL unsigned long long splitmix64(unsigned long long x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

DEVINL float hash_uniform(unsigned long long seed, unsigned long long step,
                          unsigned long long row) {
  unsigned long long h = splitmix64(seed ^ splitmix64(step ^ splitmix64(row)));
  // top 24 bits -> [0,1)
  return (float)(h >> 40) * (1.0f / 16777216.0f);
}

// wave-private LDS write->read ordering (staging buffers written and read
// by the same wave between barriers)
DEVINL void lds_fence_wave_kv() { asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory"); }

#define HIP_CHECK_LAST()                                                    \
  do {                                                                      \
    hipError_t e_ = hipGetLastError();                                      \
    if (e_ != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e_)); \
    }                                                                       \
  } while (0)
