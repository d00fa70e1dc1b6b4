// Standalone ablation of sample_topp's cost structure (no torch).
// Variants strip one stage at a time to locate the non-streaming cost
// (cdna_hip_programming.md §5 common-mistake 8: ablate before optimizing).
//
// MEASURED (MI355X, B=2048, V=151936): V0 0.406 ms (6.1 TB/s), V1 0.411,
// V2 1.693, V3 1.811 — the LDS f32 atomics (V1→V2, +1.28 ms) are 75% of
// the kernel and are THROUGHPUT-bound, not conflict-bound (lane-keyed
// spread slots measured no better).  Conclusion shipped in sampling.hip:
// a 32-bin per-thread REGISTER histogram + streaming refinement passes
// with zero atomics (1.72 → 1.28/1.11 ms random/peaked).
//   V0 pure streaming: 4 read passes, fmax only
//   V1 + exp on every element (VALU/trans cost)
//   V2 + histogram atomics (LDS)
//   V3 + serial thread-0 bin scan
//   V4 full kernel shape (max pass, hist pass, kept pass, walk)
// Build (GPU box):  hipcc --offload-arch=gfx950 -O3 tools/topp_ablate.hip -o /tmp/ta && /tmp/ta
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>

#define SBLOCK 256
#define NBINS 1024
typedef short s16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf2f(short x) {
  union { float f; unsigned u; } c;
  c.u = ((unsigned)(unsigned short)x) << 16;
  return c.f;
}

template <int VARIANT>
__global__ void ablate_kernel(const short* __restrict__ logits, float* __restrict__ out,
                              int V) {
  __shared__ float hist[NBINS];
  const long row = blockIdx.x;
  const short* lr = logits + row * (long)V;
  const int nvec = V / 8;
  float acc = 0.f;
  for (int pass = 0; pass < 4; pass++) {
    if (VARIANT >= 2 && pass == 1) {
      for (int i = threadIdx.x; i < NBINS; i += SBLOCK) hist[i] = 0.f;
      __syncthreads();
    }
    for (int i = threadIdx.x; i < nvec; i += SBLOCK) {
      s16x8 v = *reinterpret_cast<const s16x8*>(lr + i * 8);
#pragma unroll
      for (int j = 0; j < 8; j++) {
        const float u = bf2f(v[j]) * 1.43f - 18.f;
        if (VARIANT == 0) acc = fmaxf(acc, u);
        else {
          const float e = __expf(u);
          acc += e;
          if (VARIANT >= 2 && pass == 1) {
            int b = (int)((u + 32.f) * (NBINS / 32.f));
            b = max(0, min(NBINS - 1, b));
            atomicAdd(&hist[b], e);
          }
        }
      }
    }
    if (VARIANT >= 2 && pass == 1) __syncthreads();
    if (VARIANT >= 3 && pass == 1 && threadIdx.x == 0) {
      float a = 0.f;
      for (int b = NBINS - 1; b >= 0; b--) { a += hist[b]; if (a > 1e30f) break; }
      acc += a;
    }
    if (VARIANT >= 3 && pass == 1) __syncthreads();
  }
  if (threadIdx.x == 0) out[row] = acc;
}

int main() {
  const int B = 2048, V = 151936;
  short* d_logits; float* d_out;
  hipMalloc(&d_logits, (size_t)B * V * 2);
  hipMalloc(&d_out, B * 4);
  short* h = (short*)malloc((size_t)B * V * 2);
  srand(1);
  for (long i = 0; i < (long)B * V; i++) {
    float f = ((rand() % 20000) - 10000) / 1000.0f;
    unsigned u; __builtin_memcpy(&u, &f, 4);
    h[i] = (short)(u >> 16);
  }
  hipMemcpy(d_logits, h, (size_t)B * V * 2, hipMemcpyHostToDevice);
  hipEvent_t e0, e1;
  hipEventCreate(&e0); hipEventCreate(&e1);
#define RUN(VAR)                                                              \
  do {                                                                        \
    for (int w = 0; w < 3; w++)                                               \
      hipLaunchKernelGGL(ablate_kernel<VAR>, dim3(B), dim3(SBLOCK), 0, 0,     \
                         d_logits, d_out, V);                                 \
    hipEventRecord(e0);                                                       \
    for (int it = 0; it < 20; it++)                                           \
      hipLaunchKernelGGL(ablate_kernel<VAR>, dim3(B), dim3(SBLOCK), 0, 0,     \
                         d_logits, d_out, V);                                 \
    hipEventRecord(e1);                                                       \
    hipEventSynchronize(e1);                                                  \
    float ms;                                                                 \
    hipEventElapsedTime(&ms, e0, e1);                                         \
    printf("V%d: %.3f ms  (%.2f TB/s over 4 passes)\n", VAR, ms / 20,         \
           4.0 * B * (double)V * 2 / (ms / 20 * 1e-3) / 1e12);                \
  } while (0)
  RUN(0); RUN(1); RUN(2); RUN(3);
  return 0;
}
