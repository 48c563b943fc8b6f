// Empirical MFMA pipe-rate probe for gfx950: register-only MFMA chains
// (no memory traffic), several independent accumulators per wave, swept
// over waves/SIMD. Establishes the per-opcode ceiling our GEMMs should
// be priced against (the guide gives 64 FLOP/clk/SIMD for f32-input
// MFMA and the f64 rate implicitly; the SQ_VALU_MFMA_BUSY_CYCLES
// accounting for f32 measured 2x the naive expectation in r01, so the
// ceiling needs an on-silicon number).
//
// Build: hipcc --offload-arch=gfx950 -O3 -o /tmp/mfma_rate scripts/mfma_rate_probe.cpp
// Run:   /tmp/mfma_rate

#include <hip/hip_runtime.h>

#include <cstdio>

using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;
using f64x4 = __attribute__((__vector_size__(4 * sizeof(double)))) double;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;

__global__ void __launch_bounds__(256) k_f32(float* out, int iters) {
  f32x16 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
  float a = (float)threadIdx.x;
  float b = (float)(threadIdx.x ^ 7);
  for (int i = 0; i < iters; i++) {
    acc0 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc1, 0, 0, 0);
    acc2 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc2, 0, 0, 0);
    acc3 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc3, 0, 0, 0);
  }
  float s = acc0[0] + acc1[1] + acc2[2] + acc3[3];
  if (s == 12345.0f) out[0] = s;  // never true: keeps the chains live
}

__global__ void __launch_bounds__(256) k_f64(double* out, int iters) {
  f64x4 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
  double a = (double)threadIdx.x;
  double b = (double)(threadIdx.x ^ 7);
  for (int i = 0; i < iters; i++) {
    acc0 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc1, 0, 0, 0);
    acc2 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc2, 0, 0, 0);
    acc3 = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc3, 0, 0, 0);
  }
  double s = acc0[0] + acc1[1] + acc2[2] + acc3[3];
  if (s == 12345.0) out[0] = s;
}

__global__ void __launch_bounds__(256) k_bf16(float* out, int iters) {
  f32x4 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
  bf16x8 a, b;
  for (int j = 0; j < 8; j++) {
    a[j] = (__bf16)(float)(threadIdx.x + j);
    b[j] = (__bf16)(float)(threadIdx.x ^ (j + 1));
  }
  for (int i = 0; i < iters; i++) {
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc1, 0, 0, 0);
    acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc2, 0, 0, 0);
    acc3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc3, 0, 0, 0);
  }
  float s = acc0[0] + acc1[1] + acc2[2] + acc3[3];
  if (s == 12345.0f) out[0] = s;
}

template <typename K, typename T>
static void run(const char* name, K kernel, double flop_per_mfma,
                int waves_per_simd) {
  T* out;
  (void)hipMalloc(&out, sizeof(T));
  const int iters = 40000;
  // 256 threads = 4 waves/block -> 1 wave/SIMD per block; blocks/CU
  // scales waves/SIMD
  int blocks = 256 * waves_per_simd;
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  hipLaunchKernelGGL(kernel, dim3(blocks), dim3(256), 0, 0, out, 100);
  (void)hipDeviceSynchronize();  // warm
  (void)hipEventRecord(e0);
  hipLaunchKernelGGL(kernel, dim3(blocks), dim3(256), 0, 0, out, iters);
  (void)hipEventRecord(e1);
  (void)hipEventSynchronize(e1);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, e0, e1);
  double mfmas = (double)blocks * 4 /*waves*/ * 4 /*chains*/ * iters;
  double tf = mfmas * flop_per_mfma / (ms * 1e-3) / 1e12;
  // cycles per MFMA per SIMD at 2.4 GHz nominal
  double cyc = (ms * 1e-3) * 2.4e9 / (mfmas / 1024.0);
  printf("%-24s waves/SIMD=%d  %8.1f TF  %6.2f cyc/MFMA/SIMD\n", name,
         waves_per_simd, tf, cyc);
  (void)hipFree(out);
  (void)hipEventDestroy(e0);
  (void)hipEventDestroy(e1);
}

int main() {
  for (int w = 1; w <= 4; w <<= 1) {
    run<decltype(&k_f32), float>("mfma_f32_32x32x2_f32", k_f32, 4096.0, w);
    run<decltype(&k_f64), double>("mfma_f64_16x16x4_f64", k_f64, 2048.0, w);
    run<decltype(&k_bf16), float>("mfma_f32_16x16x32_bf16", k_bf16, 16384.0,
                                  w);
  }
  return 0;
}
