// f64 GEMM on the gfx950 matrix cores (v_mfma_f64_16x16x4_f64, the DGEMM
// MFMA: 78.6 TF fp64 matrix peak = the fp64 vector peak, exact f64).
// numpy's default dtype is float64, so this is the hot matmul for
// unannotated user numpy code.
//
// Structure: 64x64 block tile, 4 waves as 2x2 (each wave a 32x32 tile =
// 2x2 MFMA tiles of 16x16), BK=16 staged through LDS. A is kept row-major
// in LDS ([64][17] doubles, +1 pad: the 16-lane operand read walks rows,
// which is conflict-free at stride 17).
//
// Operand layout for mfma_f64_16x16x4_f64 (one f64 per lane for A/B):
//   A: lane l supplies A[i = l&15][k = l>>4]
//   B: lane l supplies B[k = l>>4][j = l&15]
//   C/D (4 regs): col = lane&15, row = (lane>>4) + 4*reg
// (empirically verified on gfx950 with scripts/mfma_f64_probe.cpp -- note
// the f64 C/D row map differs from the bf16/f16 16x16 map)

#include "common.h"

namespace {

constexpr int BM = 64;
constexpr int BN = 64;
constexpr int BK = 16;
constexpr int THREADS = 256;

using f64x4 = __attribute__((__vector_size__(4 * sizeof(double)))) double;

__global__ __launch_bounds__(THREADS) void gemm_f64_kernel(
    const double* __restrict__ A, const double* __restrict__ B,
    double* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
  int nwg = tiles_m * tiles_n;
  int wgid = blockIdx.x;
  {
    const int nxcd = 8;
    int q = nwg / nxcd, r = nwg % nxcd;
    int xcd = wgid % nxcd, idx = wgid / nxcd;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  int row0 = (wgid / tiles_n) * BM;
  int col0 = (wgid % tiles_n) * BN;

  // double-buffered (2 x 17 KiB: still 4 workgroups/CU); next tile's
  // global loads issue before this tile's MFMAs (T14 async-stage split)
  __shared__ double As[2][BM][BK + 1];
  __shared__ double Bs[2][BK][BN + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;
  const int wave_n = wave & 1;
  const int l15 = lane & 15;
  const int lk = lane >> 4;  // 0..3

  f64x4 acc[2][2] = {};

  const int a_m = tid >> 2;
  const int a_k = (tid & 3) * 4;
  const int b_k = tid >> 4;
  const int b_n = (tid & 15) * 4;

  double a_reg[4], b_reg[4];

  auto issue_loads = [&](int k0) {
    int gr = row0 + a_m;
#pragma unroll
    for (int j = 0; j < 4; j++) {
      int gk = k0 + a_k + j;
      a_reg[j] = (gr < M && gk < K) ? A[(int64_t)gr * K + gk] : 0.0;
    }
    int gk = k0 + b_k;
#pragma unroll
    for (int j = 0; j < 4; j++) {
      int gn = col0 + b_n + j;
      b_reg[j] = (gk < K && gn < N) ? B[(int64_t)gk * N + gn] : 0.0;
    }
  };
  auto write_lds = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 4; j++) As[buf][a_m][a_k + j] = a_reg[j];
#pragma unroll
    for (int j = 0; j < 4; j++) Bs[buf][b_k][b_n + j] = b_reg[j];
  };

  const int am0 = wave_m * 32;
  const int bn0 = wave_n * 32;
  auto compute_tile = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < BK; ks += 4) {
      double a0 = As[buf][am0 + l15][ks + lk];
      double a1 = As[buf][am0 + 16 + l15][ks + lk];
      double b0 = Bs[buf][ks + lk][bn0 + l15];
      double b1 = Bs[buf][ks + lk][bn0 + 16 + l15];
      acc[0][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a1, b1, acc[1][1], 0, 0, 0);
    }
  };

  issue_loads(0);
  write_lds(0);
  __syncthreads();
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    bool have_next = k0 + BK < K;
    if (have_next) issue_loads(k0 + BK);
    compute_tile(cur);
    if (have_next) {
      write_lds(cur ^ 1);
      cur ^= 1;
    }
    __syncthreads();
  }

  const int crow0 = row0 + wave_m * 32 + (lane >> 4);
  const int ccol0 = col0 + wave_n * 32 + l15;
#pragma unroll
  for (int mt = 0; mt < 2; mt++) {
#pragma unroll
    for (int nt = 0; nt < 2; nt++) {
      int col = ccol0 + nt * 16;
      if (col >= N) continue;
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        int row = crow0 + mt * 16 + 4 * reg;
        if (row < M) C[(int64_t)row * N + col] = acc[mt][nt][reg];
      }
    }
  }
}

}  // namespace

void launch_gemm_f64(const double* a, const double* b, double* c, int m, int n,
                     int k, hipStream_t stream) {
  int tiles_m = (m + BM - 1) / BM;
  int tiles_n = (n + BN - 1) / BN;
  hipLaunchKernelGGL(gemm_f64_kernel, dim3(tiles_m * tiles_n), dim3(THREADS),
                     0, stream, a, b, c, m, n, k, tiles_m, tiles_n);
  HIP_CHECK(hipGetLastError());
}
