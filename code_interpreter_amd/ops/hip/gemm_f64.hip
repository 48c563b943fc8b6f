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

// ---------------------------------------------------------------------------
// 128x128-tile variant: the 64x64 kernel's arithmetic intensity is
// 8 flop/B from HBM -> a 64 TF DRAM roofline, and it measures 57.8 TF =
// 90% of that WRONG roof (rocBLAS reaches 72). Doubling the tile
// doubles intensity (16 flop/B -> 128 TF cap), moving the bound back to
// the MFMA pipe. 512 threads = 8 waves as 2(M)x4(N), each wave a 64x32
// C tile (acc[4][2]); BK=16, double-buffered, 68 KB LDS -> 2 blocks/CU
// -> 4 waves/SIMD covering staging/barrier bubbles.
// ---------------------------------------------------------------------------
constexpr int BM2 = 128;
constexpr int BN2 = 128;
constexpr int THREADS2 = 512;

template <int BKT>
__global__ __launch_bounds__(THREADS2) void gemm_f64_128_kernel(
    const double* __restrict__ A, const double* __restrict__ B,
    double* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
  int tm, tn;
  {
    int nwg = tiles_m * tiles_n;
    int wgid = blockIdx.x;
    const int n_st = (tiles_m / 8) * (tiles_n / 4);
    if (tiles_m % 8 == 0 && tiles_n % 4 == 0 && n_st % 8 == 0) {
      // 2D XCD supertiling (the bf16 kernel's mapping): each XCD works
      // whole 8x4-tile supertiles so its L2 re-reads A row-bands 4x and
      // B columns 8x
      const int st_cols = tiles_n / 4;
      int xcd = wgid % 8, idx = wgid / 8;
      int st = xcd + 8 * (idx >> 5);
      int p = idx & 31;
      tm = (st / st_cols) * 8 + (p >> 2);
      tn = (st % st_cols) * 4 + (p & 3);
    } else {
      const int nxcd = 8;
      int q = nwg / nxcd, r = nwg % nxcd;
      int xcd = wgid % nxcd, idx = wgid / nxcd;
      wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
      tm = wgid / tiles_n;
      tn = wgid % tiles_n;
    }
  }
  int row0 = tm * BM2;
  int col0 = tn * BN2;

  __shared__ double As[2][BM2][BKT + 1];
  __shared__ double Bs[2][BKT][BN2 + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;        // 0..7
  const int wave_m = wave >> 2;     // 0..1 -> 64-row half
  const int wave_n = wave & 3;      // 0..3 -> 32-col band
  const int l15 = lane & 15;
  const int lk = lane >> 4;         // 0..3

  f64x4 acc[4][2] = {};

  // staging: AE = BM2*BKT/THREADS2 f64 of A, BE = BKT*BN2/THREADS2 of B
  constexpr int AE = BM2 * BKT / THREADS2;
  constexpr int BE = BKT * BN2 / THREADS2;
  const int a_m = tid / (BKT / AE);
  const int a_k = (tid % (BKT / AE)) * AE;
  const int b_k = tid / (BN2 / BE);
  const int b_n = (tid % (BN2 / BE)) * BE;

  double a_reg[AE], b_reg[BE];

  auto issue_loads = [&](int k0) {
    int gr = row0 + a_m;
#pragma unroll
    for (int j = 0; j < AE; j++) {
      int gk = k0 + a_k + j;
      a_reg[j] = (gr < M && gk < K) ? A[(int64_t)gr * K + gk] : 0.0;
    }
    int gk = k0 + b_k;
#pragma unroll
    for (int j = 0; j < BE; j++) {
      int gn = col0 + b_n + j;
      b_reg[j] = (gk < K && gn < N) ? B[(int64_t)gk * N + gn] : 0.0;
    }
  };
  auto write_lds = [&](int buf) {
#pragma unroll
    for (int j = 0; j < AE; j++) As[buf][a_m][a_k + j] = a_reg[j];
#pragma unroll
    for (int j = 0; j < BE; j++) Bs[buf][b_k][b_n + j] = b_reg[j];
  };

  const int am0 = wave_m * 64;
  const int bn0 = wave_n * 32;
  auto compute_tile = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < BKT; ks += 4) {
      double a0 = As[buf][am0 + l15][ks + lk];
      double a1 = As[buf][am0 + 16 + l15][ks + lk];
      double a2 = As[buf][am0 + 32 + l15][ks + lk];
      double a3 = As[buf][am0 + 48 + l15][ks + lk];
      double b0 = Bs[buf][ks + lk][bn0 + l15];
      double b1 = Bs[buf][ks + lk][bn0 + 16 + l15];
      // no two consecutive MFMAs share a source operand: the rate probe
      // suggests f64 MFMA stalls on back-to-back src reuse
      acc[0][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a0, b0, acc[0][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a1, b1, acc[1][1], 0, 0, 0);
      acc[2][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a2, b0, acc[2][0], 0, 0, 0);
      acc[3][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a3, b1, acc[3][1], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a1, b0, acc[1][0], 0, 0, 0);
      acc[2][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a2, b1, acc[2][1], 0, 0, 0);
      acc[3][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a3, b0, acc[3][0], 0, 0, 0);
    }
  };

  issue_loads(0);
  write_lds(0);
  __syncthreads();
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BKT) {
    bool have_next = k0 + BKT < K;
    if (have_next) issue_loads(k0 + BKT);
    compute_tile(cur);
    if (have_next) {
      write_lds(cur ^ 1);
      cur ^= 1;
    }
    __syncthreads();
  }

  const int crow0 = row0 + wave_m * 64 + (lane >> 4);
  const int ccol0 = col0 + wave_n * 32 + l15;
#pragma unroll
  for (int mt = 0; mt < 4; mt++) {
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

// ---------------------------------------------------------------------------
// 3-stage pipelined variant: the double-buffered loop above ends every
// iteration with a ds_write convoy right before the barrier (the whole
// block parks until the slowest wave finishes staging). With THREE LDS
// buffers the staging writes for tile k+1 issue BEFORE tile k's MFMAs,
// so the compiler interleaves them into the MFMA/ds_read stream, and
// the barrier only fences buffer reuse two tiles later. BK=8 keeps the
// triple buffer at ~52 KB -> 3 blocks/CU (6 waves/SIMD vs 4).
// ---------------------------------------------------------------------------
template <int BKT, bool PRIO = false>
__global__ __launch_bounds__(THREADS2) void gemm_f64_128p_kernel(
    const double* __restrict__ A, const double* __restrict__ B,
    double* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
  int tm, tn;
  {
    int nwg = tiles_m * tiles_n;
    int wgid = blockIdx.x;
    const int n_st = (tiles_m / 8) * (tiles_n / 4);
    if (tiles_m % 8 == 0 && tiles_n % 4 == 0 && n_st % 8 == 0) {
      const int st_cols = tiles_n / 4;
      int xcd = wgid % 8, idx = wgid / 8;
      int st = xcd + 8 * (idx >> 5);
      int p = idx & 31;
      tm = (st / st_cols) * 8 + (p >> 2);
      tn = (st % st_cols) * 4 + (p & 3);
    } else {
      const int nxcd = 8;
      int q = nwg / nxcd, r = nwg % nxcd;
      int xcd = wgid % nxcd, idx = wgid / nxcd;
      wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
      tm = wgid / tiles_n;
      tn = wgid % tiles_n;
    }
  }
  int row0 = tm * BM2;
  int col0 = tn * BN2;

  __shared__ double As[3][BM2][BKT + 1];
  __shared__ double Bs[3][BKT][BN2 + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 2;
  const int wave_n = wave & 3;
  const int l15 = lane & 15;
  const int lk = lane >> 4;

  f64x4 acc[4][2] = {};

  constexpr int AE = BM2 * BKT / THREADS2;
  constexpr int BE = BKT * BN2 / THREADS2;
  const int a_m = tid / (BKT / AE);
  const int a_k = (tid % (BKT / AE)) * AE;
  const int b_k = tid / (BN2 / BE);
  const int b_n = (tid % (BN2 / BE)) * BE;

  double a_reg[AE], b_reg[BE];

  auto issue_loads = [&](int k0) {
    int gr = row0 + a_m;
#pragma unroll
    for (int j = 0; j < AE; j++) {
      int gk = k0 + a_k + j;
      a_reg[j] = (gr < M && gk < K) ? A[(int64_t)gr * K + gk] : 0.0;
    }
    int gk = k0 + b_k;
#pragma unroll
    for (int j = 0; j < BE; j++) {
      int gn = col0 + b_n + j;
      b_reg[j] = (gk < K && gn < N) ? B[(int64_t)gk * N + gn] : 0.0;
    }
  };
  auto write_lds = [&](int buf) {
#pragma unroll
    for (int j = 0; j < AE; j++) As[buf][a_m][a_k + j] = a_reg[j];
#pragma unroll
    for (int j = 0; j < BE; j++) Bs[buf][b_k][b_n + j] = b_reg[j];
  };

  const int am0 = wave_m * 64;
  const int bn0 = wave_n * 32;
  auto compute_tile = [&](int buf) {
    if (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < BKT; ks += 4) {
      double a0 = As[buf][am0 + l15][ks + lk];
      double a1 = As[buf][am0 + 16 + l15][ks + lk];
      double a2 = As[buf][am0 + 32 + l15][ks + lk];
      double a3 = As[buf][am0 + 48 + l15][ks + lk];
      double b0 = Bs[buf][ks + lk][bn0 + l15];
      double b1 = Bs[buf][ks + lk][bn0 + 16 + l15];
      acc[0][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a0, b0, acc[0][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a1, b1, acc[1][1], 0, 0, 0);
      acc[2][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a2, b0, acc[2][0], 0, 0, 0);
      acc[3][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a3, b1, acc[3][1], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a1, b0, acc[1][0], 0, 0, 0);
      acc[2][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a2, b1, acc[2][1], 0, 0, 0);
      acc[3][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a3, b0, acc[3][0], 0, 0, 0);
    }
    if (PRIO) __builtin_amdgcn_s_setprio(0);
  };

  // prologue: buffer 0 staged and visible, buffer 1's loads in regs
  issue_loads(0);
  write_lds(0);
  if (BKT < K) issue_loads(BKT);
  __syncthreads();
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BKT) {
    bool have_next = k0 + BKT < K;
    if (have_next) {
      // stage k0+BKT NOW (its loads are already in registers); its
      // buffer was last read two barriers ago, so this is safe, and
      // these ds_writes overlap the MFMA stream below
      write_lds((cur + 1) % 3);
      if (k0 + 2 * BKT < K) issue_loads(k0 + 2 * BKT);
    }
    compute_tile(cur);
    cur = (cur + 1) % 3;
    __syncthreads();
  }

  const int crow0 = row0 + wave_m * 64 + (lane >> 4);
  const int ccol0 = col0 + wave_n * 32 + l15;
#pragma unroll
  for (int mt = 0; mt < 4; mt++) {
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

// ---------------------------------------------------------------------------
// 4-wave variant: 128^2 block as 2x2 waves of 64x64 C each -> 16 MFMAs
// per 8 ds_reads (2:1, vs 1.33:1 for the 8-wave shape). Probes whether
// the residual bubble is operand-read latency. BK=8, 3 LDS buffers
// (~52 KB -> 3 blocks/CU = 3 waves/SIMD; acc = 128 VGPRs).
// ---------------------------------------------------------------------------
constexpr int THREADS3 = 256;

template <int BKT>
__global__ __launch_bounds__(THREADS3) void gemm_f64_128w_kernel(
    const double* __restrict__ A, const double* __restrict__ B,
    double* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
  int tm, tn;
  {
    int nwg = tiles_m * tiles_n;
    int wgid = blockIdx.x;
    const int n_st = (tiles_m / 8) * (tiles_n / 4);
    if (tiles_m % 8 == 0 && tiles_n % 4 == 0 && n_st % 8 == 0) {
      const int st_cols = tiles_n / 4;
      int xcd = wgid % 8, idx = wgid / 8;
      int st = xcd + 8 * (idx >> 5);
      int p = idx & 31;
      tm = (st / st_cols) * 8 + (p >> 2);
      tn = (st % st_cols) * 4 + (p & 3);
    } else {
      const int nxcd = 8;
      int q = nwg / nxcd, r = nwg % nxcd;
      int xcd = wgid % nxcd, idx = wgid / nxcd;
      wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
      tm = wgid / tiles_n;
      tn = wgid % tiles_n;
    }
  }
  int row0 = tm * BM2;
  int col0 = tn * BN2;

  __shared__ double As[3][BM2][BKT + 1];
  __shared__ double Bs[3][BKT][BN2 + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;        // 0..3
  const int wave_m = wave >> 1;     // 2x2 arrangement
  const int wave_n = wave & 1;
  const int l15 = lane & 15;
  const int lk = lane >> 4;

  f64x4 acc[4][4] = {};

  constexpr int AE = BM2 * BKT / THREADS3;
  constexpr int BE = BKT * BN2 / THREADS3;
  const int a_m = tid / (BKT / AE);
  const int a_k = (tid % (BKT / AE)) * AE;
  const int b_k = tid / (BN2 / BE);
  const int b_n = (tid % (BN2 / BE)) * BE;

  double a_reg[AE], b_reg[BE];

  auto issue_loads = [&](int k0) {
    int gr = row0 + a_m;
#pragma unroll
    for (int j = 0; j < AE; j++) {
      int gk = k0 + a_k + j;
      a_reg[j] = (gr < M && gk < K) ? A[(int64_t)gr * K + gk] : 0.0;
    }
    int gk = k0 + b_k;
#pragma unroll
    for (int j = 0; j < BE; j++) {
      int gn = col0 + b_n + j;
      b_reg[j] = (gk < K && gn < N) ? B[(int64_t)gk * N + gn] : 0.0;
    }
  };
  auto write_lds = [&](int buf) {
#pragma unroll
    for (int j = 0; j < AE; j++) As[buf][a_m][a_k + j] = a_reg[j];
#pragma unroll
    for (int j = 0; j < BE; j++) Bs[buf][b_k][b_n + j] = b_reg[j];
  };

  const int am0 = wave_m * 64;
  const int bn0 = wave_n * 64;
  auto compute_tile = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < BKT; ks += 4) {
      double a0 = As[buf][am0 + l15][ks + lk];
      double a1 = As[buf][am0 + 16 + l15][ks + lk];
      double a2 = As[buf][am0 + 32 + l15][ks + lk];
      double a3 = As[buf][am0 + 48 + l15][ks + lk];
      double b0 = Bs[buf][ks + lk][bn0 + l15];
      double b1 = Bs[buf][ks + lk][bn0 + 16 + l15];
      double b2 = Bs[buf][ks + lk][bn0 + 32 + l15];
      double b3 = Bs[buf][ks + lk][bn0 + 48 + l15];
#pragma unroll
      for (int d = 0; d < 4; d++) {
        // diagonal order: consecutive MFMAs never share a source
        acc[0][d] = __builtin_amdgcn_mfma_f64_16x16x4f64(
            a0, d == 0 ? b0 : d == 1 ? b1 : d == 2 ? b2 : b3, acc[0][d], 0, 0, 0);
        acc[1][(d + 1) & 3] = __builtin_amdgcn_mfma_f64_16x16x4f64(
            a1, d == 3 ? b0 : d == 0 ? b1 : d == 1 ? b2 : b3, acc[1][(d + 1) & 3], 0, 0, 0);
        acc[2][(d + 2) & 3] = __builtin_amdgcn_mfma_f64_16x16x4f64(
            a2, d == 2 ? b0 : d == 3 ? b1 : d == 0 ? b2 : b3, acc[2][(d + 2) & 3], 0, 0, 0);
        acc[3][(d + 3) & 3] = __builtin_amdgcn_mfma_f64_16x16x4f64(
            a3, d == 1 ? b0 : d == 2 ? b1 : d == 3 ? b2 : b3, acc[3][(d + 3) & 3], 0, 0, 0);
      }
    }
  };

  issue_loads(0);
  write_lds(0);
  if (BKT < K) issue_loads(BKT);
  __syncthreads();
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BKT) {
    bool have_next = k0 + BKT < K;
    if (have_next) {
      write_lds((cur + 1) % 3);
      if (k0 + 2 * BKT < K) issue_loads(k0 + 2 * BKT);
    }
    compute_tile(cur);
    cur = (cur + 1) % 3;
    __syncthreads();
  }

  const int crow0 = row0 + wave_m * 64 + (lane >> 4);
  const int ccol0 = col0 + wave_n * 64 + l15;
#pragma unroll
  for (int mt = 0; mt < 4; mt++) {
#pragma unroll
    for (int nt = 0; nt < 4; nt++) {
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

// ---------------------------------------------------------------------------
// glds variant: stages A and B via global_load_lds (16 B/lane LDS-DMA,
// lane-linear), eliminating the staging registers AND the ds_write
// convoy -- the PMC breakdown shows the shipped pipeline is VGPR-capped
// at 4 waves/SIMD with 23% of wave cycles parked on vmcnt/barriers.
// Without a_reg/b_reg the kernel fits 5 waves/SIMD. glds's lane-linear
// LDS image cannot take the +1 padding, so the operand layout is
// XOR-swizzled instead:
//   A slot(r, kpair) = r*4 + (kpair ^ ((r>>2)&3))   (16-B slots)
//   B slot(k, cpair) = k*64 + (cpair ^ 8k)
// Both make every 64-lane ds_read_b64 operand fetch hit each qword
// bank exactly twice (the b64 minimum). Requires M,N %128==0, K%8==0;
// the launcher falls back to the 'p' kernel otherwise.
// ---------------------------------------------------------------------------
using lds_void_f64 = __attribute__((address_space(3))) void;
using global_void_f64 = const __attribute__((address_space(1))) void;

template <int MINB, int NBUF = 3>
__global__ __launch_bounds__(THREADS2, MINB) void gemm_f64_glds_kernel(
    const double* __restrict__ A, const double* __restrict__ B,
    double* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];  // 3 x 16 KB
  int tm, tn;
  {
    int nwg = tiles_m * tiles_n;
    int wgid = blockIdx.x;
    const int n_st = (tiles_m / 8) * (tiles_n / 4);
    if (tiles_m % 8 == 0 && tiles_n % 4 == 0 && n_st % 8 == 0) {
      const int st_cols = tiles_n / 4;
      int xcd = wgid % 8, idx = wgid / 8;
      int st = xcd + 8 * (idx >> 5);
      int p = idx & 31;
      tm = (st / st_cols) * 8 + (p >> 2);
      tn = (st % st_cols) * 4 + (p & 3);
    } else {
      const int nxcd = 8;
      int q = nwg / nxcd, r = nwg % nxcd;
      int xcd = wgid % nxcd, idx = wgid / nxcd;
      wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
      tm = wgid / tiles_n;
      tn = wgid % tiles_n;
    }
  }
  const int row0 = tm * BM2;
  const int col0 = tn * BN2;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;    // 0..7
  const int wave_m = wave >> 2;
  const int wave_n = wave & 3;
  const int l15 = lane & 15;
  const int lk = lane >> 4;

  f64x4 acc[4][2] = {};

  // per-thread staging sources (fixed across tiles except k0)
  const int a_r = tid >> 2;
  const int a_q = (tid & 3) ^ ((tid >> 4) & 3);
  const double* a_src0 = A + (int64_t)(row0 + a_r) * K + 2 * a_q;
  const int b_cq = lane ^ (wave * 8);
  const double* b_src0 = B + (int64_t)wave * N + col0 + 2 * b_cq;

  auto stage = [&](int ktile) {
    int buf = ktile % NBUF;
    int k0 = ktile * 8;
    unsigned dst = (unsigned)buf * 16384u + (unsigned)wave * 1024u;
    __builtin_amdgcn_global_load_lds(
        (global_void_f64*)(a_src0 + k0), (lds_void_f64*)(smem + dst), 16, 0,
        0);
    __builtin_amdgcn_global_load_lds(
        (global_void_f64*)(b_src0 + (int64_t)k0 * N),
        (lds_void_f64*)(smem + 8192u + dst), 16, 0, 0);
  };

  auto a_val = [&](int buf, int row, int k) -> double {
    unsigned slot = (unsigned)row * 4u +
                    (unsigned)((k >> 1) ^ ((row >> 2) & 3));
    return *reinterpret_cast<const double*>(
        smem + (unsigned)buf * 16384u + slot * 16u + (unsigned)(k & 1) * 8u);
  };
  auto b_val = [&](int buf, int k, int col) -> double {
    unsigned slot = (unsigned)k * 64u + (unsigned)((col >> 1) ^ (k * 8));
    return *reinterpret_cast<const double*>(
        smem + (unsigned)buf * 16384u + 8192u + slot * 16u +
        (unsigned)(col & 1) * 8u);
  };

  const int am0 = wave_m * 64;
  const int bn0 = wave_n * 32;
  auto compute_tile = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < 8; ks += 4) {
      int k = ks + lk;
      double a0 = a_val(buf, am0 + l15, k);
      double a1 = a_val(buf, am0 + 16 + l15, k);
      double a2 = a_val(buf, am0 + 32 + l15, k);
      double a3 = a_val(buf, am0 + 48 + l15, k);
      double b0 = b_val(buf, k, bn0 + l15);
      double b1 = b_val(buf, k, bn0 + 16 + l15);
      acc[0][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a0, b0, acc[0][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a1, b1, acc[1][1], 0, 0, 0);
      acc[2][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a2, b0, acc[2][0], 0, 0, 0);
      acc[3][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a3, b1, acc[3][1], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a1, b0, acc[1][0], 0, 0, 0);
      acc[2][1] = __builtin_amdgcn_mfma_f64_16x16x4f64(a2, b1, acc[2][1], 0, 0, 0);
      acc[3][0] = __builtin_amdgcn_mfma_f64_16x16x4f64(a3, b0, acc[3][0], 0, 0, 0);
    }
  };

  const int n_ktiles = K / 8;
  const int depth = NBUF - 1;  // tiles in flight beyond the current one
  for (int t = 0; t < depth && t < n_ktiles; t++) stage(t);
  {
    int inflight = n_ktiles > 1 ? 2 * (std::min(depth, n_ktiles) - 1) : 0;
    if (inflight >= 6)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if (inflight >= 4)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (inflight >= 2)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  for (int kt = 0; kt < n_ktiles; kt++) {
    if (kt + depth < n_ktiles) stage(kt + depth);
    compute_tile(kt % NBUF);
    if (kt + 1 < n_ktiles) {
      // retire everything except the loads issued after tile kt+1's
      int pending = 0;
      for (int t = kt + 2; t <= kt + depth && t < n_ktiles; t++) pending += 2;
      if (pending >= 6)
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      else if (pending >= 4)
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else if (pending >= 2)
        asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  const int crow0 = row0 + wave_m * 64 + (lane >> 4);
  const int ccol0 = col0 + wave_n * 32 + l15;
#pragma unroll
  for (int mt = 0; mt < 4; mt++) {
#pragma unroll
    for (int nt = 0; nt < 2; nt++) {
      int col = ccol0 + nt * 16;
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        int row = crow0 + mt * 16 + 4 * reg;
        C[(int64_t)row * N + col] = acc[mt][nt][reg];
      }
    }
  }
}

}  // namespace

void launch_gemm_f64(const double* a, const double* b, double* c, int m, int n,
                     int k, hipStream_t stream) {
  // default: 128^2 tile, BK=8, 3-buffer write-early pipeline (63.7 TF
  // @8192^3 — best of the r02 ladder; profiles/NOTES.md). APP_F64_VARIANT:
  // "d" = BK=16 double-buffer (prior default), "s" = the old 64^2 kernel,
  // "8"/"32" = BK sweep on the double-buffered 128^2 tile, "q" = BK=16
  // 3-buffer (1 block/CU: occupancy loss), "w" = 4-wave 64x64 wave tiles
  // (2:1 MFMA:ds_read but 3 waves/SIMD: latency-coverage loss)
  const char* v = getenv("APP_F64_VARIANT");
  if (!(v && v[0] == 's')) {
    int tiles_m2 = (m + BM2 - 1) / BM2;
    int tiles_n2 = (n + BN2 - 1) / BN2;
    bool full_tiles = (m % BM2 == 0) && (n % BN2 == 0) && (k % 8 == 0);
    // default for full 128-multiple tiles: the 4-buffer glds kernel
    // (67.7 TF @8192^3 same-box vs 63.5 for the register-staged
    // pipeline; depth ladder in profiles/NOTES.md)
    if (full_tiles && !v) {
      hipLaunchKernelGGL((gemm_f64_glds_kernel<1, 4>),
                         dim3(tiles_m2 * tiles_n2), dim3(THREADS2),
                         4 * 16384, stream, a, b, c, m, n, k, tiles_m2,
                         tiles_n2);
      HIP_CHECK(hipGetLastError());
      return;
    }
    if (v && v[0] == 'g' && full_tiles) {
      if (v[1] == '5')
        hipLaunchKernelGGL((gemm_f64_glds_kernel<1, 5>),
                           dim3(tiles_m2 * tiles_n2), dim3(THREADS2),
                           5 * 16384, stream, a, b, c, m, n, k, tiles_m2,
                           tiles_n2);
      else if (v[1] == '4')
        hipLaunchKernelGGL((gemm_f64_glds_kernel<1, 4>),
                           dim3(tiles_m2 * tiles_n2), dim3(THREADS2),
                           4 * 16384, stream, a, b, c, m, n, k, tiles_m2,
                           tiles_n2);
      else
        hipLaunchKernelGGL((gemm_f64_glds_kernel<1, 3>),
                           dim3(tiles_m2 * tiles_n2), dim3(THREADS2),
                           3 * 16384, stream, a, b, c, m, n, k, tiles_m2,
                           tiles_n2);
      HIP_CHECK(hipGetLastError());
      return;
    }
    if (v && v[0] == 'P')
      hipLaunchKernelGGL((gemm_f64_128p_kernel<8, true>),
                         dim3(tiles_m2 * tiles_n2), dim3(THREADS2), 0, stream,
                         a, b, c, m, n, k, tiles_m2, tiles_n2);
    else if (v && v[0] == 'w')
      hipLaunchKernelGGL((gemm_f64_128w_kernel<8>), dim3(tiles_m2 * tiles_n2),
                         dim3(THREADS3), 0, stream, a, b, c, m, n, k,
                         tiles_m2, tiles_n2);
    else if (v && v[0] == 'q')
      hipLaunchKernelGGL((gemm_f64_128p_kernel<16>), dim3(tiles_m2 * tiles_n2),
                         dim3(THREADS2), 0, stream, a, b, c, m, n, k,
                         tiles_m2, tiles_n2);
    else if (v && v[0] == '8')
      hipLaunchKernelGGL((gemm_f64_128_kernel<8>), dim3(tiles_m2 * tiles_n2),
                         dim3(THREADS2), 0, stream, a, b, c, m, n, k,
                         tiles_m2, tiles_n2);
    else if (v && v[0] == '3')
      hipLaunchKernelGGL((gemm_f64_128_kernel<32>), dim3(tiles_m2 * tiles_n2),
                         dim3(THREADS2), 0, stream, a, b, c, m, n, k,
                         tiles_m2, tiles_n2);
    else if (v && v[0] == 'd')
      hipLaunchKernelGGL((gemm_f64_128_kernel<16>), dim3(tiles_m2 * tiles_n2),
                         dim3(THREADS2), 0, stream, a, b, c, m, n, k,
                         tiles_m2, tiles_n2);
    else
      hipLaunchKernelGGL((gemm_f64_128p_kernel<8>), dim3(tiles_m2 * tiles_n2),
                         dim3(THREADS2), 0, stream, a, b, c, m, n, k,
                         tiles_m2, tiles_n2);
    HIP_CHECK(hipGetLastError());
    return;
  }
  int tiles_m = (m + BM - 1) / BM;
  int tiles_n = (n + BN - 1) / BN;
  hipLaunchKernelGGL(gemm_f64_kernel, dim3(tiles_m * tiles_n), dim3(THREADS),
                     0, stream, a, b, c, m, n, k, tiles_m, tiles_n);
  HIP_CHECK(hipGetLastError());
}
