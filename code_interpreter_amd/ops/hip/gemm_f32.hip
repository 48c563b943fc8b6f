// f32 GEMM on the gfx950 matrix cores.
//
// Uses v_mfma_f32_32x32x2_f32 (exact f32 in/accumulate at the 157 TF f32
// vector rate -- there is no TF32/xf32 on CDNA4, and this is ~2.4x an f32
// VALU kernel at identical numerics). Structure: 128x128 block tile,
// 4 waves as 2x2, each wave 2x2 tiles of 32x32 (16-reg f32 accumulators),
// BK=32 K-steps staged through LDS, A transposed into LDS at stage time so
// both MFMA operand reads are bank-conflict-free (A rows padded +1 lane).
//
// Operand layout for mfma_f32_32x32x2_f32 (one f32 VGPR per lane each):
//   A: lane l supplies A[i = l&31][k = l>>5]
//   B: lane l supplies B[k = l>>5][j = l&31]
//   C/D (16 regs): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//
// Arbitrary M, N, K: out-of-range stage reads are zero-filled, C stores are
// bounds-guarded. blockIdx is remapped XCD-aware (bijective) so neighbor
// tiles share an XCD's L2.

#include "common.h"

namespace {

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 32;
constexpr int THREADS = 256;  // 4 waves: 2x2

using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;

__global__ __launch_bounds__(THREADS) void gemm_f32_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
  // bijective XCD-aware tile remap (8 XCDs)
  int nwg = tiles_m * tiles_n;
  int wgid = blockIdx.x;
  {
    const int nxcd = 8;
    int q = nwg / nxcd, r = nwg % nxcd;
    int xcd = wgid % nxcd, idx = wgid / nxcd;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  int tile_m = wgid / tiles_n;
  int tile_n = wgid % tiles_n;
  int row0 = tile_m * BM;
  int col0 = tile_n * BN;

  // LDS: As transposed [BK][BM+1] (pad kills the k-major write conflict),
  // Bs natural [BK][BN]
  __shared__ float As[BK][BM + 1];
  __shared__ float Bs[BK][BN];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;      // 0..3
  const int wave_m = wave >> 1;   // 0..1
  const int wave_n = wave & 1;    // 0..1

  // each wave owns a 64x64 output tile: 2x2 MFMA tiles of 32x32
  f32x16 acc[2][2] = {};

  const int l31 = lane & 31;
  const int lk = lane >> 5;  // 0/1: the k-slot this lane supplies

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- stage A[row0:row0+128][k0:k0+32] -> As[k][m] (transposed) ----
    // 256 threads x 4 floats: thread t covers row m = t>>3, cols 4*(t&7)..
    {
      int m = tid >> 3;          // 0..31 x4 iterations
      int kq = (tid & 7) * 4;    // 0,4,..28
      for (int mm = m; mm < BM; mm += 32) {
        int gr = row0 + mm;
        float v0 = 0, v1 = 0, v2 = 0, v3 = 0;
        if (gr < M) {
          int gk = k0 + kq;
          const float* src = A + (int64_t)gr * K + gk;
          if (gk + 3 < K) {
            v0 = src[0]; v1 = src[1]; v2 = src[2]; v3 = src[3];
          } else {
            if (gk + 0 < K) v0 = src[0];
            if (gk + 1 < K) v1 = src[1];
            if (gk + 2 < K) v2 = src[2];
            if (gk + 3 < K) v3 = src[3];
          }
        }
        As[kq + 0][mm] = v0;
        As[kq + 1][mm] = v1;
        As[kq + 2][mm] = v2;
        As[kq + 3][mm] = v3;
      }
    }
    // ---- stage B[k0:k0+32][col0:col0+128] -> Bs[k][n] ----
    {
      int kq = tid >> 5;         // 0..7 x4 iterations
      int n = (tid & 31) * 4;    // 0,4,..124
      for (int kk = kq; kk < BK; kk += 8) {
        int gk = k0 + kk;
        float v0 = 0, v1 = 0, v2 = 0, v3 = 0;
        if (gk < K) {
          int gn = col0 + n;
          const float* src = B + (int64_t)gk * N + gn;
          if (gn + 3 < N) {
            v0 = src[0]; v1 = src[1]; v2 = src[2]; v3 = src[3];
          } else {
            if (gn + 0 < N) v0 = src[0];
            if (gn + 1 < N) v1 = src[1];
            if (gn + 2 < N) v2 = src[2];
            if (gn + 3 < N) v3 = src[3];
          }
        }
        float4 v = {v0, v1, v2, v3};
        *reinterpret_cast<float4*>(&Bs[kk][n]) = v;
      }
    }
    __syncthreads();

    // ---- MFMA inner loop: BK/2 = 16 steps of K=2 ----
    const int am0 = wave_m * 64;
    const int bn0 = wave_n * 64;
#pragma unroll
    for (int ks = 0; ks < BK; ks += 2) {
      float a0 = As[ks + lk][am0 + l31];
      float a1 = As[ks + lk][am0 + 32 + l31];
      float b0 = Bs[ks + lk][bn0 + l31];
      float b1 = Bs[ks + lk][bn0 + 32 + l31];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: C writes (bounds-guarded) ----
  const int crow0 = row0 + wave_m * 64 + 4 * (lane >> 5);
  const int ccol0 = col0 + wave_n * 64 + l31;
#pragma unroll
  for (int mt = 0; mt < 2; mt++) {
#pragma unroll
    for (int nt = 0; nt < 2; nt++) {
      int col = ccol0 + nt * 32;
      if (col >= N) continue;
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        int row = crow0 + mt * 32 + (reg & 3) + 8 * (reg >> 2);
        if (row < M) C[(int64_t)row * N + col] = acc[mt][nt][reg];
      }
    }
  }
}

}  // namespace

void launch_gemm_f32(const float* a, const float* b, float* c, int m, int n,
                     int k, hipStream_t stream) {
  int tiles_m = (m + BM - 1) / BM;
  int tiles_n = (n + BN - 1) / BN;
  hipLaunchKernelGGL(gemm_f32_kernel, dim3(tiles_m * tiles_n), dim3(THREADS),
                     0, stream, a, b, c, m, n, k, tiles_m, tiles_n);
  HIP_CHECK(hipGetLastError());
}
