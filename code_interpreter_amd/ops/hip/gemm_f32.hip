// f32 GEMM on the gfx950 matrix cores.
//
// Uses v_mfma_f32_32x32x2_f32 (exact f32 in/accumulate at the 157 TF f32
// vector rate -- no TF32/xf32 on CDNA4; ~2.4x an f32 VALU kernel at
// identical numerics). Structure: 128x128 block tile, 4 waves as 2x2,
// each wave 2x2 tiles of 32x32 (16-reg f32 accumulators), BK=32 K-steps,
// double-buffered LDS with the async-stage split (T14): the next tile's
// global loads are issued BEFORE this tile's MFMA loop (HBM latency
// hides under compute), and written to the other LDS buffer after a
// counted drain -- one barrier per K-tile. A is transposed into LDS at
// write time (As[k][m], +1 pad) so both operand reads are
// bank-conflict-free.
//
// Operand layout for mfma_f32_32x32x2_f32 (one f32 per lane each):
//   A: lane l supplies A[i = l&31][k = l>>5]
//   B: lane l supplies B[k = l>>5][j = l&31]
//   C/D (16 regs): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//
// Arbitrary M, N, K (bounds-checked staging + guarded C stores);
// XCD-aware bijective tile remap.

#include "common.h"

namespace {

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int BK = 32;
constexpr int THREADS = 256;

using f32x16 = __attribute__((__vector_size__(16 * sizeof(float)))) float;

__global__ __launch_bounds__(THREADS) void gemm_f32_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
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

  // double-buffered: As transposed [BK][BM+1], Bs natural [BK][BN]
  __shared__ float As[2][BK][BM + 1];
  __shared__ float Bs[2][BK][BN];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;
  const int wave_n = wave & 1;
  const int l31 = lane & 31;
  const int lk = lane >> 5;

  f32x16 acc[2][2] = {};

  // staging registers: 4 float4 of A (rows tid>>3 + {0,32,64,96}, cols
  // 4*(tid&7)) and 4 float4 of B (rows tid>>5 + {0,8,16,24}, cols
  // 4*(tid&31))
  const int a_m = tid >> 3;
  const int a_k = (tid & 7) * 4;
  const int b_k = tid >> 5;
  const int b_n = (tid & 31) * 4;

  float4 a_reg[4], b_reg[4];

  auto issue_loads = [&](int k0) {
#pragma unroll
    for (int i = 0; i < 4; i++) {
      int gr = row0 + a_m + i * 32;
      int gk = k0 + a_k;
      if (gr < M && gk + 3 < K) {
        a_reg[i] = *reinterpret_cast<const float4*>(&A[(int64_t)gr * K + gk]);
      } else {
        float v[4];
#pragma unroll
        for (int j = 0; j < 4; j++)
          v[j] = (gr < M && gk + j < K) ? A[(int64_t)gr * K + gk + j] : 0.0f;
        a_reg[i] = {v[0], v[1], v[2], v[3]};
      }
    }
#pragma unroll
    for (int i = 0; i < 4; i++) {
      int gk = k0 + b_k + i * 8;
      int gn = col0 + b_n;
      if (gk < K && gn + 3 < N) {
        b_reg[i] = *reinterpret_cast<const float4*>(&B[(int64_t)gk * N + gn]);
      } else {
        float v[4];
#pragma unroll
        for (int j = 0; j < 4; j++)
          v[j] = (gk < K && gn + j < N) ? B[(int64_t)gk * N + gn + j] : 0.0f;
        b_reg[i] = {v[0], v[1], v[2], v[3]};
      }
    }
  };

  auto write_lds = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 4; i++) {
      As[buf][a_k + 0][a_m + i * 32] = a_reg[i].x;
      As[buf][a_k + 1][a_m + i * 32] = a_reg[i].y;
      As[buf][a_k + 2][a_m + i * 32] = a_reg[i].z;
      As[buf][a_k + 3][a_m + i * 32] = a_reg[i].w;
    }
#pragma unroll
    for (int i = 0; i < 4; i++) {
      *reinterpret_cast<float4*>(&Bs[buf][b_k + i * 8][b_n]) = b_reg[i];
    }
  };

  const int am0 = wave_m * 64;
  const int bn0 = wave_n * 64;

  auto compute_tile = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < BK; ks += 2) {
      float a0 = As[buf][ks + lk][am0 + l31];
      float a1 = As[buf][ks + lk][am0 + 32 + l31];
      float b0 = Bs[buf][ks + lk][bn0 + l31];
      float b1 = Bs[buf][ks + lk][bn0 + 32 + l31];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
  };

  // prologue: tile 0 staged synchronously
  issue_loads(0);
  write_lds(0);
  __syncthreads();

  // one barrier per K-tile: the write targets the OTHER buffer, whose
  // last cross-wave readers were separated by the previous iteration's
  // barrier; this iteration's readers only touch buf cur
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    bool have_next = k0 + BK < K;
    if (have_next) issue_loads(k0 + BK);  // HBM latency hides under MFMA
    compute_tile(cur);
    if (have_next) {
      write_lds(cur ^ 1);
      cur ^= 1;
    }
    __syncthreads();
  }

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

// ---------------------------------------------------------------------------
// BK=16 variant: half the LDS (33 KB/block) -> 4 blocks/CU -> 4 waves
// per SIMD. The MFMA-rate probe (scripts/mfma_rate_probe.cpp) measures
// the f32 pipe ceiling 156 TF reachable from 2 waves/SIMD; the BK=32
// kernel at exactly 2 waves/SIMD leaves no thread-level slack to cover
// barrier/staging bubbles -- this variant trades 2x barrier frequency
// for 2x the waves covering them.
// ---------------------------------------------------------------------------
template <int BKT>
__global__ __launch_bounds__(THREADS) void gemm_f32_bk16_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
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
  int row0 = tm * BM;
  int col0 = tn * BN;

  __shared__ float As[2][BKT][BM + 1];
  __shared__ float Bs[2][BKT][BN];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;
  const int wave_n = wave & 1;
  const int l31 = lane & 31;
  const int lk = lane >> 5;

  f32x16 acc[2][2] = {};

  // staging: AQ = BM*BKT/THREADS/4 float4 of A, BQ = BKT*BN/THREADS/4
  // of B per thread (AQ=BQ=2 at BKT=16, 1 at BKT=8)
  constexpr int AQ = BM * BKT / THREADS / 4;
  constexpr int BQ = BKT * BN / THREADS / 4;
  constexpr int KV = BKT / 4;            // float4 columns per A row
  const int a_m = tid / KV;              // BM/AQ rows per chunk
  const int a_k = (tid % KV) * 4;
  constexpr int NV = BN / 4;             // float4 columns per B row
  const int b_k = tid / NV;
  const int b_n = (tid % NV) * 4;
  constexpr int AROWS = THREADS / KV;    // rows covered per i-step
  constexpr int BROWS = THREADS / NV;

  float4 a_reg[AQ], b_reg[BQ];

  auto issue_loads = [&](int k0) {
#pragma unroll
    for (int i = 0; i < AQ; i++) {
      int gr = row0 + a_m + i * AROWS;
      int gk = k0 + a_k;
      if (gr < M && gk + 3 < K) {
        a_reg[i] = *reinterpret_cast<const float4*>(&A[(int64_t)gr * K + gk]);
      } else {
        float v[4];
#pragma unroll
        for (int j = 0; j < 4; j++)
          v[j] = (gr < M && gk + j < K) ? A[(int64_t)gr * K + gk + j] : 0.0f;
        a_reg[i] = {v[0], v[1], v[2], v[3]};
      }
    }
#pragma unroll
    for (int i = 0; i < BQ; i++) {
      int gk = k0 + b_k + i * BROWS;
      int gn = col0 + b_n;
      if (gk < K && gn + 3 < N) {
        b_reg[i] = *reinterpret_cast<const float4*>(&B[(int64_t)gk * N + gn]);
      } else {
        float v[4];
#pragma unroll
        for (int j = 0; j < 4; j++)
          v[j] = (gk < K && gn + j < N) ? B[(int64_t)gk * N + gn + j] : 0.0f;
        b_reg[i] = {v[0], v[1], v[2], v[3]};
      }
    }
  };

  auto write_lds = [&](int buf) {
#pragma unroll
    for (int i = 0; i < AQ; i++) {
      As[buf][a_k + 0][a_m + i * AROWS] = a_reg[i].x;
      As[buf][a_k + 1][a_m + i * AROWS] = a_reg[i].y;
      As[buf][a_k + 2][a_m + i * AROWS] = a_reg[i].z;
      As[buf][a_k + 3][a_m + i * AROWS] = a_reg[i].w;
    }
#pragma unroll
    for (int i = 0; i < BQ; i++) {
      *reinterpret_cast<float4*>(&Bs[buf][b_k + i * BROWS][b_n]) = b_reg[i];
    }
  };

  const int am0 = wave_m * 64;
  const int bn0 = wave_n * 64;

  auto compute_tile = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < BKT; ks += 2) {
      float a0 = As[buf][ks + lk][am0 + l31];
      float a1 = As[buf][ks + lk][am0 + 32 + l31];
      float b0 = Bs[buf][ks + lk][bn0 + l31];
      float b1 = Bs[buf][ks + lk][bn0 + 32 + l31];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
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

// ---------------------------------------------------------------------------
// B-glds variant: B staged via global_load_lds into the PLAIN [BK][128]
// image (glds of row-major B rows needs no swizzle: the 64-lane b32
// operand read is exactly 2-way, the minimum), removing B's staging
// registers and ds_write convoy; A keeps the register path (the
// 32x32x2f32 A-read's k-parity split fights any 16-B-slot glds image).
// Full 128-multiple tiles only; the launcher falls back otherwise.
// ---------------------------------------------------------------------------
using lds_void_f32 = __attribute__((address_space(3))) void;
using global_void_f32 = const __attribute__((address_space(1))) void;

template <int BKT>
__global__ __launch_bounds__(THREADS) void gemm_f32_gldsb_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int N, int K, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char bsmem[];  // 2 x BKT*512
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
  int row0 = tm * BM;
  int col0 = tn * BN;

  __shared__ float As[2][BKT][BM + 1];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wave_m = wave >> 1;
  const int wave_n = wave & 1;
  const int l31 = lane & 31;
  const int lk = lane >> 5;

  f32x16 acc[2][2] = {};

  constexpr int AQ = BM * BKT / THREADS / 4;
  constexpr int KV = BKT / 4;
  const int a_m = tid / KV;
  const int a_k = (tid % KV) * 4;
  constexpr int AROWS = THREADS / KV;

  float4 a_reg[AQ];

  auto issue_a = [&](int k0) {
#pragma unroll
    for (int i = 0; i < AQ; i++) {
      int gr = row0 + a_m + i * AROWS;
      a_reg[i] =
          *reinterpret_cast<const float4*>(&A[(int64_t)gr * K + k0 + a_k]);
    }
  };
  auto write_a = [&](int buf) {
#pragma unroll
    for (int i = 0; i < AQ; i++) {
      As[buf][a_k + 0][a_m + i * AROWS] = a_reg[i].x;
      As[buf][a_k + 1][a_m + i * AROWS] = a_reg[i].y;
      As[buf][a_k + 2][a_m + i * AROWS] = a_reg[i].z;
      As[buf][a_k + 3][a_m + i * AROWS] = a_reg[i].w;
    }
  };
  // B: BKT*128 floats = BKT*32 16-B slots; slot s covers
  // (k = s>>5, n4 = s&31); wave w's glds instr j covers
  // s = j*256 + w*64 + lane (plain layout: lds addr = s*16)
  constexpr unsigned kBBuf = (unsigned)BKT * 512u;
  auto stage_b = [&](int ktile) {
    unsigned base = (unsigned)(ktile & 1) * kBBuf;
    int k0 = ktile * BKT;
#pragma unroll
    for (int j = 0; j < BKT * 32 / THREADS; j++) {
      int slot = j * THREADS + wave * 64;  // + lane, implicit in glds
      int k = (slot + lane) >> 5;
      int n4 = (slot + lane) & 31;
      const float* gsrc = B + (int64_t)(k0 + k) * N + col0 + n4 * 4;
      __builtin_amdgcn_global_load_lds(
          (global_void_f32*)gsrc,
          (lds_void_f32*)(bsmem + base + (unsigned)slot * 16u), 16, 0, 0);
    }
  };
  auto b_val = [&](int buf, int k, int n) -> float {
    return *reinterpret_cast<const float*>(
        bsmem + (unsigned)buf * kBBuf + (unsigned)k * 512u + (unsigned)n * 4u);
  };

  const int am0 = wave_m * 64;
  const int bn0 = wave_n * 64;
  auto compute_tile = [&](int buf) {
#pragma unroll
    for (int ks = 0; ks < BKT; ks += 2) {
      float a0 = As[buf][ks + lk][am0 + l31];
      float a1 = As[buf][ks + lk][am0 + 32 + l31];
      float b0 = b_val(buf, ks + lk, bn0 + l31);
      float b1 = b_val(buf, ks + lk, bn0 + 32 + l31);
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
  };

  stage_b(0);
  issue_a(0);
  write_a(0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BKT) {
    bool have_next = k0 + BKT < K;
    if (have_next) {
      stage_b((k0 / BKT) + 1);  // overlaps the MFMA stream below
      issue_a(k0 + BKT);
    }
    compute_tile(cur);
    if (have_next) {
      write_a(cur ^ 1);
      cur ^= 1;
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  const int crow0 = row0 + wave_m * 64 + 4 * (lane >> 5);
  const int ccol0 = col0 + wave_n * 64 + l31;
#pragma unroll
  for (int mt = 0; mt < 2; mt++) {
#pragma unroll
    for (int nt = 0; nt < 2; nt++) {
      int col = ccol0 + nt * 32;
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        int row = crow0 + mt * 32 + (reg & 3) + 8 * (reg >> 2);
        C[(int64_t)row * N + col] = acc[mt][nt][reg];
      }
    }
  }
}

}  // namespace

void launch_gemm_f32(const float* a, const float* b, float* c, int m, int n,
                     int k, hipStream_t stream) {
  int tiles_m = (m + BM - 1) / BM;
  int tiles_n = (n + BN - 1) / BN;
  // default: the BK=16 high-occupancy kernel (124.8 TF @8192 vs 108 for
  // BK=32 -- same-box A/B, profiles/NOTES.md r02); APP_F32_VARIANT=k32
  // keeps the old kernel for comparison
  const char* v = getenv("APP_F32_VARIANT");
  bool full_tiles = (m % BM == 0) && (n % BN == 0) && (k % 16 == 0);
  if (v && v[0] == 'g' && full_tiles) {
    hipLaunchKernelGGL((gemm_f32_gldsb_kernel<16>), dim3(tiles_m * tiles_n),
                       dim3(THREADS), 2 * 16 * 512, stream, a, b, c, m, n, k,
                       tiles_m, tiles_n);
    HIP_CHECK(hipGetLastError());
    return;
  }
  if (v && v[0] == 'k') {
    hipLaunchKernelGGL(gemm_f32_kernel, dim3(tiles_m * tiles_n), dim3(THREADS),
                       0, stream, a, b, c, m, n, k, tiles_m, tiles_n);
  } else if (v && v[0] == '8') {
    hipLaunchKernelGGL((gemm_f32_bk16_kernel<8>), dim3(tiles_m * tiles_n),
                       dim3(THREADS), 0, stream, a, b, c, m, n, k, tiles_m,
                       tiles_n);
  } else {
    hipLaunchKernelGGL((gemm_f32_bk16_kernel<16>), dim3(tiles_m * tiles_n),
                       dim3(THREADS), 0, stream, a, b, c, m, n, k, tiles_m,
                       tiles_n);
  }
  HIP_CHECK(hipGetLastError());
}
