// bf16 GEMM, 256^2-tile 8-phase structure for gfx950.
//
// The fast MFMA GEMM shape for this chip (CDNA4 guide's verified
// template class): 256x256 block tile, BK=64, 512 threads as 8 waves,
// one workgroup per CU (128 KiB LDS), double-buffered K-tiles filled by
// global_load_lds_dwordx4 (direct-to-LDS DMA, 16 B/lane), raw
// s_barrier + counted s_waitcnt vmcnt(N) so prefetched half-tiles stay
// in flight ACROSS barriers, st_16x32 XOR swizzle applied to BOTH the
// glds source position and the ds_read_b128 offset (same involution) to
// kill bank conflicts, s_setprio(1) around each MFMA cluster, XCD-aware
// bijective tile remap.
//
// Both operands are consumed K-contiguous: A is row-major [M][K]; B is
// pre-transposed to Bt[N][K] by transpose_bf16 (driven by the launcher)
// so the LDS images stay lane-linear for glds.
//
// Schedule (4 phases per K-tile X, cooperative 128x128 C-quadrants:
// all 8 waves work on one quadrant per phase):
//   ph1: Q(0,0)  reads A-half0+B-half0   stages B-half0 of tile X+1
//   ph2: Q(0,1)  reads B-half1 (A reuse)  stages A-half1 of tile X+1
//   ph3: Q(1,1)  reads A-half1 (B reuse)  stages A-half0 of tile X+2
//   ph4: Q(1,0)  reads B-half0 (A reuse)  stages B-half1 of tile X+2
//        + s_waitcnt vmcnt(4) (once per K-tile)
// TWO barriers per K-tile (not per phase): ph4's barrier (after the
// per-wave vmcnt) publishes the next tile's staged halves before any
// wave reads them; ph3's barrier fences this tile's ph1/ph2 cross-wave
// reads before the ph3/ph4 stages overwrite those slots. ph1/ph2 need no
// barrier: their reads hit the buffer published at the previous ph4, and
// their stages target slots last read before that same barrier. Each
// MFMA cluster is guarded by its own per-wave lgkmcnt(0) +
// sched_barrier(0) (hipcc can hoist register-only MFMAs past an
// inline-asm wait, guide rule 18).
// Slot-reuse: a stage targets a slot only >= 1 barrier after its last
// read; landing: vmcnt(4) at each ph4 retires every stage older than
// the last two, which covers every read deadline (B0/A1 of X+1 staged
// at X.ph1/2 are retired by X.ph4's wait; A0/B1 of X+2 staged at
// X.ph3/4 are retired by X+1.ph4's wait, read at X+2.ph1/2).
//
// Fast path requires M%256==0, N%256==0, K%128==0 (dispatch falls back
// to the general 128^2 kernel otherwise).

#include "common.h"

#include <hip/hip_bf16.h>

namespace {

constexpr int BM = 256;
constexpr int BN = 256;
constexpr int BK = 64;
constexpr int THREADS = 512;

using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

__device__ __forceinline__ uint16_t f32_to_bf16_rne(float f) {
  union {
    uint32_t u;
    float f;
  } cvt;
  cvt.f = f;
  uint32_t lsb = (cvt.u >> 16) & 1;
  cvt.u += 0x7FFF + lsb;
  return (uint16_t)(cvt.u >> 16);
}

// st_16x32 swizzle: flip bit5 (32 B) conditioned on bit9 (512 B).
// Involution; row bits untouched.
__device__ __forceinline__ unsigned swz(unsigned byte_off) {
  return byte_off ^ (((byte_off >> 9) & 1u) << 5);
}

// LDS: [buf(2)][op(2: A=0, Bt=1)][half(2)][128 rows][128 B] = 128 KiB
constexpr unsigned kHalfBytes = 128 * 128;
constexpr unsigned kOpBytes = 2 * kHalfBytes;
constexpr unsigned kBufBytes = 2 * kOpBytes;
constexpr unsigned kLdsBytes = 2 * kBufBytes;

__device__ __forceinline__ unsigned lds_off(int buf, int op, int half) {
  return (unsigned)buf * kBufBytes + (unsigned)op * kOpBytes +
         (unsigned)half * kHalfBytes;
}

using lds_void = __attribute__((address_space(3))) void;
using global_void = const __attribute__((address_space(1))) void;

__global__ __launch_bounds__(THREADS, 1) void gemm_bf16_256_kernel(
    const uint16_t* __restrict__ A,   // [M][K]
    const uint16_t* __restrict__ Bt,  // [N][K]
    uint16_t* __restrict__ C,         // [M][N]
    int M, int N, int K, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  int nwg = tiles_m * tiles_n;
  int wgid = blockIdx.x;
  {
    const int nxcd = 8;
    int q = nwg / nxcd, r = nwg % nxcd;
    int xcd = wgid % nxcd, idx = wgid / nxcd;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int row0 = (wgid / tiles_n) * BM;
  const int col0 = (wgid % tiles_n) * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;      // 0..7
  const int wrow = wave >> 2;     // 0..1: 64-row band inside a quadrant
  const int wcol = wave & 3;      // 0..3: 32-col band inside a quadrant
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  // acc[quadrant(mq*2+nq)][mt 0..3][nt 0..1]
  f32x4 acc[4][4][2] = {};

  const uint16_t* op_src[2][2] = {
      {A + (int64_t)row0 * K, A + (int64_t)(row0 + 128) * K},
      {Bt + (int64_t)col0 * K, Bt + (int64_t)(col0 + 128) * K},
  };

  // stage one half-tile (128 rows x 64 k bf16 = 16 KiB): 2 glds per
  // thread; LDS image is lane-linear, swizzle goes on the SOURCE column
  auto stage_half = [&](int op, int half, int ktile) {
    const uint16_t* src = op_src[op][half];
    unsigned base = lds_off(ktile & 1, op, half);
    int k0 = ktile * BK;
#pragma unroll
    for (int q = 0; q < 2; q++) {
      int row = q * 64 + (tid >> 3);
      unsigned image_byte = (unsigned)row * 128 + (unsigned)(tid & 7) * 16;
      int lcol = (int)((swz(image_byte) >> 4) & 7);
      const uint16_t* gsrc = src + (int64_t)row * K + k0 + lcol * 8;
      unsigned dst_off =
          base + (unsigned)(q * 64 * 128) + (unsigned)(tid >> 6) * 1024u;
      __builtin_amdgcn_global_load_lds(
          (global_void*)gsrc, (lds_void*)(smem + dst_off), 16, 0, 0);
    }
  };

  bf16x8 a_frag[4][2];  // 4 m-frags x 2 k-chunks (one quadrant, this wave)
  bf16x8 b_frag[2][2];  // 2 n-frags x 2 k-chunks

  auto load_a = [&](int buf, int mq) {
    unsigned base = lds_off(buf, 0, mq);
#pragma unroll
    for (int t = 0; t < 4; t++)
#pragma unroll
      for (int ch = 0; ch < 2; ch++) {
        unsigned row = (unsigned)(wrow * 64 + t * 16 + l15);
        a_frag[t][ch] = *reinterpret_cast<const bf16x8*>(
            &smem[base + swz(row * 128 + (unsigned)(ch * 4 + l4) * 16)]);
      }
  };
  auto load_b = [&](int buf, int nq) {
    unsigned base = lds_off(buf, 1, nq);
#pragma unroll
    for (int t = 0; t < 2; t++)
#pragma unroll
      for (int ch = 0; ch < 2; ch++) {
        unsigned row = (unsigned)(wcol * 32 + t * 16 + l15);
        b_frag[t][ch] = *reinterpret_cast<const bf16x8*>(
            &smem[base + swz(row * 128 + (unsigned)(ch * 4 + l4) * 16)]);
      }
  };

#define MFMA_QUADRANT(mq, nq)                                              \
  do {                                                                     \
    __builtin_amdgcn_s_setprio(1);                                         \
    _Pragma("unroll") for (int t = 0; t < 4; t++)                          \
        _Pragma("unroll") for (int n = 0; n < 2; n++)                      \
            _Pragma("unroll") for (int ch = 0; ch < 2; ch++) {             \
      acc[(mq)*2 + (nq)][t][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(  \
          a_frag[t][ch], b_frag[n][ch], acc[(mq)*2 + (nq)][t][n], 0, 0, 0); \
    }                                                                      \
    __builtin_amdgcn_s_setprio(0);                                         \
  } while (0)

  const int n_ktiles = K / BK;  // even (K % 128 == 0)

  // prologue: tiles 0 and 1 fully staged, plus tile 2's A0 + B1 (the
  // halves the steady-state pattern would have staged before tile 0)
  stage_half(0, 0, 0);
  stage_half(0, 1, 0);
  stage_half(1, 0, 0);
  stage_half(1, 1, 0);
  stage_half(0, 0, 1);
  stage_half(0, 1, 1);
  stage_half(1, 0, 1);
  stage_half(1, 1, 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  // (no pre-stage of tile 2 here: its slots are tile 0's live buffer;
  // the steady-state pattern stages them at tile 0's ph3/ph4, and the
  // vmcnt(4) at tile 1's ph4 retires them before tile 2 reads)

  for (int kt = 0; kt < n_ktiles; kt++) {
    const int buf = kt & 1;
    const bool s1 = kt + 1 < n_ktiles;   // stage tile kt+1 halves
    const bool s2 = kt + 2 < n_ktiles;   // stage tile kt+2 halves

    // ph1: Q(0,0)
    load_a(buf, 0);
    load_b(buf, 0);
    if (s1) stage_half(1, 0, kt + 1);  // B-half0(kt+1)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    MFMA_QUADRANT(0, 0);

    // ph2: Q(0,1)
    load_b(buf, 1);
    if (s1) stage_half(0, 1, kt + 1);  // A-half1(kt+1)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    MFMA_QUADRANT(0, 1);

    // ph3: Q(1,1) -- barrier BEFORE the stage: all waves must be past
    // their ph1/ph2 reads of the slots ph3/ph4 overwrite
    load_a(buf, 1);
    __builtin_amdgcn_s_barrier();
    if (s2) stage_half(0, 0, kt + 2);  // A-half0(kt+2)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    MFMA_QUADRANT(1, 1);

    // ph4: Q(1,0) -- counted vmcnt + publication barrier for next tile
    load_b(buf, 0);
    if (s2) stage_half(1, 1, kt + 2);  // B-half1(kt+2)
    if (s1) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    MFMA_QUADRANT(1, 0);
  }

  // epilogue: acc -> bf16 C stores
#pragma unroll
  for (int mq = 0; mq < 2; mq++) {
#pragma unroll
    for (int nq = 0; nq < 2; nq++) {
      const int r0 = row0 + mq * 128 + wrow * 64 + 4 * l4;
      const int c0 = col0 + nq * 128 + wcol * 32 + l15;
#pragma unroll
      for (int mt = 0; mt < 4; mt++)
#pragma unroll
        for (int nt = 0; nt < 2; nt++)
#pragma unroll
          for (int r = 0; r < 4; r++) {
            C[(int64_t)(r0 + mt * 16 + r) * N + c0 + nt * 16] =
                f32_to_bf16_rne(acc[mq * 2 + nq][mt][nt][r]);
          }
    }
  }
}

// ---------------------------------------------------------------------------
// Variant B: per-wave-owned output tiles (the CDNA4 guide's verified
// 256^2 8-phase template shape): 8 waves as 2M x 4N, each wave owns a
// 128x64 C tile (acc[8][4]); per phase a wave computes 2 m-fragments x
// 4 n-fragments x 2 K-chunks = 16 MFMAs. B fragments are read once per
// K-tile and held in registers; A fragments 4 ds_read_b128 per phase.
// Hazard-minimal synchronization (2 barriers per K-tile, derivation in
// the phase loop comments); the two waves sharing a SIMD drift into
// antiphase so one issues MFMAs while the other is in its load/stage
// region. Staging schedule (1 half per phase, exact-counted vmcnt):
//   X.ph0: stage A0(X+1)   (slot last read at X-1.ph3, >= 1 barrier ago)
//   X.ph1: stage A1(X+1)
//   X.ph2: stage B0(X+2)   (B(X) regs were loaded+waited at X.ph0)
//   X.ph3: stage B1(X+2); s_waitcnt vmcnt(4 if B(X+2) staged else 0)
//          -> retires A(X+1) and everything older before X+1.ph0 reads;
//          ph3's post-MFMA barrier publishes.
__global__ __launch_bounds__(THREADS, 1) void gemm_bf16_256b_kernel(
    const uint16_t* __restrict__ A,   // [M][K]
    const uint16_t* __restrict__ Bt,  // [N][K]
    uint16_t* __restrict__ C,         // [M][N]
    int M, int N, int K, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  int nwg = tiles_m * tiles_n;
  int wgid = blockIdx.x;
  int tm, tn;
  const int n_st = (tiles_m / 8) * (tiles_n / 4);
  if (tiles_m % 8 == 0 && tiles_n % 4 == 0 && n_st % 8 == 0) {
    // 2D supertiling: each XCD (blockIdx%8 under round-robin dispatch)
    // works whole 8x4-tile supertiles, so its L2 reuses each A row-band
    // 4x and each B column 8x (1D banding leaves the B operand's HBM
    // traffic as the 8k+ roof). Requires n_st % 8 == 0 so every XCD
    // gets whole supertiles (the bijective 1D remap covers the rest).
    const int st_cols = tiles_n / 4;
    int xcd = wgid % 8, idx = wgid / 8;
    int st = xcd + 8 * (idx >> 5);     // 32 CUs per XCD, 1 WG each
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
  const int row0 = tm * BM;
  const int col0 = tn * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;   // 0..7
  const int wm = wave >> 2;    // 0..1: which 128-row A half this wave owns
  const int wn = wave & 3;     // 0..3: 64-col band of B
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  f32x4 acc[8][4] = {};        // [m-frag][n-frag]

  const uint16_t* op_src[2][2] = {
      {A + (int64_t)row0 * K, A + (int64_t)(row0 + 128) * K},
      {Bt + (int64_t)col0 * K, Bt + (int64_t)(col0 + 128) * K},
  };

  auto stage_half = [&](int op, int half, int ktile) {
    const uint16_t* src = op_src[op][half];
    unsigned base = lds_off(ktile & 1, op, half);
    int k0 = ktile * BK;
#pragma unroll
    for (int q = 0; q < 2; q++) {
      int row = q * 64 + (tid >> 3);
      unsigned image_byte = (unsigned)row * 128 + (unsigned)(tid & 7) * 16;
      int lcol = (int)((swz(image_byte) >> 4) & 7);
      const uint16_t* gsrc = src + (int64_t)row * K + k0 + lcol * 8;
      unsigned dst_off =
          base + (unsigned)(q * 64 * 128) + (unsigned)(tid >> 6) * 1024u;
      __builtin_amdgcn_global_load_lds(
          (global_void*)gsrc, (lds_void*)(smem + dst_off), 16, 0, 0);
    }
  };

  bf16x8 a_frag[2][2];  // 2 m-frags (this phase) x 2 k-chunks
  bf16x8 b_frag[4][2];  // 4 n-frags (whole tile)  x 2 k-chunks

  // A rows for phase p: within-half rows [32p, 32p+32) of half wm
  auto load_a_phase = [&](int buf, int p) {
    unsigned base = lds_off(buf, 0, wm);
#pragma unroll
    for (int t = 0; t < 2; t++)
#pragma unroll
      for (int ch = 0; ch < 2; ch++) {
        unsigned row = (unsigned)(p * 32 + t * 16 + l15);
        a_frag[t][ch] = *reinterpret_cast<const bf16x8*>(
            &smem[base + swz(row * 128 + (unsigned)(ch * 4 + l4) * 16)]);
      }
  };
  // B rows for the whole tile: [wn*64, wn*64+64) across the two halves
  auto load_b_tile = [&](int buf) {
#pragma unroll
    for (int n = 0; n < 4; n++)
#pragma unroll
      for (int ch = 0; ch < 2; ch++) {
        unsigned ncol = (unsigned)(wn * 64 + n * 16 + l15);
        unsigned base = lds_off(buf, 1, (int)(ncol >> 7));
        unsigned row = ncol & 127u;
        b_frag[n][ch] = *reinterpret_cast<const bf16x8*>(
            &smem[base + swz(row * 128 + (unsigned)(ch * 4 + l4) * 16)]);
      }
  };

  const int n_ktiles = K / BK;  // even (K % 128 == 0)

  // prologue: tile 0 fully + B halves of tile 1; retire tile 0 before ph0
  stage_half(0, 0, 0);
  stage_half(0, 1, 0);
  stage_half(1, 0, 0);
  stage_half(1, 1, 0);
  if (n_ktiles > 1) {
    stage_half(1, 0, 1);
    stage_half(1, 1, 1);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < n_ktiles; kt++) {
    const int buf = kt & 1;
    const bool s1 = kt + 1 < n_ktiles;  // stage A halves of kt+1
    const bool s2 = kt + 2 < n_ktiles;  // stage B halves of kt+2

#pragma unroll
    for (int p = 0; p < 4; p++) {
      load_a_phase(buf, p);
      if (p == 0) load_b_tile(buf);
      // one staged half per phase (even vmem pressure):
      // A halves of kt+1 at ph0/ph1, B halves of kt+2 at ph2/ph3
      if (p == 0 && s1) {
        stage_half(0, 0, kt + 1);
      } else if (p == 1 && s1) {
        stage_half(0, 1, kt + 1);
      } else if (p == 2 && s2) {
        stage_half(1, 0, kt + 2);
      } else if (p == 3) {
        if (s2) {
          stage_half(1, 1, kt + 2);
          // retire A(kt+1) and older; keep only B(kt+2) outstanding
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        } else if (s1) {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
      }
      // NO barrier before the MFMA cluster: every stage above writes a
      // slot whose last reader finished before that reader's post-MFMA
      // barrier (>= 1 barrier ago), and staged data is published by the
      // per-wave vmcnt + the post-MFMA barrier of its deadline phase.
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ch = 0; ch < 2; ch++)
#pragma unroll
        for (int t = 0; t < 2; t++)
#pragma unroll
          for (int n = 0; n < 4; n++) {
            acc[p * 2 + t][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag[t][ch], b_frag[n][ch], acc[p * 2 + t][n], 0, 0, 0);
          }
      __builtin_amdgcn_s_setprio(0);
      // hazard-minimal barriers, 2 per K-tile: ph1-post separates this
      // tile's ph0 B-reads from ph2/ph3's B(kt+2) stages; ph3-post both
      // publishes the vmcnt-retired halves of kt+1 and separates ph0's
      // A(kt+2) stages (next tile) from this tile's A reads.
      if (p & 1) __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: per-wave 128x64 tile -> C (MFMA D rows = 4*(lane>>4)+reg)
#pragma unroll
  for (int mt = 0; mt < 8; mt++) {
    const int r0 = row0 + wm * 128 + mt * 16 + 4 * l4;
    const int c0 = col0 + wn * 64 + l15;
#pragma unroll
    for (int nt = 0; nt < 4; nt++)
#pragma unroll
      for (int r = 0; r < 4; r++) {
        C[(int64_t)(r0 + r) * N + c0 + nt * 16] =
            f32_to_bf16_rne(acc[mt][nt][r]);
      }
  }
}

// ---------------------------------------------------------------------------
// bf16 transpose: out[N][K] = in[K][N]^T. 64x64 LDS tiles (pad 72 u16).
// ---------------------------------------------------------------------------
using ushort8 = __attribute__((ext_vector_type(8))) unsigned short;

__global__ __launch_bounds__(256) void transpose_bf16_kernel(
    const uint16_t* __restrict__ in, uint16_t* __restrict__ out, int K,
    int N) {
  __shared__ uint16_t tile[64][72];
  int k0 = blockIdx.x * 64;
  int n0 = blockIdx.y * 64;
  int tid = threadIdx.x;
  int lr = tid >> 3;
  int lc = (tid & 7) * 8;
  const bool interior =
      (k0 + 63 < K) && (n0 + 63 < N);  // vector fast path (16 B I/O)
#pragma unroll
  for (int q = 0; q < 2; q++) {
    int k = k0 + lr + q * 32;
    if (interior) {
      ushort8 v = *reinterpret_cast<const ushort8*>(&in[(int64_t)k * N + n0 + lc]);
#pragma unroll
      for (int j = 0; j < 8; j++) tile[lc + j][lr + q * 32] = v[j];
    } else {
#pragma unroll
      for (int j = 0; j < 8; j++) {
        int n = n0 + lc + j;
        tile[lc + j][lr + q * 32] =
            (k < K && n < N) ? in[(int64_t)k * N + n] : 0;
      }
    }
  }
  __syncthreads();
#pragma unroll
  for (int q = 0; q < 2; q++) {
    int n = n0 + lr + q * 32;
    if (interior) {
      ushort8 v;
#pragma unroll
      for (int j = 0; j < 8; j++) v[j] = tile[lr + q * 32][lc + j];
      *reinterpret_cast<ushort8*>(&out[(int64_t)n * K + k0 + lc]) = v;
    } else {
      if (n >= N) continue;
      uint16_t* dst = out + (int64_t)n * K + k0 + lc;
#pragma unroll
      for (int j = 0; j < 8; j++)
        if (k0 + lc + j < K) dst[j] = tile[lr + q * 32][lc + j];
    }
  }
}

}  // namespace

void launch_transpose_bf16(const uint16_t* in, uint16_t* out, int k, int n,
                           hipStream_t stream) {
  dim3 grid((k + 63) / 64, (n + 63) / 64);
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, stream, in,
                     out, k, n);
  HIP_CHECK(hipGetLastError());
}

bool gemm_bf16_256_supported(int m, int n, int k) {
  return m % 256 == 0 && n % 256 == 0 && k % 128 == 0;
}

void launch_gemm_bf16_256(const uint16_t* a, const uint16_t* bt, uint16_t* c,
                          int m, int n, int k, hipStream_t stream) {
  // variant B (default): per-wave-owned output tiles, hazard-minimal
  // barriers, 2D XCD supertiling -- measured faster AND race-free.
  // variant A (cooperative quadrants) kept for A/B only: the multi-run
  // screen showed an intermittent race in it (see profiles/NOTES.md).
  static const char* v = getenv("APP_BF16_256_VARIANT");
  static const bool use_b = (v == nullptr || v[0] != 'a');
  int tiles_m = m / BM;
  int tiles_n = n / BN;
  if (use_b)
    hipLaunchKernelGGL(gemm_bf16_256b_kernel, dim3(tiles_m * tiles_n),
                       dim3(THREADS), kLdsBytes, stream, a, bt, c, m, n, k,
                       tiles_m, tiles_n);
  else
    hipLaunchKernelGGL(gemm_bf16_256_kernel, dim3(tiles_m * tiles_n),
                       dim3(THREADS), kLdsBytes, stream, a, bt, c, m, n, k,
                       tiles_m, tiles_n);
  HIP_CHECK(hipGetLastError());
}
