// bf16 GEMM, 256^2-tile 8-phase structure for gfx950 (variant B).
//
// 256x256 block tile, BK=64, 512 threads as 8 waves (2M x 4N), one
// workgroup per CU (128 KiB LDS). Each wave owns a 128x64 C tile
// (acc[8][4] fragments); per phase it computes 2 m-frags x 4 n-frags x
// 2 K-chunks = 16 MFMAs (mfma_f32_16x16x32_bf16). B fragments are read
// once per K-tile and held in registers; A fragments are 4 ds_read_b128
// per phase. K-tiles are double-buffered and filled by
// global_load_lds_dwordx4 (16 B/lane LDS-DMA) with the st_16x32 XOR
// swizzle on both the source position and the ds_read offset.
// Synchronization is hazard-minimal (2 raw s_barriers per K-tile +
// exact-counted per-wave s_waitcnt vmcnt) -- the derivation is inline in
// the phase loop; with no lockstep barrier around the MFMA cluster the
// two waves sharing a SIMD drift into antiphase, keeping the MFMA pipe
// fed while the co-wave is in its load/stage region. Workgroups map to
// XCDs via 2D 8x4-tile supertiles so each XCD's L2 reuses A row-bands
// 4x and B columns 8x.
//
// History note (profiles/NOTES.md): the previous cooperative-quadrant
// variant measured ~995 TF but a multi-run screen caught an intermittent
// slot-reuse race; this per-wave-tile structure is both faster
// (~975 TF @4k / ~1035 TF @8k end-to-end incl. the B pre-transpose) and
// race-free under the 3x multi-size screen (scripts/bf16_ab.py).
//
// Both operands are consumed K-contiguous: A is row-major [M][K]; B is
// pre-transposed to Bt[N][K] by transpose_bf16 (driven by the launcher).
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

// ---------------------------------------------------------------------------
// Variant T: same 8-phase structure, but the B operand is consumed in its
// NATIVE [K][N] layout -- no pre-transpose kernel, no Bt scratch, and
// ~134 MB less HBM traffic per 8192^3 GEMM (the transpose pass was the
// measured ~2-7% wall gap to hipBLASLt, profiles/NOTES.md r01).
//
// Mechanism: gfx950's ds_read_b64_tr_b16 hardware transpose-read
// (semantics established empirically by scripts/tr16_probe.cpp: within a
// 16-lane group, each lane fetches its own 8-B slice at
// base + (lane&15)*8, and the engine transposes lane-quads so lane l's
// element j = u16[(l&15) + 16*j] of the group's 128-B span -- i.e.
// column (l&15) of a row-major [4k][16n] bf16 block).
//
// LDS image per 128-col B half (16 KiB): [nb(8)][pos(16)] blocks of
// [4k][16n] u16, where pos permutes the k-blocks as
//   pos = ch*8 + r*4 + gq   <->   kb = ch*8 + gq*2 + r
// so one tr16 read (4 16-lane groups = 4 consecutive pos blocks) hands
// group g exactly its MFMA k-chunk (fragment k = 32ch + 8g + 4r + j),
// with bases 128 B apart: banks (a/4)%64 are disjoint within each
// 32-lane conflict half -- conflict-free. The global granule (fixed k,
// 8 consecutive n = 16 B) stays contiguous in the image, so the fill is
// the same lane-linear global_load_lds as variant B, just with the
// (k, n) source coordinates derived from the permuted destination.
// A-operand path, phase schedule, vmcnt counting and barriers are
// identical to variant B.
namespace {

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
using lds_bf16x4 = __attribute__((address_space(3))) bf16x4;

__global__ __launch_bounds__(THREADS, 1) void gemm_bf16_256t_kernel(
    const uint16_t* __restrict__ A,  // [M][K]
    const uint16_t* __restrict__ B,  // [K][N] (native row-major)
    uint16_t* __restrict__ C,        // [M][N]
    int M, int N, int K, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];

  int nwg = tiles_m * tiles_n;
  int wgid = blockIdx.x;
  int tm, tn;
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
  const int row0 = tm * BM;
  const int col0 = tn * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  f32x4 acc[8][4] = {};

  const uint16_t* a_src[2] = {A + (int64_t)row0 * K,
                              A + (int64_t)(row0 + 128) * K};

  // A halves: identical to variant B (swizzled glds image)
  auto stage_a_half = [&](int half, int ktile) {
    const uint16_t* src = a_src[half];
    unsigned base = lds_off(ktile & 1, 0, half);
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

  // B halves: native [K][N] source, permuted-block tr16 image (above)
  auto stage_b_half = [&](int half, int ktile) {
    const uint16_t* src = B + col0 + half * 128;
    unsigned base = lds_off(ktile & 1, 1, half);
    int k0 = ktile * BK;
#pragma unroll
    for (int q = 0; q < 2; q++) {
      unsigned p = (unsigned)q * 8192u + (unsigned)tid * 16u;  // dst byte
      unsigned blk = p >> 7;
      unsigned within = p & 127u;
      unsigned nb = blk >> 4, pos = blk & 15u;
      unsigned ch = pos >> 3, r = (pos >> 2) & 1u, gq = pos & 3u;
      unsigned kb = ch * 8u + gq * 2u + r;
      unsigned k = kb * 4u + (within >> 5);
      unsigned n0 = nb * 16u + ((within >> 4) & 1u) * 8u;
      const uint16_t* gsrc = src + (int64_t)(k0 + (int)k) * N + n0;
      unsigned dst_wave =
          base + (unsigned)(q * 64 * 128) + (unsigned)(tid >> 6) * 1024u;
      __builtin_amdgcn_global_load_lds(
          (global_void*)gsrc, (lds_void*)(smem + dst_wave), 16, 0, 0);
    }
  };

  auto stage_half = [&](int op, int half, int ktile) {
    if (op == 0)
      stage_a_half(half, ktile);
    else
      stage_b_half(half, ktile);
  };

  bf16x8 a_frag[2][2];
  bf16x8 b_frag[4][2];

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

  // one tr16 read per (fragment, ch, r): group g = lane>>4 reads pos
  // block (nb*16 + ch*8 + r*4 + g) at its own (lane&15)*8 slice
  const unsigned tr_lane_off =
      (unsigned)(lane & 15) * 8u + (unsigned)(lane >> 4) * 128u;
  auto load_b_tile = [&](int buf) {
#pragma unroll
    for (int n = 0; n < 4; n++) {
      unsigned ncol = (unsigned)(wn * 64 + n * 16);
      unsigned hbase = lds_off(buf, 1, (int)(ncol >> 7));
      unsigned nb = (ncol >> 4) & 7u;
#pragma unroll
      for (int ch = 0; ch < 2; ch++) {
        union {
          bf16x8 v8;
          bf16x4 v4[2];
        } u;
#pragma unroll
        for (int r = 0; r < 2; r++) {
          unsigned blk_byte =
              (nb * 16u + (unsigned)ch * 8u + (unsigned)r * 4u) * 128u;
          u.v4[r] = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (lds_bf16x4*)(smem + hbase + blk_byte + tr_lane_off));
        }
        b_frag[n][ch] = u.v8;
      }
    }
  };

  const int n_ktiles = K / BK;

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
    const bool s1 = kt + 1 < n_ktiles;
    const bool s2 = kt + 2 < n_ktiles;

#pragma unroll
    for (int p = 0; p < 4; p++) {
      load_a_phase(buf, p);
      if (p == 0) load_b_tile(buf);
      if (p == 0 && s1) {
        stage_half(0, 0, kt + 1);
      } else if (p == 1 && s1) {
        stage_half(0, 1, kt + 1);
      } else if (p == 2 && s2) {
        stage_half(1, 0, kt + 2);
      } else if (p == 3) {
        if (s2) {
          stage_half(1, 1, kt + 2);
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        } else if (s1) {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
      }
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
      if (p & 1) __builtin_amdgcn_s_barrier();
    }
  }

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

}  // namespace

void launch_gemm_bf16_256t(const uint16_t* a, const uint16_t* b, uint16_t* c,
                           int m, int n, int k, hipStream_t stream) {
  int tiles_m = m / BM;
  int tiles_n = n / BN;
  hipLaunchKernelGGL(gemm_bf16_256t_kernel, dim3(tiles_m * tiles_n),
                     dim3(THREADS), kLdsBytes, stream, a, b, c, m, n, k,
                     tiles_m, tiles_n);
  HIP_CHECK(hipGetLastError());
}

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
  int tiles_m = m / BM;
  int tiles_n = n / BN;
  hipLaunchKernelGGL(gemm_bf16_256b_kernel, dim3(tiles_m * tiles_n),
                     dim3(THREADS), kLdsBytes, stream, a, bt, c, m, n, k,
                     tiles_m, tiles_n);
  HIP_CHECK(hipGetLastError());
}
