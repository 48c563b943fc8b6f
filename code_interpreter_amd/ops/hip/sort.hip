// Device radix sort (np.sort / np.argsort) for f32/f64, written for
// wave64: LSD over 8-bit digits, stable, with IEEE-754 key transforms
// (monotone float -> unsigned map; NaNs map above +inf, matching
// numpy's NaNs-last ordering).
//
// Structure per digit pass:
//   K1  one wave per chunk builds a 256-bin digit histogram (LDS
//       atomics, lanes stride the chunk), written to counts[chunk][256]
//   K2  one 256-thread block turns counts into global stable scatter
//       bases: digit-major prefix across chunks + digit base offsets
//   K3  one wave per chunk scatters its chunk: per-64-element tile,
//       equal-digit lane groups are found with an 8-step ballot
//       bit-split, each lane's stable rank is popcount(same & below),
//       and a per-wave LDS running[256] carries the chunk's offsets
// Keys ping-pong between two buffers; an optional int64 payload
// (argsort indices) rides along through the same scatters.

#include <cstring>

#include "common.h"

namespace {

constexpr int kSortChunk = 4096;   // elements per wave
constexpr int kWavesPerBlock = 4;  // 256 threads

__device__ __forceinline__ uint64_t key64_of_f64(double v) {
  if (v != v) return ~0ull;  // all NaNs (either sign) last, like numpy
  uint64_t u = __double_as_longlong(v);
  return (u & 0x8000000000000000ull) ? ~u : (u | 0x8000000000000000ull);
}
__device__ __forceinline__ double f64_of_key64(uint64_t k) {
  uint64_t u = (k & 0x8000000000000000ull) ? (k & 0x7fffffffffffffffull)
                                           : ~k;
  return __longlong_as_double((long long)u);
}
__device__ __forceinline__ uint32_t key32_of_f32(float v) {
  if (v != v) return ~0u;  // all NaNs (either sign) last, like numpy
  uint32_t u = __float_as_uint(v);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
__device__ __forceinline__ float f32_of_key32(uint32_t k) {
  uint32_t u = (k & 0x80000000u) ? (k & 0x7fffffffu) : ~k;
  return __uint_as_float(u);
}

// ROWDIG: digit taken from the element's row id (idx[i] / L) instead of
// the key -- the row passes of the 2-D (axis=-1) sort. LSD stability
// makes "key passes, then row passes" == an independent sort per row.
template <typename K, bool ROWDIG>
__global__ void radix_hist_kernel(const K* __restrict__ keys,
                                  const long long* __restrict__ idx,
                                  unsigned int L, int64_t n, int shift,
                                  unsigned int* __restrict__ counts,
                                  int64_t nchunks) {
  __shared__ unsigned int hist[kWavesPerBlock][256];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  int64_t chunk = (int64_t)blockIdx.x * kWavesPerBlock + wave;
  for (int i = lane; i < 256; i += 64) hist[wave][i] = 0;
  // no cross-wave sharing: wave-local LDS, no barrier needed beyond
  // the implicit wave lockstep
  if (chunk < nchunks) {
    int64_t start = chunk * kSortChunk;
    int64_t end = start + kSortChunk < n ? start + kSortChunk : n;
    for (int64_t i = start + lane; i < end; i += 64) {
      unsigned d = ROWDIG ? (((unsigned)idx[i] / L) >> shift) & 0xff
                          : (unsigned)((keys[i] >> shift) & 0xff);
      atomicAdd(&hist[wave][d], 1u);
    }
    unsigned int* out = counts + chunk * 256;
    for (int i = lane; i < 256; i += 64) out[i] = hist[wave][i];
  }
}

// counts[chunk][256] -> stable scatter bases, parallel three-phase scan
// (a single-block sequential scan over 24k chunks would serialize on
// one CU and dominate the sort):
//   P1: thread per (digit, segment of 256 chunks): segment sums
//   P2: one block: per-digit scan over segments + digit-base scan
//   P3: thread per (digit, segment): fold prefix through the segment
constexpr int kSeg = 256;  // chunks per segment

__global__ void radix_segsum_kernel(const unsigned int* __restrict__ counts,
                                    int64_t nchunks, int64_t nseg,
                                    unsigned int* __restrict__ segsum) {
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= nseg * 256) return;
  int64_t seg = t >> 8;
  int d = (int)(t & 0xff);
  int64_t c0 = seg * kSeg;
  int64_t c1 = c0 + kSeg < nchunks ? c0 + kSeg : nchunks;
  unsigned int acc = 0;
  for (int64_t c = c0; c < c1; c++) acc += counts[c * 256 + d];
  segsum[seg * 256 + d] = acc;
}

__global__ void radix_segscan_kernel(unsigned int* __restrict__ segsum,
                                     int64_t nseg,
                                     unsigned long long* __restrict__ dig) {
  const int d = threadIdx.x;  // 256 threads
  unsigned long long run = 0;
  for (int64_t sg = 0; sg < nseg; sg++) {
    unsigned int v = segsum[sg * 256 + d];
    segsum[sg * 256 + d] = (unsigned int)run;
    run += v;
  }
  dig[d] = run;
  __syncthreads();
  if (d == 0) {
    unsigned long long acc = 0;
    for (int i = 0; i < 256; i++) {
      unsigned long long v = dig[i];
      dig[i] = acc;
      acc += v;
    }
  }
  __syncthreads();
  unsigned long long base = dig[d];
  for (int64_t sg = 0; sg < nseg; sg++)
    segsum[sg * 256 + d] += (unsigned int)base;
}

__global__ void radix_fold_kernel(unsigned int* __restrict__ counts,
                                  int64_t nchunks, int64_t nseg,
                                  const unsigned int* __restrict__ segsum) {
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= nseg * 256) return;
  int64_t seg = t >> 8;
  int d = (int)(t & 0xff);
  int64_t c0 = seg * kSeg;
  int64_t c1 = c0 + kSeg < nchunks ? c0 + kSeg : nchunks;
  unsigned int run = segsum[seg * 256 + d];
  for (int64_t c = c0; c < c1; c++) {
    unsigned int v = counts[c * 256 + d];
    counts[c * 256 + d] = run;
    run += v;
  }
}

template <typename K, bool PAYLOAD>
__global__ void radix_scatter_kernel(const K* __restrict__ in_keys,
                                     K* __restrict__ out_keys,
                                     const long long* __restrict__ in_idx,
                                     long long* __restrict__ out_idx,
                                     int64_t n, int shift,
                                     const unsigned int* __restrict__ bases,
                                     int64_t nchunks) {
  __shared__ unsigned int running[kWavesPerBlock][256];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  int64_t chunk = (int64_t)blockIdx.x * kWavesPerBlock + wave;
  if (chunk >= nchunks) return;
  for (int i = lane; i < 256; i += 64)
    running[wave][i] = bases[chunk * 256 + i];
  int64_t start = chunk * kSortChunk;
  int64_t end = start + kSortChunk < n ? start + kSortChunk : n;
  const uint64_t lanes_below = ((uint64_t)1 << lane) - 1;
  for (int64_t i0 = start; i0 < end; i0 += 64) {
    int64_t i = i0 + lane;
    bool valid = i < end;
    K key = valid ? in_keys[i] : (K)0;
    unsigned d = (unsigned)((key >> shift) & 0xff);
    // 8-step bit-split: lanes with MY digit
    uint64_t same = __ballot(valid);
#pragma unroll
    for (int b = 0; b < 8; b++) {
      uint64_t bal = __ballot(valid && ((d >> b) & 1u));
      same &= ((d >> b) & 1u) ? bal : ~bal;
    }
    if (valid) {
      unsigned rank = (unsigned)__popcll(same & lanes_below);
      unsigned pos = running[wave][d] + rank;
      out_keys[pos] = key;
      if (PAYLOAD) out_idx[pos] = in_idx[i];
      // group leader advances the running offset for this digit
      if (rank == 0) running[wave][d] += (unsigned)__popcll(same);
    }
  }
}

// Block-cooperative scatter (default variant): one 4096-element chunk
// per 256-thread block. The wave-autonomous scatter above issues one
// 4/8-B global write per element to a near-random line — ~16x write
// amplification, measured 64% of sort time (profiles/
// sort_kernel_stats_r02.md). Here the chunk is reordered digit-
// contiguously through LDS first, so global writes go out in ~16-
// element runs (64-128 B), and the argsort payload is gathered through
// a u16 in-chunk origin (the 32 KiB in_idx window stays cache-
// resident) instead of scattered 8-B stores.
template <typename K, bool PAYLOAD, bool ROWDIG = false>
__global__ void radix_scatter_block_kernel(
    const K* __restrict__ in_keys, K* __restrict__ out_keys,
    const long long* __restrict__ in_idx, long long* __restrict__ out_idx,
    int64_t n, int shift, const unsigned int* __restrict__ bases,
    int64_t nchunks, unsigned int L = 0) {
  static_assert(!ROWDIG || PAYLOAD, "row digits come from the payload");
  constexpr int kQuarter = kSortChunk / 4;  // elements per wave
  constexpr int kTiles = kQuarter / 64;     // 64-lane tiles per wave
  __shared__ K stage[kSortChunk];
  __shared__ unsigned short origin[PAYLOAD ? kSortChunk : 64];
  __shared__ unsigned int wcnt[4][256];  // per-wave digit counts -> wave-
                                         // exclusive prefixes (phase 2)
  __shared__ unsigned int pref[256];     // chunk-local digit prefix
  __shared__ unsigned int scan_tmp[256];

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t chunk = blockIdx.x;
  if (chunk >= nchunks) return;
  const int64_t start = chunk * kSortChunk;
  const int64_t end = start + kSortChunk < n ? start + kSortChunk : n;
  const int len = (int)(end - start);
  const uint64_t lanes_below = ((uint64_t)1 << lane) - 1;

  for (int i = lane; i < 256; i += 64) wcnt[wave][i] = 0;
  // phase 1: wave-local stable ranks over this wave's quarter
  K key[kTiles];
  unsigned short rank[kTiles];
  const int q0 = wave * kQuarter;
  for (int t = 0; t < kTiles; t++) {
    int e = q0 + t * 64 + lane;
    bool valid = e < len;
    key[t] = valid ? in_keys[start + e] : (K)0;
    unsigned d;
    if (ROWDIG)  // valid-guarded: in_idx[start+e] is OOB past len
      d = valid ? (((unsigned)in_idx[start + e] / L) >> shift) & 0xff : 0u;
    else
      d = (unsigned)((key[t] >> shift) & 0xff);
    uint64_t same = __ballot(valid);
#pragma unroll
    for (int b = 0; b < 8; b++) {
      uint64_t bal = __ballot(valid && ((d >> b) & 1u));
      same &= ((d >> b) & 1u) ? bal : ~bal;
    }
    if (valid) {
      unsigned r = (unsigned)__popcll(same & lanes_below);
      rank[t] = (unsigned short)(wcnt[wave][d] + r);
      if (r == 0) wcnt[wave][d] += (unsigned)__popcll(same);
    }
  }
  __syncthreads();
  // phase 2 (one thread per digit): wave-exclusive prefixes in wcnt,
  // chunk totals -> exclusive digit prefix in pref (Hillis-Steele)
  {
    const int d = threadIdx.x;
    unsigned c0 = wcnt[0][d], c1 = wcnt[1][d], c2 = wcnt[2][d];
    unsigned tot = c0 + c1 + c2 + wcnt[3][d];
    wcnt[0][d] = 0;
    wcnt[1][d] = c0;
    wcnt[2][d] = c0 + c1;
    wcnt[3][d] = c0 + c1 + c2;
    unsigned v = tot;
    scan_tmp[d] = v;
    for (int off = 1; off < 256; off <<= 1) {
      __syncthreads();
      unsigned add = d >= off ? scan_tmp[d - off] : 0;
      __syncthreads();
      scan_tmp[d] = v = v + add;
    }
    pref[d] = v - tot;  // inclusive -> exclusive
  }
  __syncthreads();
  // phase 3: scatter into LDS, digit-contiguous
  for (int t = 0; t < kTiles; t++) {
    int e = q0 + t * 64 + lane;
    if (e < len) {
      unsigned d = ROWDIG
          ? (((unsigned)in_idx[start + e] / L) >> shift) & 0xff
          : (unsigned)((key[t] >> shift) & 0xff);
      unsigned pos = pref[d] + wcnt[wave][d] + rank[t];
      stage[pos] = key[t];
      if (PAYLOAD) origin[pos] = (unsigned short)e;
    }
  }
  __syncthreads();
  // phase 4: linear readout -> coalesced digit-run global writes
  const unsigned int* base_row = bases + chunk * 256;
  for (int t = 0; t < kTiles; t++) {
    int li = (int)threadIdx.x + t * 256;
    if (li < len) {
      K k = stage[li];
      long long iv = PAYLOAD ? in_idx[start + origin[li]] : 0;
      unsigned d = ROWDIG ? (((unsigned)iv / L) >> shift) & 0xff
                          : (unsigned)((k >> shift) & 0xff);
      unsigned gpos = base_row[d] + (unsigned)li - pref[d];
      out_keys[gpos] = k;
      if (PAYLOAD) out_idx[gpos] = iv;
    }
  }
}

template <typename T, typename K>
__global__ void sort_encode_kernel(const T* __restrict__ in,
                                   K* __restrict__ keys,
                                   long long* __restrict__ idx, int64_t n,
                                   int with_idx) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if constexpr (sizeof(T) == 8)
      keys[i] = key64_of_f64((double)in[i]);
    else
      keys[i] = key32_of_f32((float)in[i]);
    if (with_idx) idx[i] = (long long)i;
  }
}

template <typename T, typename K>
__global__ void sort_decode_kernel(const K* __restrict__ keys,
                                   T* __restrict__ out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if constexpr (sizeof(T) == 8)
      out[i] = (T)f64_of_key64(keys[i]);
    else
      out[i] = (T)f32_of_key32(keys[i]);
  }
}

// APP_SORT_VARIANT=wave selects the wave-autonomous scatter for A/B
// measurement; default is the LDS block scatter
static bool use_block_scatter() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("APP_SORT_VARIANT");
    v = (e && strcmp(e, "wave") == 0) ? 0 : 1;
  }
  return v == 1;
}

template <typename T, typename K>
static void radix_sort_impl(const T* in, T* out, long long* idx_out,
                            void* keys_a, void* keys_b, void* idx_a,
                            void* idx_b, void* counts, void* dig_scratch,
                            int64_t n, bool want_idx, hipStream_t s) {
  int64_t nchunks = (n + kSortChunk - 1) / kSortChunk;
  int grid_c = (int)((nchunks + kWavesPerBlock - 1) / kWavesPerBlock);
  int grid_e = (int)std::min<int64_t>((n + 255) / 256, 4096);

  K* ka = (K*)keys_a;
  K* kb = (K*)keys_b;
  long long* ia = (long long*)idx_a;
  long long* ib = (long long*)idx_b;

  hipLaunchKernelGGL((sort_encode_kernel<T, K>), dim3(grid_e), dim3(256), 0,
                     s, in, ka, ia, n, want_idx ? 1 : 0);
  const int passes = (int)sizeof(K);
  int64_t nseg = (nchunks + kSeg - 1) / kSeg;
  unsigned int* segsum =
      (unsigned int*)((char*)dig_scratch + 256 * 8);  // after dig[256]
  int grid_s = (int)((nseg * 256 + 255) / 256);
  for (int p = 0; p < passes; p++) {
    int shift = p * 8;
    hipLaunchKernelGGL((radix_hist_kernel<K, false>), dim3(grid_c), dim3(256),
                       0, s, ka, nullptr, 0u, n, shift,
                       (unsigned int*)counts, nchunks);
    hipLaunchKernelGGL(radix_segsum_kernel, dim3(grid_s), dim3(256), 0, s,
                       (const unsigned int*)counts, nchunks, nseg, segsum);
    hipLaunchKernelGGL(radix_segscan_kernel, dim3(1), dim3(256), 0, s,
                       segsum, nseg, (unsigned long long*)dig_scratch);
    hipLaunchKernelGGL(radix_fold_kernel, dim3(grid_s), dim3(256), 0, s,
                       (unsigned int*)counts, nchunks, nseg, segsum);
    if (use_block_scatter()) {
      if (want_idx)
        hipLaunchKernelGGL((radix_scatter_block_kernel<K, true>),
                           dim3((unsigned)nchunks), dim3(256), 0, s, ka, kb,
                           ia, ib, n, shift, (const unsigned int*)counts,
                           nchunks);
      else
        hipLaunchKernelGGL((radix_scatter_block_kernel<K, false>),
                           dim3((unsigned)nchunks), dim3(256), 0, s, ka, kb,
                           nullptr, nullptr, n, shift,
                           (const unsigned int*)counts, nchunks);
    } else if (want_idx)
      hipLaunchKernelGGL((radix_scatter_kernel<K, true>), dim3(grid_c),
                         dim3(256), 0, s, ka, kb, ia, ib, n, shift,
                         (const unsigned int*)counts, nchunks);
    else
      hipLaunchKernelGGL((radix_scatter_kernel<K, false>), dim3(grid_c),
                         dim3(256), 0, s, ka, kb, nullptr, nullptr, n, shift,
                         (const unsigned int*)counts, nchunks);
    K* tk = ka; ka = kb; kb = tk;
    long long* ti = ia; ia = ib; ib = ti;
  }
  // after an even number of passes the result is back in buffer A
  hipLaunchKernelGGL((sort_decode_kernel<T, K>), dim3(grid_e), dim3(256), 0,
                     s, ka, out, n);
  if (want_idx && idx_out != ia) {
    HIP_CHECK(hipMemcpyAsync(idx_out, ia, n * sizeof(long long),
                             hipMemcpyDeviceToDevice, s));
  }
  HIP_CHECK(hipGetLastError());
}

// per-row argsort positions: idx holds flat positions in [0, R*L);
// the user-facing index is position-within-row
__global__ void rowpos_kernel(const long long* __restrict__ idx,
                              long long* __restrict__ out, unsigned int L,
                              int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (long long)((unsigned)idx[i] % L);
}

// 2-D axis=-1 sort: LSD key passes with the flat index as payload, then
// LSD passes over the row id (idx / L). Stability makes the composite
// equal an independent stable sort of every row. Row passes use the
// block scatter unconditionally (the wave variant has no ROWDIG mode).
template <typename T, typename K>
static void radix_sort_rows_impl(const T* in, T* out, long long* idx_out,
                                 void* keys_a, void* keys_b, void* idx_a,
                                 void* idx_b, void* counts, void* dig_scratch,
                                 int64_t rows, int64_t cols, bool want_idx,
                                 hipStream_t s) {
  const int64_t n = rows * cols;
  int64_t nchunks = (n + kSortChunk - 1) / kSortChunk;
  int grid_c = (int)((nchunks + kWavesPerBlock - 1) / kWavesPerBlock);
  int grid_e = (int)std::min<int64_t>((n + 255) / 256, 4096);

  K* ka = (K*)keys_a;
  K* kb = (K*)keys_b;
  long long* ia = (long long*)idx_a;
  long long* ib = (long long*)idx_b;

  hipLaunchKernelGGL((sort_encode_kernel<T, K>), dim3(grid_e), dim3(256), 0,
                     s, in, ka, ia, n, 1);
  int row_passes = 1;
  while ((rows - 1) >> (8 * row_passes)) row_passes++;
  const int passes = (int)sizeof(K) + row_passes;
  int64_t nseg = (nchunks + kSeg - 1) / kSeg;
  unsigned int* segsum = (unsigned int*)((char*)dig_scratch + 256 * 8);
  int grid_s = (int)((nseg * 256 + 255) / 256);
  const unsigned int L = (unsigned int)cols;
  for (int p = 0; p < passes; p++) {
    bool rowdig = p >= (int)sizeof(K);
    int shift = (rowdig ? p - (int)sizeof(K) : p) * 8;
    if (rowdig)
      hipLaunchKernelGGL((radix_hist_kernel<K, true>), dim3(grid_c),
                         dim3(256), 0, s, ka, ia, L, n, shift,
                         (unsigned int*)counts, nchunks);
    else
      hipLaunchKernelGGL((radix_hist_kernel<K, false>), dim3(grid_c),
                         dim3(256), 0, s, ka, nullptr, 0u, n, shift,
                         (unsigned int*)counts, nchunks);
    hipLaunchKernelGGL(radix_segsum_kernel, dim3(grid_s), dim3(256), 0, s,
                       (const unsigned int*)counts, nchunks, nseg, segsum);
    hipLaunchKernelGGL(radix_segscan_kernel, dim3(1), dim3(256), 0, s,
                       segsum, nseg, (unsigned long long*)dig_scratch);
    hipLaunchKernelGGL(radix_fold_kernel, dim3(grid_s), dim3(256), 0, s,
                       (unsigned int*)counts, nchunks, nseg, segsum);
    if (rowdig)
      hipLaunchKernelGGL((radix_scatter_block_kernel<K, true, true>),
                         dim3((unsigned)nchunks), dim3(256), 0, s, ka, kb,
                         ia, ib, n, shift, (const unsigned int*)counts,
                         nchunks, L);
    else if (use_block_scatter())
      hipLaunchKernelGGL((radix_scatter_block_kernel<K, true>),
                         dim3((unsigned)nchunks), dim3(256), 0, s, ka, kb,
                         ia, ib, n, shift, (const unsigned int*)counts,
                         nchunks);
    else
      hipLaunchKernelGGL((radix_scatter_kernel<K, true>), dim3(grid_c),
                         dim3(256), 0, s, ka, kb, ia, ib, n, shift,
                         (const unsigned int*)counts, nchunks);
    K* tk = ka; ka = kb; kb = tk;
    long long* ti = ia; ia = ib; ib = ti;
  }
  hipLaunchKernelGGL((sort_decode_kernel<T, K>), dim3(grid_e), dim3(256), 0,
                     s, ka, out, n);
  if (want_idx)
    hipLaunchKernelGGL(rowpos_kernel, dim3(grid_e), dim3(256), 0, s, ia,
                       idx_out, L, n);
  HIP_CHECK(hipGetLastError());
}

}  // namespace

// scratch requirements (bytes), all caller-allocated:
//   keys_a/keys_b: n * sizeof(key) (8 for f64, 4 for f32)
//   idx_a/idx_b:   n * 8 when want_idx (idx_a doubles as the result)
//   counts:        nchunks(n) * 256 * 4
//   dig_scratch:   256*8 + nseg(n) * 256 * 4   (dig[256] then segsum)
int64_t radix_sort_nchunks(int64_t n) { return (n + kSortChunk - 1) / kSortChunk; }

// 2-D axis=-1 variant: idx_a/idx_b are REQUIRED (the row id rides in
// the payload); idx_out only read when want_idx. rows*cols <= 2^31.
void launch_radix_sort_rows(DType dt, const void* in, void* out,
                            void* idx_out, void* keys_a, void* keys_b,
                            void* idx_a, void* idx_b, void* counts,
                            void* dig_scratch, int64_t rows, int64_t cols,
                            bool want_idx, hipStream_t s) {
  if (dt == DType::F64)
    radix_sort_rows_impl<double, uint64_t>(
        (const double*)in, (double*)out, (long long*)idx_out, keys_a, keys_b,
        idx_a, idx_b, counts, dig_scratch, rows, cols, want_idx, s);
  else
    radix_sort_rows_impl<float, uint32_t>(
        (const float*)in, (float*)out, (long long*)idx_out, keys_a, keys_b,
        idx_a, idx_b, counts, dig_scratch, rows, cols, want_idx, s);
}

void launch_radix_sort(DType dt, const void* in, void* out, void* idx_out,
                       void* keys_a, void* keys_b, void* idx_a, void* idx_b,
                       void* counts, void* dig_scratch, int64_t n,
                       bool want_idx, hipStream_t s) {
  if (dt == DType::F64)
    radix_sort_impl<double, uint64_t>((const double*)in, (double*)out,
                                      (long long*)idx_out, keys_a, keys_b,
                                      idx_a, idx_b, counts, dig_scratch, n,
                                      want_idx, s);
  else
    radix_sort_impl<float, uint32_t>((const float*)in, (float*)out,
                                     (long long*)idx_out, keys_a, keys_b,
                                     idx_a, idx_b, counts, dig_scratch, n,
                                     want_idx, s);
}
