// Elementwise, reduction and RNG kernels for gfx950.
//
// Design notes (per the CDNA4 performance rules):
//  - memory-bound kernels are vectorized to 16 B/lane (double2 / float4):
//    scalar loads leave >2x bandwidth on the table;
//  - grids are capped at 8 blocks/CU x 256 CUs = 2048 workgroups with
//    grid-stride loops (fills all 8 XCDs, avoids launch overhead);
//  - reductions: wave64 __shfl_down tree -> per-wave LDS slot -> block
//    reduce -> one partial per block -> single-block stage 2 (no global
//    atomics, deterministic for a fixed grid);
//  - RNG is Philox4x32-10 (counter-based): stateless, any index range can
//    be generated independently at full bandwidth.

#include <algorithm>

#include "common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kMaxBlocks = 2048;  // 256 CUs x 8 blocks

inline int grid_for(int64_t work_items) {
  int64_t blocks = (work_items + kBlock - 1) / kBlock;
  if (blocks > kMaxBlocks) blocks = kMaxBlocks;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// ---------------------------------------------------------------------------
// elementwise
// ---------------------------------------------------------------------------
template <typename T>
struct Vec2 {
  T x, y;
};

template <typename T, int OP>
__device__ __forceinline__ T apply_unary(T v) {
  if constexpr (OP == (int)UnaryOp::Square) return v * v;
  if constexpr (OP == (int)UnaryOp::Neg) return -v;
  if constexpr (OP == (int)UnaryOp::Abs) return v < T(0) ? -v : v;
  if constexpr (OP == (int)UnaryOp::Sqrt) return (T)sqrt((double)v);
  if constexpr (OP == (int)UnaryOp::Exp) return (T)exp((double)v);
  if constexpr (OP == (int)UnaryOp::Log) return (T)log((double)v);
  if constexpr (OP == (int)UnaryOp::Sin) return (T)sin((double)v);
  if constexpr (OP == (int)UnaryOp::Cos) return (T)cos((double)v);
  if constexpr (OP == (int)UnaryOp::Tanh) return (T)tanh((double)v);
  if constexpr (OP == (int)UnaryOp::Floor) return (T)floor((double)v);
  if constexpr (OP == (int)UnaryOp::Ceil) return (T)ceil((double)v);
  // numpy round/rint: round-half-to-even
  if constexpr (OP == (int)UnaryOp::Rint) return (T)rint((double)v);
  if constexpr (OP == (int)UnaryOp::Trunc) return (T)trunc((double)v);
  if constexpr (OP == (int)UnaryOp::Sign)
    return v != v ? v : (v > T(0) ? T(1) : (v < T(0) ? T(-1) : v));
  if constexpr (OP == (int)UnaryOp::Log2) return (T)log2((double)v);
  if constexpr (OP == (int)UnaryOp::Log10) return (T)log10((double)v);
  if constexpr (OP == (int)UnaryOp::Exp2) return (T)exp2((double)v);
  if constexpr (OP == (int)UnaryOp::Expm1) return (T)expm1((double)v);
  if constexpr (OP == (int)UnaryOp::Log1p) return (T)log1p((double)v);
  if constexpr (OP == (int)UnaryOp::Cbrt) return (T)cbrt((double)v);
  if constexpr (OP == (int)UnaryOp::Tan) return (T)tan((double)v);
  if constexpr (OP == (int)UnaryOp::Arcsin) return (T)asin((double)v);
  if constexpr (OP == (int)UnaryOp::Arccos) return (T)acos((double)v);
  if constexpr (OP == (int)UnaryOp::Arctan) return (T)atan((double)v);
  if constexpr (OP == (int)UnaryOp::Sinh) return (T)sinh((double)v);
  if constexpr (OP == (int)UnaryOp::Cosh) return (T)cosh((double)v);
  return v;
}

template <typename T, int OP>
__device__ __forceinline__ T apply_bin(T a, T b) {
  if constexpr (OP == (int)BinOp::Add) return a + b;
  if constexpr (OP == (int)BinOp::Sub) return a - b;
  if constexpr (OP == (int)BinOp::Mul) return a * b;
  if constexpr (OP == (int)BinOp::Div) return a / b;
  // numpy NaN semantics: maximum/minimum propagate NaN from either side
  if constexpr (OP == (int)BinOp::Max)
    return (a != a) ? a : ((b != b) ? b : (a > b ? a : b));
  if constexpr (OP == (int)BinOp::Min)
    return (a != a) ? a : ((b != b) ? b : (a < b ? a : b));
  if constexpr (OP == (int)BinOp::Pow) return (T)pow((double)a, (double)b);
  return a;
}

// processes 2 elements of T per lane per step (16 B for f64, 8 B for f32;
// f32 uses 4/lane below)
template <typename T, int OP>
__global__ void unary_kernel(const T* __restrict__ in, T* __restrict__ out,
                             int64_t n) {
  using V2 = Vec2<T>;
  int64_t n2 = n / 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    V2 v = reinterpret_cast<const V2*>(in)[i];
    v.x = apply_unary<T, OP>(v.x);
    v.y = apply_unary<T, OP>(v.y);
    reinterpret_cast<V2*>(out)[i] = v;
  }
  // tail
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    out[n - 1] = apply_unary<T, OP>(in[n - 1]);
  }
}

template <typename T, int OP>
__global__ void binary_kernel(const T* __restrict__ a, const T* __restrict__ b,
                              T* __restrict__ out, int64_t n) {
  using V2 = Vec2<T>;
  int64_t n2 = n / 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    V2 va = reinterpret_cast<const V2*>(a)[i];
    V2 vb = reinterpret_cast<const V2*>(b)[i];
    va.x = apply_bin<T, OP>(va.x, vb.x);
    va.y = apply_bin<T, OP>(va.y, vb.y);
    reinterpret_cast<V2*>(out)[i] = va;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    out[n - 1] = apply_bin<T, OP>(a[n - 1], b[n - 1]);
  }
}

template <typename T, int OP>
__global__ void binary_scalar_kernel(const T* __restrict__ a, T scalar,
                                     T* __restrict__ out, int64_t n) {
  using V2 = Vec2<T>;
  int64_t n2 = n / 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    V2 va = reinterpret_cast<const V2*>(a)[i];
    va.x = apply_bin<T, OP>(va.x, scalar);
    va.y = apply_bin<T, OP>(va.y, scalar);
    reinterpret_cast<V2*>(out)[i] = va;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    out[n - 1] = apply_bin<T, OP>(a[n - 1], scalar);
  }
}

template <typename S, typename D>
__global__ void convert_kernel(const S* __restrict__ in, D* __restrict__ out,
                               int64_t n) {
  using VS = Vec2<S>;
  using VD = Vec2<D>;
  int64_t n2 = n / 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    VS v = reinterpret_cast<const VS*>(in)[i];
    reinterpret_cast<VD*>(out)[i] = {(D)v.x, (D)v.y};
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    out[n - 1] = (D)in[n - 1];
  }
}

// ---------------------------------------------------------------------------
// reduction (sum / sum-of-squares)
// ---------------------------------------------------------------------------
template <typename T>
__device__ __forceinline__ double wave_reduce_sum(double v) {
  // wave64 butterfly; __shfl_down over the full 64-lane wave
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

template <typename T, bool SQUARE>
__global__ void sum_stage1(const T* __restrict__ in, T* __restrict__ partials,
                           int64_t n) {
  using V2 = Vec2<T>;
  int64_t n2 = n / 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  double acc = 0.0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    V2 v = reinterpret_cast<const V2*>(in)[i];
    if constexpr (SQUARE) {
      acc += (double)v.x * (double)v.x + (double)v.y * (double)v.y;
    } else {
      acc += (double)v.x + (double)v.y;
    }
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    double t = (double)in[n - 1];
    acc += SQUARE ? t * t : t;
  }
  acc = wave_reduce_sum<T>(acc);
  __shared__ double wave_sums[kBlock / 64];
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) wave_sums[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    double total = 0.0;
    for (int w = 0; w < (int)blockDim.x / 64; w++) total += wave_sums[w];
    partials[blockIdx.x] = (T)total;
  }
}

template <typename T>
__global__ void sum_stage2(const T* __restrict__ partials, T* __restrict__ out,
                           int n) {
  double acc = 0.0;
  for (int i = threadIdx.x; i < n; i += blockDim.x) acc += (double)partials[i];
  acc = wave_reduce_sum<T>(acc);
  __shared__ double wave_sums[kBlock / 64];
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) wave_sums[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    double total = 0.0;
    for (int w = 0; w < (int)blockDim.x / 64; w++) total += wave_sums[w];
    out[0] = (T)total;
  }
}

template <bool MAXOP>
__device__ __forceinline__ double minmax_combine(double acc, double v) {
  if (v != v) return v;  // NaN propagates (numpy np.max/np.min)
  if (acc != acc) return acc;
  if constexpr (MAXOP) return v > acc ? v : acc;
  else return v < acc ? v : acc;
}

template <bool MAXOP>
__device__ __forceinline__ double wave_reduce_minmax(double v) {
  for (int off = 32; off > 0; off >>= 1)
    v = minmax_combine<MAXOP>(v, __shfl_down(v, off, 64));
  return v;
}

template <typename T, bool MAXOP>
__global__ void minmax_stage1(const T* __restrict__ in,
                              T* __restrict__ partials, int64_t n) {
  using V2 = Vec2<T>;
  int64_t n2 = n / 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  double acc = MAXOP ? -INFINITY : INFINITY;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    V2 v = reinterpret_cast<const V2*>(in)[i];
    acc = minmax_combine<MAXOP>(acc, (double)v.x);
    acc = minmax_combine<MAXOP>(acc, (double)v.y);
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && (n & 1)) {
    acc = minmax_combine<MAXOP>(acc, (double)in[n - 1]);
  }
  acc = wave_reduce_minmax<MAXOP>(acc);
  __shared__ double wave_vals[kBlock / 64];
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) wave_vals[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    double total = MAXOP ? -INFINITY : INFINITY;
    for (int w = 0; w < (int)blockDim.x / 64; w++)
      total = minmax_combine<MAXOP>(total, wave_vals[w]);
    partials[blockIdx.x] = (T)total;
  }
}

template <typename T, bool MAXOP>
__global__ void minmax_stage2(const T* __restrict__ partials,
                              T* __restrict__ out, int n) {
  double acc = MAXOP ? -INFINITY : INFINITY;
  for (int i = threadIdx.x; i < n; i += blockDim.x)
    acc = minmax_combine<MAXOP>(acc, (double)partials[i]);
  acc = wave_reduce_minmax<MAXOP>(acc);
  __shared__ double wave_vals[kBlock / 64];
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) wave_vals[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    double total = MAXOP ? -INFINITY : INFINITY;
    for (int w = 0; w < (int)blockDim.x / 64; w++)
      total = minmax_combine<MAXOP>(total, wave_vals[w]);
    out[0] = (T)total;
  }
}

// ---------------------------------------------------------------------------
// cumsum: three-pass block scan. Pass 1: block b scans its contiguous
// chunk (per-thread sequential sums -> LDS exclusive scan of thread
// totals -> rewrite with prefixes) and stores its chunk total. Pass 2:
// one block exclusive-scans the totals. Pass 3: add each block's offset.
// Accumulation in double (f32 in -> f32 out, f64 exact); element order
// within a thread is sequential, so rounding matches numpy to ~1 ulp
// per partial-regrouping (documented).
// ---------------------------------------------------------------------------
template <typename T>
__global__ void cumsum_block_kernel(const T* __restrict__ in,
                                    T* __restrict__ out,
                                    double* __restrict__ totals, int64_t n,
                                    int64_t chunk) {
  __shared__ double tsum[kBlock];
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t end = start + chunk < n ? start + chunk : n;
  int64_t per = (chunk + blockDim.x - 1) / blockDim.x;
  int64_t t0 = start + threadIdx.x * per;
  int64_t t1 = t0 + per < end ? t0 + per : end;
  double acc = 0.0;
  for (int64_t i = t0; i < t1; i++) acc += (double)in[i];
  tsum[threadIdx.x] = acc;
  __syncthreads();
  // exclusive scan of thread totals (single thread: kBlock=256 adds,
  // negligible vs the per-thread sweeps)
  if (threadIdx.x == 0) {
    double run = 0.0;
    for (int i = 0; i < (int)blockDim.x; i++) {
      double v = tsum[i];
      tsum[i] = run;
      run += v;
    }
    if (totals) totals[blockIdx.x] = run;
  }
  __syncthreads();
  double run = tsum[threadIdx.x];
  for (int64_t i = t0; i < t1; i++) {
    run += (double)in[i];
    out[i] = (T)run;
  }
}

__global__ void cumsum_totals_kernel(double* __restrict__ totals, int nb) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    double run = 0.0;
    for (int i = 0; i < nb; i++) {
      double v = totals[i];
      totals[i] = run;
      run += v;
    }
  }
}

template <typename T>
__global__ void cumsum_add_offsets_kernel(T* __restrict__ out,
                                          const double* __restrict__ totals,
                                          int64_t n, int64_t chunk) {
  int64_t start = (int64_t)blockIdx.x * chunk;
  int64_t end = start + chunk < n ? start + chunk : n;
  double off = totals[blockIdx.x];
  if (off == 0.0) return;
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
    out[i] = (T)((double)out[i] + off);
}

// ---------------------------------------------------------------------------
// u8 mask logic: 0=and 1=or 2=xor 3=andnot(a & ~b); unary not via
// xor with an all-ones operand is avoided -- NOT uses op 4 (b ignored)
// ---------------------------------------------------------------------------
__global__ void mask_logic_kernel(const unsigned char* __restrict__ a,
                                  const unsigned char* __restrict__ b,
                                  unsigned char* __restrict__ out, int64_t n,
                                  int op) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    unsigned char av = a[i];
    unsigned char bv = b ? b[i] : 0;
    unsigned char r;
    switch (op) {
      case 0: r = (av && bv) ? 1 : 0; break;
      case 1: r = (av || bv) ? 1 : 0; break;
      case 2: r = (!av != !bv) ? 1 : 0; break;
      case 3: r = (av && !bv) ? 1 : 0; break;
      default: r = av ? 0 : 1; break;  // NOT
    }
    out[i] = r;
  }
}

// ---------------------------------------------------------------------------
// histogram-selection primitives (np.median/percentile without a sort):
// range histogram with per-block LDS accumulation, and range-compaction
// extraction for the exact finish on a narrowed candidate bin.
// ---------------------------------------------------------------------------
constexpr int kHistBins = 4096;

// exact != 0: after the float bin estimate, correct against the actual
// linspace edges (edge_i = lo + i*step, step = (hi-lo)/bins) the way
// numpy's histogram does -- boundary-landing values (integer data on
// integer edges) then bin EXACTLY like numpy.
template <typename T>
__global__ void hist_range_kernel(const T* __restrict__ in, int64_t n,
                                  double lo, double hi, double inv_width,
                                  int bins, int exact, double step,
                                  unsigned long long* __restrict__ counts,
                                  unsigned long long* __restrict__ extra) {
  // extra[0] = NaN count, extra[1] = count below lo, extra[2] = above hi
  __shared__ unsigned int local[kHistBins];
  for (int i = threadIdx.x; i < bins; i += blockDim.x) local[i] = 0;
  __syncthreads();
  unsigned long long nan_c = 0, below_c = 0, above_c = 0;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double v = (double)in[i];
    if (v != v) {
      nan_c++;
    } else if (v < lo) {
      below_c++;
    } else if (v > hi) {
      above_c++;
    } else {
      int b = (int)((v - lo) * inv_width);
      if (b >= bins) b = bins - 1;  // v == hi (or fp rounding at the edge)
      if (b < 0) b = 0;
      if (exact) {
        if (v < lo + b * step) b--;
        else if (b != bins - 1 && v >= lo + (b + 1) * step) b++;
        if (b < 0) b = 0;
      }
      atomicAdd(&local[b], 1u);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < bins; i += blockDim.x)
    if (local[i]) atomicAdd(&counts[i], (unsigned long long)local[i]);
  if (nan_c) atomicAdd(&extra[0], nan_c);
  if (below_c) atomicAdd(&extra[1], below_c);
  if (above_c) atomicAdd(&extra[2], above_c);
}

template <typename T>
__global__ void extract_range_kernel(const T* __restrict__ in, int64_t n,
                                     double lo, double hi,
                                     double* __restrict__ out,
                                     unsigned long long* __restrict__ counter,
                                     int64_t cap) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double v = (double)in[i];
    if (v >= lo && v <= hi) {
      unsigned long long idx = atomicAdd(counter, 1ull);
      if ((int64_t)idx < cap) out[idx] = v;
    }
  }
}

// ---------------------------------------------------------------------------
// boolean-mask ops: comparisons -> u8 masks, select (np.where),
// masked fill, mask popcount. CmpOp: 0 lt, 1 le, 2 gt, 3 ge, 4 eq, 5 ne
// (NaN compares false like numpy, except ne where NaN != x is true).
// ---------------------------------------------------------------------------
__device__ __forceinline__ bool cmp_apply(int op, double x, double y) {
  switch (op) {
    case 0: return x < y;
    case 1: return x <= y;
    case 2: return x > y;
    case 3: return x >= y;
    case 4: return x == y;
    default: return x != y;
  }
}

template <typename T, bool SCALAR>
__global__ void compare_kernel(const T* __restrict__ a,
                               const T* __restrict__ b, double scalar,
                               unsigned char* __restrict__ out, int64_t n,
                               int op) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double y = SCALAR ? scalar : (double)b[i];
    out[i] = cmp_apply(op, (double)a[i], y) ? 1 : 0;
  }
}

// out[i] = mask[i] ? a : b, with a/b each independently an array or a
// scalar (null pointer selects the scalar)
template <typename T>
__global__ void where_kernel(const unsigned char* __restrict__ mask,
                             const T* __restrict__ pa, double sa,
                             const T* __restrict__ pb, double sb,
                             T* __restrict__ out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (mask[i])
      out[i] = pa ? pa[i] : (T)sa;
    else
      out[i] = pb ? pb[i] : (T)sb;
  }
}

template <typename T>
__global__ void masked_fill_kernel(T* __restrict__ data,
                                   const unsigned char* __restrict__ mask,
                                   double value, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (mask[i]) data[i] = (T)value;
  }
}

__global__ void mask_count_stage1(const unsigned char* __restrict__ mask,
                                  int64_t* __restrict__ partials, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t acc = 0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride)
    acc += mask[i] ? 1 : 0;
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  __shared__ int64_t wsum[kBlock / 64];
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) wsum[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    int64_t total = 0;
    for (int w = 0; w < (int)blockDim.x / 64; w++) total += wsum[w];
    partials[blockIdx.x] = total;
  }
}

__global__ void mask_count_stage2(const int64_t* __restrict__ partials,
                                  int64_t* __restrict__ out, int nparts) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    int64_t total = 0;
    for (int i = 0; i < nparts; i++) total += partials[i];
    out[0] = total;
  }
}

// ---------------------------------------------------------------------------
// broadcast binary: a viewed [outer][inner] combined with a vector b --
// mode 0: b[inner] broadcast along outer (row vector, e.g. x - mean0)
// mode 1: b[outer] broadcast along inner (column vector, keepdims shape)
// ---------------------------------------------------------------------------
template <typename T, int MODE>
__global__ void binary_bcast_kernel(const T* __restrict__ a,
                                    const T* __restrict__ b,
                                    T* __restrict__ out, int64_t outer,
                                    int64_t inner, int op) {
  int64_t n = outer * inner;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    T av = a[i];
    T bv = MODE == 0 ? b[i % inner] : b[i / inner];
    double x = (double)av, y = (double)bv, r;
    switch (op) {
      case 0: r = x + y; break;
      case 1: r = x - y; break;
      case 2: r = x * y; break;
      case 3: r = x / y; break;
      case 4: r = (y != y) ? y : (x != x ? x : (x > y ? x : y)); break;
      case 5: r = (y != y) ? y : (x != x ? x : (x < y ? x : y)); break;
      default: r = pow(x, y); break;
    }
    out[i] = (T)r;
  }
}

// ---------------------------------------------------------------------------
// argmax/argmin: two-stage (value, index) reduction. numpy semantics:
// FIRST occurrence wins ties; NaN wins (propagates) like np.argmax.
// ---------------------------------------------------------------------------
template <bool MAXOP>
__device__ __forceinline__ bool arg_better(double v, int64_t i, double bv,
                                           int64_t bi) {
  // NaN beats everything (numpy returns the first NaN's index);
  // among equals the SMALLER index wins
  bool v_nan = v != v, b_nan = bv != bv;
  if (v_nan != b_nan) return v_nan;
  if (v_nan && b_nan) return i < bi;
  if (v != bv) return MAXOP ? v > bv : v < bv;
  return i < bi;
}

template <typename T, bool MAXOP>
__global__ void argminmax_stage1(const T* __restrict__ in,
                                 double* __restrict__ pvals,
                                 int64_t* __restrict__ pidx, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  double best = MAXOP ? -INFINITY : INFINITY;
  int64_t besti = 0;
  bool any = false;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double v = (double)in[i];
    if (!any || arg_better<MAXOP>(v, i, best, besti)) {
      best = v;
      besti = i;
      any = true;
    }
  }
  if (!any) besti = n;  // lane saw nothing: neutral (index n loses ties)
  // wave64 butterfly on (value, index) pairs
  for (int off = 32; off > 0; off >>= 1) {
    double ov = __shfl_down(best, off, 64);
    int64_t oi = __shfl_down(besti, off, 64);
    if (oi < n && (besti >= n || arg_better<MAXOP>(ov, oi, best, besti))) {
      best = ov;
      besti = oi;
    }
  }
  __shared__ double wv[kBlock / 64];
  __shared__ int64_t wi[kBlock / 64];
  int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) {
    wv[wave] = best;
    wi[wave] = besti;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < (int)blockDim.x / 64; w++) {
      if (wi[w] < n &&
          (wi[0] >= n || arg_better<MAXOP>(wv[w], wi[w], wv[0], wi[0]))) {
        wv[0] = wv[w];
        wi[0] = wi[w];
      }
    }
    pvals[blockIdx.x] = wv[0];
    pidx[blockIdx.x] = wi[0];
  }
}

template <bool MAXOP>
__global__ void argminmax_stage2(const double* __restrict__ pvals,
                                 const int64_t* __restrict__ pidx,
                                 int64_t* __restrict__ out, int nparts,
                                 int64_t n) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    double best = pvals[0];
    int64_t besti = pidx[0];
    for (int i = 1; i < nparts; i++) {
      if (pidx[i] < n &&
          (besti >= n || arg_better<MAXOP>(pvals[i], pidx[i], best, besti))) {
        best = pvals[i];
        besti = pidx[i];
      }
    }
    out[0] = besti;
  }
}

// ---------------------------------------------------------------------------
// axis-wise reduction: in viewed as [outer][red][inner] -> out[outer][inner]
// (covers any single-axis reduce of a contiguous N-D array). Two shapes:
//  - inner > 1: one thread per (outer, inner) output element; lanes read
//    consecutive `inner` addresses per step -> coalesced;
//  - inner == 1 (last-axis reduce): one wave64 per slice, lanes stride
//    the reduced axis, cross-lane butterfly combines.
// MODE matches ReduceOp: 0 sum, 1 sum-of-squares, 2 max, 3 min.
// ---------------------------------------------------------------------------
template <int MODE>
__device__ __forceinline__ double axis_init() {
  if constexpr (MODE == 2) return -INFINITY;
  else if constexpr (MODE == 3) return INFINITY;
  else return 0.0;
}

template <int MODE>
__device__ __forceinline__ double axis_combine(double acc, double v) {
  if constexpr (MODE == 0) return acc + v;
  else if constexpr (MODE == 1) return acc + v * v;
  else return minmax_combine<MODE == 2>(acc, v);
}

template <int MODE>
__device__ __forceinline__ double axis_wave_reduce(double v) {
  if constexpr (MODE <= 1) return wave_reduce_sum<double>(v);
  else return wave_reduce_minmax<MODE == 2>(v);
}

template <typename T, int MODE>
__global__ void reduce_axis_inner_kernel(const T* __restrict__ in,
                                         T* __restrict__ out, int64_t outer,
                                         int64_t red, int64_t inner) {
  int64_t total = outer * inner;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    int64_t o = idx / inner, i = idx % inner;
    const T* p = in + o * red * inner + i;
    double acc = axis_init<MODE>();
    for (int64_t r = 0; r < red; r++)
      acc = axis_combine<MODE>(acc, (double)p[r * inner]);
    out[idx] = (T)acc;
  }
}

template <typename T, int MODE>
__global__ void reduce_axis_last_kernel(const T* __restrict__ in,
                                        T* __restrict__ out, int64_t outer,
                                        int64_t red) {
  const int waves_per_block = blockDim.x >> 6;
  const int lane = threadIdx.x & 63;
  int64_t stride = (int64_t)gridDim.x * waves_per_block;
  for (int64_t slice = (int64_t)blockIdx.x * waves_per_block +
                       (threadIdx.x >> 6);
       slice < outer; slice += stride) {
    const T* p = in + slice * red;
    double acc = axis_init<MODE>();
    for (int64_t r = lane; r < red; r += 64)
      acc = axis_combine<MODE>(acc, (double)p[r]);
    acc = axis_wave_reduce<MODE>(acc);
    if (lane == 0) out[slice] = (T)acc;
  }
}

// ---------------------------------------------------------------------------
// Philox4x32-10 uniform RNG
// ---------------------------------------------------------------------------
__device__ __forceinline__ void philox_round(uint32_t& c0, uint32_t& c1,
                                             uint32_t& c2, uint32_t& c3,
                                             uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  uint32_t hi0 = __umulhi(M0, c0), lo0 = M0 * c0;
  uint32_t hi1 = __umulhi(M1, c2), lo1 = M1 * c2;
  uint32_t n0 = hi1 ^ c1 ^ k0;
  uint32_t n1 = lo1;
  uint32_t n2 = hi0 ^ c3 ^ k1;
  uint32_t n3 = lo0;
  c0 = n0; c1 = n1; c2 = n2; c3 = n3;
}

__device__ __forceinline__ void philox10(uint64_t ctr, uint64_t seed,
                                         uint32_t out[4]) {
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
  uint32_t c0 = (uint32_t)ctr, c1 = (uint32_t)(ctr >> 32), c2 = 0x85EBCA6Bu,
           c3 = 0xC2B2AE35u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; r++) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += W0;
    k1 += W1;
  }
  out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
}

// each counter yields 2 doubles (53-bit mantissas) or 4 floats
__global__ void rand_f64_kernel(double* __restrict__ out, int64_t n,
                                uint64_t seed, uint64_t offset) {
  int64_t pairs = (n + 1) / 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const double scale = 1.0 / 9007199254740992.0;  // 2^-53
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < pairs;
       i += stride) {
    uint32_t r[4];
    philox10(offset + (uint64_t)i, seed, r);
    uint64_t u0 = ((uint64_t)r[1] << 32) | r[0];
    uint64_t u1 = ((uint64_t)r[3] << 32) | r[2];
    double d0 = (double)(u0 >> 11) * scale;
    double d1 = (double)(u1 >> 11) * scale;
    int64_t j = i * 2;
    if (j + 1 < n) {
      reinterpret_cast<Vec2<double>*>(out)[i] = {d0, d1};
    } else if (j < n) {
      out[j] = d0;
    }
  }
}

// Box-Muller on Philox pairs: each counter yields two N(mu, sigma^2)
// doubles. u1 is mapped to (0,1] so log(u1) is always finite.
__global__ void randn_f64_kernel(double* __restrict__ out, int64_t n,
                                 uint64_t seed, uint64_t offset, double mu,
                                 double sigma) {
  int64_t pairs = (n + 1) / 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const double scale = 1.0 / 9007199254740992.0;  // 2^-53
  const double two_pi = 6.283185307179586476925286766559;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < pairs;
       i += stride) {
    uint32_t r[4];
    philox10(offset + (uint64_t)i, seed, r);
    uint64_t u0 = ((uint64_t)r[1] << 32) | r[0];
    uint64_t u1 = ((uint64_t)r[3] << 32) | r[2];
    double a = (double)((u0 >> 11) + 1) * scale;  // (0, 1]
    double b = (double)(u1 >> 11) * scale;        // [0, 1)
    double radius = sqrt(-2.0 * log(a));
    // f32 trig for the angle: ~1e-7 relative error on a uniform angle is
    // statistically invisible in the variates and halves the kernel cost
    // (f64 sincos dominates; the radius keeps full f64 precision)
    float s_f, c_f;
    __sincosf((float)(two_pi * b), &s_f, &c_f);
    double z0 = mu + sigma * (radius * (double)c_f);
    double z1 = mu + sigma * (radius * (double)s_f);
    int64_t j = i * 2;
    if (j + 1 < n) {
      reinterpret_cast<Vec2<double>*>(out)[i] = {z0, z1};
    } else if (j < n) {
      out[j] = z0;
    }
  }
}

__global__ void rand_f32_kernel(float* __restrict__ out, int64_t n,
                                uint64_t seed, uint64_t offset) {
  int64_t quads = (n + 3) / 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float scale = 1.0f / 16777216.0f;  // 2^-24
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < quads;
       i += stride) {
    uint32_t r[4];
    philox10(offset + (uint64_t)i, seed, r);
    int64_t j = i * 4;
    if (j + 3 < n) {
      float4 v = {(float)(r[0] >> 8) * scale, (float)(r[1] >> 8) * scale,
                  (float)(r[2] >> 8) * scale, (float)(r[3] >> 8) * scale};
      reinterpret_cast<float4*>(out)[i] = v;
    } else {
      for (int q = 0; q < 4 && j + q < n; q++)
        out[j + q] = (float)(r[q] >> 8) * scale;
    }
  }
}

// LDS-tiled 2-D transpose: 32x32 element tiles, 256 threads (each
// thread moves 4 elements), +1-padded LDS so both the row-coalesced
// read and the row-coalesced (transposed) write are conflict-free.
template <typename T>
__global__ void transpose_kernel(const T* __restrict__ in, T* __restrict__ out,
                                 int64_t rows, int64_t cols,
                                 int64_t tiles_c) {
  __shared__ T tile[32][33];
  int64_t tr = blockIdx.x / tiles_c;
  int64_t tc = blockIdx.x % tiles_c;
  int64_t r0 = tr * 32, c0 = tc * 32;
  int tx = threadIdx.x & 31;
  int ty = threadIdx.x >> 5;  // 0..7
#pragma unroll
  for (int i = 0; i < 4; i++) {
    int64_t r = r0 + ty + i * 8;
    int64_t c = c0 + tx;
    if (r < rows && c < cols) tile[ty + i * 8][tx] = in[r * cols + c];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 4; i++) {
    int64_t r = c0 + ty + i * 8;   // output row = input col
    int64_t c = r0 + tx;           // output col = input row
    if (r < cols && c < rows) out[r * rows + c] = tile[tx][ty + i * 8];
  }
}

// row-wise inclusive prefix sum (np.cumsum axis=-1 on 2-D): one block
// per row, 1024-element chunks (4 thread-serial elements + a 256-wide
// Hillis-Steele over thread sums), running double carry across chunks
// -- accumulation in double like the flat cumsum.
template <typename T>
__global__ void cumsum_rows_kernel(const T* __restrict__ in,
                                   T* __restrict__ out, int64_t rows,
                                   int64_t cols) {
  __shared__ double sums[256];
  const int tid = threadIdx.x;
  const int64_t base = (int64_t)blockIdx.x * cols;
  double carry = 0.0;
  for (int64_t chunk0 = 0; chunk0 < cols; chunk0 += 1024) {
    int64_t e0 = chunk0 + tid * 4;
    double v[4];
    double s = 0.0;
#pragma unroll
    for (int j = 0; j < 4; j++) {
      int64_t e = e0 + j;
      v[j] = e < cols ? (double)in[base + e] : 0.0;
      s += v[j];
    }
    sums[tid] = s;
    double x = s;
    for (int off = 1; off < 256; off <<= 1) {
      __syncthreads();
      double add = tid >= off ? sums[tid - off] : 0.0;
      __syncthreads();
      x += add;
      sums[tid] = x;
    }
    __syncthreads();
    double total = sums[255];
    double run = x - s + carry;  // exclusive prefix for this thread
#pragma unroll
    for (int j = 0; j < 4; j++) {
      int64_t e = e0 + j;
      if (e < cols) {
        run += v[j];
        out[base + e] = (T)run;
      }
    }
    carry += total;
    __syncthreads();  // sums reused next chunk
  }
}

// np.diff along the last axis: out[o][i] = in[o][i+1] - in[o][i]
template <typename T>
__global__ void diff_kernel(const T* __restrict__ in, T* __restrict__ out,
                            int64_t outer, int64_t inner) {
  int64_t n_out = outer * (inner - 1);
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; t < n_out;
       t += stride) {
    int64_t o = t / (inner - 1);
    int64_t i = t % (inner - 1);
    out[t] = in[o * inner + i + 1] - in[o * inner + i];
  }
}

// np.searchsorted: one thread per query, binary search with numpy's
// sort ordering (NaN compares greater than everything, so NaN queries
// land among the trailing NaNs exactly like numpy)
template <typename T>
__device__ __forceinline__ bool np_lt(T x, T y) {
  return x < y || (y != y && x == x);
}

template <typename T, bool RIGHT>
__global__ void searchsorted_kernel(const T* __restrict__ a, int64_t n,
                                    const T* __restrict__ v, int64_t m,
                                    long long* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; t < m;
       t += stride) {
    T val = v[t];
    int64_t lo = 0, hi = n;
    while (lo < hi) {
      int64_t mid = (lo + hi) >> 1;
      bool go_right = RIGHT ? !np_lt(val, a[mid]) : np_lt(a[mid], val);
      if (go_right)
        lo = mid + 1;
      else
        hi = mid;
    }
    out[t] = lo;
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
template <typename T>
static void launch_unary_t(UnaryOp op, const T* in, T* out, int64_t n,
                           hipStream_t s) {
  int grid = grid_for(n / 2 + 1);
  switch (op) {
#define CASE(OP)                                                         \
  case UnaryOp::OP:                                                      \
    hipLaunchKernelGGL((unary_kernel<T, (int)UnaryOp::OP>), dim3(grid),  \
                       dim3(kBlock), 0, s, in, out, n);                  \
    break;
    CASE(Square) CASE(Neg) CASE(Abs) CASE(Sqrt) CASE(Exp)
    CASE(Log) CASE(Sin) CASE(Cos) CASE(Tanh)
    CASE(Floor) CASE(Ceil) CASE(Rint) CASE(Trunc) CASE(Sign)
    CASE(Log2) CASE(Log10) CASE(Exp2) CASE(Expm1) CASE(Log1p)
    CASE(Cbrt) CASE(Tan) CASE(Arcsin) CASE(Arccos) CASE(Arctan)
    CASE(Sinh) CASE(Cosh)
#undef CASE
  }
  HIP_CHECK(hipGetLastError());
}

void launch_unary(DType dt, UnaryOp op, const void* in, void* out, int64_t n,
                  hipStream_t stream) {
  if (dt == DType::F64)
    launch_unary_t(op, (const double*)in, (double*)out, n, stream);
  else
    launch_unary_t(op, (const float*)in, (float*)out, n, stream);
}

template <typename T>
static void launch_binary_t(BinOp op, const T* a, const T* b, T* out,
                            int64_t n, hipStream_t s) {
  int grid = grid_for(n / 2 + 1);
  switch (op) {
#define CASE(OP)                                                          \
  case BinOp::OP:                                                         \
    hipLaunchKernelGGL((binary_kernel<T, (int)BinOp::OP>), dim3(grid),    \
                       dim3(kBlock), 0, s, a, b, out, n);                 \
    break;
    CASE(Add) CASE(Sub) CASE(Mul) CASE(Div) CASE(Max) CASE(Min) CASE(Pow)
#undef CASE
  }
  HIP_CHECK(hipGetLastError());
}

void launch_binary(DType dt, BinOp op, const void* a, const void* b, void* out,
                   int64_t n, hipStream_t stream) {
  if (dt == DType::F64)
    launch_binary_t(op, (const double*)a, (const double*)b, (double*)out, n,
                    stream);
  else
    launch_binary_t(op, (const float*)a, (const float*)b, (float*)out, n,
                    stream);
}

template <typename T>
static void launch_binary_scalar_t(BinOp op, const T* a, T scalar, T* out,
                                   int64_t n, hipStream_t s) {
  int grid = grid_for(n / 2 + 1);
  switch (op) {
#define CASE(OP)                                                             \
  case BinOp::OP:                                                            \
    hipLaunchKernelGGL((binary_scalar_kernel<T, (int)BinOp::OP>), dim3(grid), \
                       dim3(kBlock), 0, s, a, scalar, out, n);               \
    break;
    CASE(Add) CASE(Sub) CASE(Mul) CASE(Div) CASE(Max) CASE(Min) CASE(Pow)
#undef CASE
  }
  HIP_CHECK(hipGetLastError());
}

void launch_binary_scalar(DType dt, BinOp op, const void* a, double scalar,
                          void* out, int64_t n, hipStream_t stream) {
  if (dt == DType::F64)
    launch_binary_scalar_t(op, (const double*)a, scalar, (double*)out, n,
                           stream);
  else
    launch_binary_scalar_t(op, (const float*)a, (float)scalar, (float*)out, n,
                           stream);
}

int reduce_num_partials(int64_t n) { return grid_for(n / 2 + 1); }

template <typename T>
static void launch_sum_t(ReduceOp mode, const T* in, T* partials,
                         T* out_scalar, int64_t n, hipStream_t s) {
  int grid = grid_for(n / 2 + 1);
  switch (mode) {
    case ReduceOp::Sum:
      hipLaunchKernelGGL((sum_stage1<T, false>), dim3(grid), dim3(kBlock), 0,
                         s, in, partials, n);
      break;
    case ReduceOp::SumSquares:
      hipLaunchKernelGGL((sum_stage1<T, true>), dim3(grid), dim3(kBlock), 0,
                         s, in, partials, n);
      break;
    case ReduceOp::Max:
      hipLaunchKernelGGL((minmax_stage1<T, true>), dim3(grid), dim3(kBlock),
                         0, s, in, partials, n);
      break;
    case ReduceOp::Min:
      hipLaunchKernelGGL((minmax_stage1<T, false>), dim3(grid), dim3(kBlock),
                         0, s, in, partials, n);
      break;
  }
  HIP_CHECK(hipGetLastError());
  if (mode == ReduceOp::Max)
    hipLaunchKernelGGL((minmax_stage2<T, true>), dim3(1), dim3(kBlock), 0, s,
                       partials, out_scalar, grid);
  else if (mode == ReduceOp::Min)
    hipLaunchKernelGGL((minmax_stage2<T, false>), dim3(1), dim3(kBlock), 0, s,
                       partials, out_scalar, grid);
  else
    hipLaunchKernelGGL((sum_stage2<T>), dim3(1), dim3(kBlock), 0, s, partials,
                       out_scalar, grid);
  HIP_CHECK(hipGetLastError());
}

template <typename T>
static void launch_cumsum_t(const T* in, T* out, void* totals, int64_t n,
                            hipStream_t s) {
  int nb = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 2048);
  int64_t chunk = (n + nb - 1) / nb;
  hipLaunchKernelGGL((cumsum_block_kernel<T>), dim3(nb), dim3(kBlock), 0, s,
                     in, out, (double*)totals, n, chunk);
  hipLaunchKernelGGL(cumsum_totals_kernel, dim3(1), dim3(64), 0, s,
                     (double*)totals, nb);
  hipLaunchKernelGGL((cumsum_add_offsets_kernel<T>), dim3(nb), dim3(kBlock),
                     0, s, out, (const double*)totals, n, chunk);
  HIP_CHECK(hipGetLastError());
}

void launch_cumsum(DType dt, const void* in, void* out, void* totals,
                   int64_t n, hipStream_t s) {
  if (dt == DType::F64)
    launch_cumsum_t((const double*)in, (double*)out, totals, n, s);
  else
    launch_cumsum_t((const float*)in, (float*)out, totals, n, s);
}

void launch_mask_logic(const void* a, const void* b, void* out, int64_t n,
                       int op, hipStream_t s) {
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 4096);
  hipLaunchKernelGGL(mask_logic_kernel, dim3(grid), dim3(kBlock), 0, s,
                     (const unsigned char*)a, (const unsigned char*)b,
                     (unsigned char*)out, n, op);
  HIP_CHECK(hipGetLastError());
}

void launch_hist_range(DType dt, const void* in, int64_t n, double lo,
                       double hi, double inv_width, int bins, int exact,
                       void* counts, void* extra, hipStream_t s) {
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 2048);
  double step = (hi - lo) / bins;  // numpy linspace's step
  if (dt == DType::F64)
    hipLaunchKernelGGL((hist_range_kernel<double>), dim3(grid), dim3(kBlock),
                       0, s, (const double*)in, n, lo, hi, inv_width, bins,
                       exact, step, (unsigned long long*)counts,
                       (unsigned long long*)extra);
  else
    hipLaunchKernelGGL((hist_range_kernel<float>), dim3(grid), dim3(kBlock),
                       0, s, (const float*)in, n, lo, hi, inv_width, bins,
                       exact, step, (unsigned long long*)counts,
                       (unsigned long long*)extra);
  HIP_CHECK(hipGetLastError());
}

void launch_extract_range(DType dt, const void* in, int64_t n, double lo,
                          double hi, void* out, void* counter, int64_t cap,
                          hipStream_t s) {
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 2048);
  if (dt == DType::F64)
    hipLaunchKernelGGL((extract_range_kernel<double>), dim3(grid),
                       dim3(kBlock), 0, s, (const double*)in, n, lo, hi,
                       (double*)out, (unsigned long long*)counter, cap);
  else
    hipLaunchKernelGGL((extract_range_kernel<float>), dim3(grid),
                       dim3(kBlock), 0, s, (const float*)in, n, lo, hi,
                       (double*)out, (unsigned long long*)counter, cap);
  HIP_CHECK(hipGetLastError());
}

template <typename T>
static void launch_compare_t(int op, const T* a, const T* b, double scalar,
                             unsigned char* out, int64_t n, hipStream_t s) {
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 4096);
  if (b)
    hipLaunchKernelGGL((compare_kernel<T, false>), dim3(grid), dim3(kBlock),
                       0, s, a, b, scalar, out, n, op);
  else
    hipLaunchKernelGGL((compare_kernel<T, true>), dim3(grid), dim3(kBlock),
                       0, s, a, b, scalar, out, n, op);
  HIP_CHECK(hipGetLastError());
}

void launch_compare(DType dt, int op, const void* a, const void* b,
                    double scalar, void* out_u8, int64_t n, hipStream_t s) {
  if (dt == DType::F64)
    launch_compare_t(op, (const double*)a, (const double*)b, scalar,
                     (unsigned char*)out_u8, n, s);
  else
    launch_compare_t(op, (const float*)a, (const float*)b, scalar,
                     (unsigned char*)out_u8, n, s);
}

void launch_where(DType dt, const void* mask, const void* pa, double sa,
                  const void* pb, double sb, void* out, int64_t n,
                  hipStream_t s) {
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 4096);
  if (dt == DType::F64)
    hipLaunchKernelGGL((where_kernel<double>), dim3(grid), dim3(kBlock), 0, s,
                       (const unsigned char*)mask, (const double*)pa, sa,
                       (const double*)pb, sb, (double*)out, n);
  else
    hipLaunchKernelGGL((where_kernel<float>), dim3(grid), dim3(kBlock), 0, s,
                       (const unsigned char*)mask, (const float*)pa, sa,
                       (const float*)pb, sb, (float*)out, n);
  HIP_CHECK(hipGetLastError());
}

void launch_masked_fill(DType dt, void* data, const void* mask, double value,
                        int64_t n, hipStream_t s) {
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 4096);
  if (dt == DType::F64)
    hipLaunchKernelGGL((masked_fill_kernel<double>), dim3(grid), dim3(kBlock),
                       0, s, (double*)data, (const unsigned char*)mask, value,
                       n);
  else
    hipLaunchKernelGGL((masked_fill_kernel<float>), dim3(grid), dim3(kBlock),
                       0, s, (float*)data, (const unsigned char*)mask, value,
                       n);
  HIP_CHECK(hipGetLastError());
}

void launch_mask_count(const void* mask, void* scratch, void* out_i64,
                       int64_t n, hipStream_t s) {
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 2048);
  hipLaunchKernelGGL(mask_count_stage1, dim3(grid), dim3(kBlock), 0, s,
                     (const unsigned char*)mask, (int64_t*)scratch, n);
  hipLaunchKernelGGL(mask_count_stage2, dim3(1), dim3(64), 0, s,
                     (const int64_t*)scratch, (int64_t*)out_i64, grid);
  HIP_CHECK(hipGetLastError());
}

template <typename T>
static void launch_binary_bcast_t(BinOp op, int mode, const T* a, const T* b,
                                  T* out, int64_t outer, int64_t inner,
                                  hipStream_t s) {
  int64_t n = outer * inner;
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 4096);
  if (mode == 0)
    hipLaunchKernelGGL((binary_bcast_kernel<T, 0>), dim3(grid), dim3(kBlock),
                       0, s, a, b, out, outer, inner, (int)op);
  else
    hipLaunchKernelGGL((binary_bcast_kernel<T, 1>), dim3(grid), dim3(kBlock),
                       0, s, a, b, out, outer, inner, (int)op);
  HIP_CHECK(hipGetLastError());
}

void launch_binary_bcast(DType dt, BinOp op, int mode, const void* a,
                         const void* b, void* out, int64_t outer,
                         int64_t inner, hipStream_t s) {
  if (dt == DType::F64)
    launch_binary_bcast_t(op, mode, (const double*)a, (const double*)b,
                          (double*)out, outer, inner, s);
  else
    launch_binary_bcast_t(op, mode, (const float*)a, (const float*)b,
                          (float*)out, outer, inner, s);
}

template <typename T>
static void launch_argminmax_t(bool maxop, const T* in, void* scratch,
                               void* out_idx, int64_t n, hipStream_t s) {
  // scratch layout: [kBlock doubles][kBlock int64] partials
  int grid = (int)std::min<int64_t>((n + kBlock - 1) / kBlock, 2048);
  double* pvals = (double*)scratch;
  int64_t* pidx = (int64_t*)((char*)scratch + 2048 * sizeof(double));
  if (maxop) {
    hipLaunchKernelGGL((argminmax_stage1<T, true>), dim3(grid), dim3(kBlock),
                       0, s, in, pvals, pidx, n);
    hipLaunchKernelGGL((argminmax_stage2<true>), dim3(1), dim3(64), 0, s,
                       pvals, pidx, (int64_t*)out_idx, grid, n);
  } else {
    hipLaunchKernelGGL((argminmax_stage1<T, false>), dim3(grid), dim3(kBlock),
                       0, s, in, pvals, pidx, n);
    hipLaunchKernelGGL((argminmax_stage2<false>), dim3(1), dim3(64), 0, s,
                       pvals, pidx, (int64_t*)out_idx, grid, n);
  }
  HIP_CHECK(hipGetLastError());
}

void launch_argminmax(DType dt, bool maxop, const void* in, void* scratch,
                      void* out_idx, int64_t n, hipStream_t s) {
  if (dt == DType::F64)
    launch_argminmax_t(maxop, (const double*)in, scratch, out_idx, n, s);
  else
    launch_argminmax_t(maxop, (const float*)in, scratch, out_idx, n, s);
}

template <typename T>
static void launch_reduce_axis_t(ReduceOp mode, const T* in, T* out,
                                 int64_t outer, int64_t red, int64_t inner,
                                 hipStream_t s) {
  if (inner == 1) {
    // one wave per slice; >= 2048 workgroups saturates the 8 XCDs
    int waves_per_block = kBlock / 64;
    int grid = (int)std::min<int64_t>(
        (outer + waves_per_block - 1) / waves_per_block, 4096);
    switch ((int)mode) {
      case 0: hipLaunchKernelGGL((reduce_axis_last_kernel<T, 0>), dim3(grid), dim3(kBlock), 0, s, in, out, outer, red); break;
      case 1: hipLaunchKernelGGL((reduce_axis_last_kernel<T, 1>), dim3(grid), dim3(kBlock), 0, s, in, out, outer, red); break;
      case 2: hipLaunchKernelGGL((reduce_axis_last_kernel<T, 2>), dim3(grid), dim3(kBlock), 0, s, in, out, outer, red); break;
      default: hipLaunchKernelGGL((reduce_axis_last_kernel<T, 3>), dim3(grid), dim3(kBlock), 0, s, in, out, outer, red); break;
    }
  } else {
    int64_t total = outer * inner;
    int grid = (int)std::min<int64_t>((total + kBlock - 1) / kBlock, 4096);
    switch ((int)mode) {
      case 0: hipLaunchKernelGGL((reduce_axis_inner_kernel<T, 0>), dim3(grid), dim3(kBlock), 0, s, in, out, outer, red, inner); break;
      case 1: hipLaunchKernelGGL((reduce_axis_inner_kernel<T, 1>), dim3(grid), dim3(kBlock), 0, s, in, out, outer, red, inner); break;
      case 2: hipLaunchKernelGGL((reduce_axis_inner_kernel<T, 2>), dim3(grid), dim3(kBlock), 0, s, in, out, outer, red, inner); break;
      default: hipLaunchKernelGGL((reduce_axis_inner_kernel<T, 3>), dim3(grid), dim3(kBlock), 0, s, in, out, outer, red, inner); break;
    }
  }
  HIP_CHECK(hipGetLastError());
}

void launch_reduce_axis(DType dt, ReduceOp mode, const void* in, void* out,
                        int64_t outer, int64_t red, int64_t inner,
                        hipStream_t s) {
  if (dt == DType::F64)
    launch_reduce_axis_t(mode, (const double*)in, (double*)out, outer, red,
                         inner, s);
  else
    launch_reduce_axis_t(mode, (const float*)in, (float*)out, outer, red,
                         inner, s);
}

void launch_sum(DType dt, ReduceOp mode, const void* in, void* partials,
                void* out_scalar, int64_t n, hipStream_t stream) {
  if (dt == DType::F64)
    launch_sum_t(mode, (const double*)in, (double*)partials,
                 (double*)out_scalar, n, stream);
  else
    launch_sum_t(mode, (const float*)in, (float*)partials,
                 (float*)out_scalar, n, stream);
}

void launch_convert(DType src, DType dst, const void* in, void* out,
                    int64_t n, hipStream_t stream) {
  int grid = grid_for(n / 2 + 1);
  if (src == DType::F64 && dst == DType::F32)
    hipLaunchKernelGGL((convert_kernel<double, float>), dim3(grid),
                       dim3(kBlock), 0, stream, (const double*)in,
                       (float*)out, n);
  else if (src == DType::F32 && dst == DType::F64)
    hipLaunchKernelGGL((convert_kernel<float, double>), dim3(grid),
                       dim3(kBlock), 0, stream, (const float*)in,
                       (double*)out, n);
  else
    return;  // same-dtype convert is a no-op at the caller
  HIP_CHECK(hipGetLastError());
}

void launch_rand_normal(void* out, int64_t n, uint64_t seed, uint64_t offset,
                        double mu, double sigma, hipStream_t stream) {
  int grid = grid_for((n + 1) / 2);
  hipLaunchKernelGGL(randn_f64_kernel, dim3(grid), dim3(kBlock), 0, stream,
                     (double*)out, n, seed, offset, mu, sigma);
  HIP_CHECK(hipGetLastError());
}

void launch_rand_uniform(DType dt, void* out, int64_t n, uint64_t seed,
                         uint64_t offset, hipStream_t stream) {
  if (dt == DType::F64) {
    int grid = grid_for((n + 1) / 2);
    hipLaunchKernelGGL(rand_f64_kernel, dim3(grid), dim3(kBlock), 0, stream,
                       (double*)out, n, seed, offset);
  } else {
    int grid = grid_for((n + 3) / 4);
    hipLaunchKernelGGL(rand_f32_kernel, dim3(grid), dim3(kBlock), 0, stream,
                       (float*)out, n, seed, offset);
  }
  HIP_CHECK(hipGetLastError());
}

void launch_transpose(DType dt, const void* in, void* out, int64_t rows,
                      int64_t cols, hipStream_t stream) {
  int64_t tiles_r = (rows + 31) / 32;
  int64_t tiles_c = (cols + 31) / 32;
  if (dt == DType::F64)
    hipLaunchKernelGGL(transpose_kernel<double>, dim3(tiles_r * tiles_c),
                       dim3(256), 0, stream, (const double*)in, (double*)out,
                       rows, cols, tiles_c);
  else
    hipLaunchKernelGGL(transpose_kernel<float>, dim3(tiles_r * tiles_c),
                       dim3(256), 0, stream, (const float*)in, (float*)out,
                       rows, cols, tiles_c);
  HIP_CHECK(hipGetLastError());
}

void launch_cumsum_rows(DType dt, const void* in, void* out, int64_t rows,
                        int64_t cols, hipStream_t stream) {
  if (dt == DType::F64)
    hipLaunchKernelGGL(cumsum_rows_kernel<double>, dim3((unsigned)rows),
                       dim3(256), 0, stream, (const double*)in, (double*)out,
                       rows, cols);
  else
    hipLaunchKernelGGL(cumsum_rows_kernel<float>, dim3((unsigned)rows),
                       dim3(256), 0, stream, (const float*)in, (float*)out,
                       rows, cols);
  HIP_CHECK(hipGetLastError());
}

void launch_diff(DType dt, const void* in, void* out, int64_t outer,
                 int64_t inner, hipStream_t stream) {
  int64_t n_out = outer * (inner - 1);
  int grid = (int)std::min<int64_t>((n_out + 255) / 256, 4096);
  if (dt == DType::F64)
    hipLaunchKernelGGL(diff_kernel<double>, dim3(grid), dim3(256), 0, stream,
                       (const double*)in, (double*)out, outer, inner);
  else
    hipLaunchKernelGGL(diff_kernel<float>, dim3(grid), dim3(256), 0, stream,
                       (const float*)in, (float*)out, outer, inner);
  HIP_CHECK(hipGetLastError());
}

void launch_searchsorted(DType dt, const void* a, int64_t n, const void* v,
                         int64_t m, int right, void* out,
                         hipStream_t stream) {
  int grid = (int)std::min<int64_t>((m + 255) / 256, 4096);
  if (dt == DType::F64) {
    if (right)
      hipLaunchKernelGGL((searchsorted_kernel<double, true>), dim3(grid),
                         dim3(256), 0, stream, (const double*)a, n,
                         (const double*)v, m, (long long*)out);
    else
      hipLaunchKernelGGL((searchsorted_kernel<double, false>), dim3(grid),
                         dim3(256), 0, stream, (const double*)a, n,
                         (const double*)v, m, (long long*)out);
  } else {
    if (right)
      hipLaunchKernelGGL((searchsorted_kernel<float, true>), dim3(grid),
                         dim3(256), 0, stream, (const float*)a, n,
                         (const float*)v, m, (long long*)out);
    else
      hipLaunchKernelGGL((searchsorted_kernel<float, false>), dim3(grid),
                         dim3(256), 0, stream, (const float*)a, n,
                         (const float*)v, m, (long long*)out);
  }
  HIP_CHECK(hipGetLastError());
}
