// Shared declarations for the gfx950 HIP kernel library.
// All kernels are written CDNA4-first: wave64, MFMA matrix cores, LDS
// tiling sized for 160 KiB/CU, grid sizing for 256 CUs / 8 XCDs.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <stdexcept>
#include <string>

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      throw std::runtime_error(std::string("HIP error at " __FILE__ ":") +  \
                               std::to_string(__LINE__) + ": " +            \
                               hipGetErrorString(_e));                      \
    }                                                                       \
  } while (0)

// element dtypes understood by the generic elementwise/reduce entry points
enum class DType : int { F32 = 0, F64 = 1 };

enum class UnaryOp : int {
  Square = 0, Neg = 1, Abs = 2, Sqrt = 3, Exp = 4,
  Log = 5, Sin = 6, Cos = 7, Tanh = 8,
  Floor = 9, Ceil = 10, Rint = 11, Trunc = 12, Sign = 13,
  Log2 = 14, Log10 = 15, Exp2 = 16, Expm1 = 17, Log1p = 18,
  Cbrt = 19, Tan = 20, Arcsin = 21, Arccos = 22, Arctan = 23,
  Sinh = 24, Cosh = 25,
};
enum class BinOp : int {
  Add = 0, Sub = 1, Mul = 2, Div = 3, Max = 4, Min = 5, Pow = 6,
};
// full-array reduction flavors (the python-side `mode` int)
enum class ReduceOp : int { Sum = 0, SumSquares = 1, Max = 2, Min = 3 };

// launchers (defined in kernels_ew.hip / gemm_*.hip); all take raw device
// pointers and run on `stream`
void launch_unary(DType dt, UnaryOp op, const void* in, void* out,
                  int64_t n, hipStream_t stream);
void launch_binary(DType dt, BinOp op, const void* a, const void* b, void* out,
                   int64_t n, hipStream_t stream);
void launch_binary_scalar(DType dt, BinOp op, const void* a, double scalar,
                          void* out, int64_t n, hipStream_t stream);
// reduction: out_partials must hold >= reduce_num_partials(n) elements of dt;
// final scalar (in dt) is written to out_scalar (device ptr) by stage 2
int reduce_num_partials(int64_t n);
void launch_sum(DType dt, ReduceOp mode, const void* in, void* partials,
                void* out_scalar, int64_t n, hipStream_t stream);
// single-axis reduction of a contiguous array viewed [outer][red][inner]
void launch_reduce_axis(DType dt, ReduceOp mode, const void* in, void* out,
                        int64_t outer, int64_t red, int64_t inner,
                        hipStream_t stream);
// radix sort (np.sort/argsort): see sort.hip header for scratch sizes
int64_t radix_sort_nchunks(int64_t n);
void launch_searchsorted(DType dt, const void* a, int64_t n, const void* v,
                         int64_t m, int right, void* out, hipStream_t stream);
void launch_diff(DType dt, const void* in, void* out, int64_t outer,
                 int64_t inner, hipStream_t stream);
void launch_cumsum_rows(DType dt, const void* in, void* out, int64_t rows,
                        int64_t cols, hipStream_t stream);
void launch_transpose(DType dt, const void* in, void* out, int64_t rows,
                      int64_t cols, hipStream_t stream);
void launch_radix_sort_rows(DType dt, const void* in, void* out,
                            void* idx_out, void* keys_a, void* keys_b,
                            void* idx_a, void* idx_b, void* counts,
                            void* dig_scratch, int64_t rows, int64_t cols,
                            bool want_idx, hipStream_t s);
void launch_radix_sort(DType dt, const void* in, void* out, void* idx_out,
                       void* keys_a, void* keys_b, void* idx_a, void* idx_b,
                       void* counts, void* dig_scratch, int64_t n,
                       bool want_idx, hipStream_t stream);
// cumulative sum (flat): totals scratch >= 2048 doubles
void launch_cumsum(DType dt, const void* in, void* out, void* totals,
                   int64_t n, hipStream_t stream);
// u8 mask logic: 0 and, 1 or, 2 xor, 3 andnot, 4 not (b may be null)
void launch_mask_logic(const void* a, const void* b, void* out, int64_t n,
                       int op, hipStream_t stream);
// histogram selection (median/percentile): range histogram (counts:
// bins u64 + extra[3] = {nan, below, above}) and range compaction
void launch_hist_range(DType dt, const void* in, int64_t n, double lo,
                       double hi, double inv_width, int bins, int exact,
                       void* counts, void* extra, hipStream_t s);
void launch_extract_range(DType dt, const void* in, int64_t n, double lo,
                          double hi, void* out, void* counter, int64_t cap,
                          hipStream_t stream);
// boolean masks: compare -> u8, np.where select, masked fill, popcount
void launch_compare(DType dt, int op, const void* a, const void* b,
                    double scalar, void* out_u8, int64_t n, hipStream_t s);
void launch_where(DType dt, const void* mask, const void* pa, double sa,
                  const void* pb, double sb, void* out, int64_t n,
                  hipStream_t s);
void launch_masked_fill(DType dt, void* data, const void* mask, double value,
                        int64_t n, hipStream_t s);
void launch_mask_count(const void* mask, void* scratch, void* out_i64,
                       int64_t n, hipStream_t s);
// broadcast binary over [outer][inner]: mode 0 = b[inner] along outer,
// mode 1 = b[outer] along inner
void launch_binary_bcast(DType dt, BinOp op, int mode, const void* a,
                         const void* b, void* out, int64_t outer,
                         int64_t inner, hipStream_t stream);
// argmax/argmin (numpy tie/NaN semantics): scratch >= 2048*(8+8) bytes,
// result int64 index written to out_idx (device)
void launch_argminmax(DType dt, bool maxop, const void* in, void* scratch,
                      void* out_idx, int64_t n, hipStream_t stream);
// philox4x32-10 uniform doubles/floats in [0, 1)
void launch_rand_uniform(DType dt, void* out, int64_t n, uint64_t seed,
                         uint64_t offset, hipStream_t stream);
// dtype conversion (f64 <-> f32), vectorized
void launch_convert(DType src, DType dst, const void* in, void* out,
                    int64_t n, hipStream_t stream);
// normal(mu, sigma) via Philox + Box-Muller (f64 only: numpy's normal
// family returns float64)
void launch_rand_normal(void* out, int64_t n, uint64_t seed, uint64_t offset,
                        double mu, double sigma, hipStream_t stream);

// row-major GEMM: C[M,N] = A[M,K] @ B[K,N]
void launch_gemm_f32(const float* a, const float* b, float* c, int m, int n,
                     int k, hipStream_t stream);
void launch_gemm_f64(const double* a, const double* b, double* c, int m, int n,
                     int k, hipStream_t stream);
// bf16 in (as uint16 storage), f32 accumulate, bf16 out
void launch_gemm_bf16(const uint16_t* a, const uint16_t* b, uint16_t* c, int m,
                      int n, int k, hipStream_t stream);
// fast 256^2-tile 8-phase path (takes B pre-transposed [N][K])
bool gemm_bf16_256_supported(int m, int n, int k);
void launch_gemm_bf16_256(const uint16_t* a, const uint16_t* bt, uint16_t* c,
                          int m, int n, int k, hipStream_t stream);
// variant T: same tile structure, B consumed in native [K][N] via
// ds_read_b64_tr_b16 (no pre-transpose pass, no Bt scratch)
void launch_gemm_bf16_256t(const uint16_t* a, const uint16_t* b, uint16_t* c,
                           int m, int n, int k, hipStream_t stream);
void launch_transpose_bf16(const uint16_t* in, uint16_t* out, int k, int n,
                           hipStream_t stream);
