// _hipops: CPython extension exposing the gfx950 kernel library to the
// sandbox runtime -- no torch dependency, so importing it costs
// milliseconds (torch costs ~1.5 s per sandbox) and the module can be
// pre-initialized in the warm child.
//
// Host-side design:
//  - lazy init: importing never touches HIP (the module must be loadable
//    on CPU-only boxes); init() creates the device context, a COMPUTE
//    stream and a COPY stream;
//  - pinned staging: two hipHostMalloc buffers; uploads/downloads stream
//    through them with hipMemcpyAsync on the copy stream, overlapping the
//    host memcpy of chunk i+1 with the DMA of chunk i (the BASELINE
//    "pinned hipHostMalloc + hipMemcpyAsync on a side stream" obligation);
//  - a size-bucketed free-list over hipMalloc so repeated same-shape
//    allocations in one sandbox are O(us);
//  - all entry points release the GIL around device work.

#define PY_SSIZE_T_CLEAN
#include <Python.h>

#include <atomic>
#include <cstring>
#include <mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <unordered_set>
#include <vector>

#include "common.h"

namespace {

struct DevBuf {
  void* ptr = nullptr;
  int64_t size = 0;
};

struct State {
  bool initialized = false;
  bool use_mempool = false;
  int device = 0;
  hipStream_t compute = nullptr;
  hipStream_t copy = nullptr;
  // pinned staging (double buffer)
  static constexpr int64_t kStage = 32ll << 20;
  void* pin[2] = {nullptr, nullptr};
  hipEvent_t pin_evt[2] = {nullptr, nullptr};
  // reduction scratch + pinned scalar
  void* reduce_scratch = nullptr;
  void* scalar_dev = nullptr;
  double* scalar_pin = nullptr;
  // handles
  uint64_t next_handle = 1;
  std::unordered_map<uint64_t, DevBuf> bufs;
  // free-list: size -> free device pointers
  std::unordered_map<int64_t, std::vector<void*>> free_list;
  // pointers that came from plain hipMalloc (mempool fallback): they must
  // be released with hipFree, never hipFreeAsync
  std::unordered_set<void*> plain_allocs;
  int64_t pool_bytes = 0;
  std::atomic<uint64_t> rand_offset{0};  // claimed with fetch_add: two
                                         // concurrent rand() calls must
                                         // never share counter ranges
  // allocator stats
  int64_t outstanding = 0;
  long mempool_allocs = 0;
  long fallback_allocs = 0;
  long oom_trims = 0;
  // The daemon serves one thread per sandbox connection and every entry
  // point releases the GIL around device work, so shared host-side
  // state needs real locks (the GIL only guards the Python-visible maps,
  // which are touched outside the allow-threads regions):
  std::mutex alloc_mu;   // free_list / plain_allocs / pool counters
  std::mutex stage_mu;   // the two pinned staging buffers + their events
  std::mutex reduce_mu;  // reduce scratch + the scalar landing slot
};

State g;

int64_t round_size(int64_t n) { return (n + 255) & ~255ll; }

// release a cached pointer with the API that matches its provenance
void release_ptr(void* p) {
  if (g.plain_allocs.erase(p)) {
    (void)hipFree(p);
    return;
  }
  if (!g.use_mempool || hipFreeAsync(p, g.compute) != hipSuccess) (void)hipFree(p);
}

// under memory pressure our exact-size cache is the first thing to give
// back: hipMemPoolTrimTo cannot reclaim blocks WE are holding.
// Caller holds alloc_mu.
void flush_free_list_locked() {
  for (auto& kv : g.free_list)
    for (void* q : kv.second) release_ptr(q);
  g.free_list.clear();
  g.pool_bytes = 0;
  (void)hipStreamSynchronize(g.compute);
}

void ensure_init() {
  if (!g.initialized) throw std::runtime_error("_hipops not initialized; call init()");
}

constexpr int64_t kFreeListCapBytes = 96ll << 30;  // per-daemon cache cap

void* pool_alloc(int64_t size) {
  size = round_size(size);
  std::lock_guard<std::mutex> lk(g.alloc_mu);
  // exact-size reuse first: steady-state request streams allocate the
  // same shapes over and over; a hit costs no HIP call at all (observed:
  // sustained alloc/free churn through the async mempool causes ~300 ms
  // runtime-level stalls that block every other HIP call)
  {
    auto it = g.free_list.find(size);
    if (it != g.free_list.end() && !it->second.empty()) {
      void* p = it->second.back();
      it->second.pop_back();
      g.pool_bytes -= size;
      g.outstanding += size;
      return p;
    }
  }
  if (g.use_mempool) {
    // stream-ordered allocator with an unbounded release threshold: the
    // driver caches freed blocks, so the steady-state cost of the hot
    // path's repeated 800 MB allocations is O(us), not a fresh hipMalloc
    // (which serializes in the driver at ~10/s for GB-scale blocks)
    void* p = nullptr;
    hipError_t e = hipMallocAsync(&p, size, g.compute);
    if (e == hipErrorOutOfMemory) {
      g.oom_trims++;
      flush_free_list_locked();
      hipMemPool_t pool = nullptr;
      if (hipDeviceGetDefaultMemPool(&pool, g.device) == hipSuccess)
        (void)hipMemPoolTrimTo(pool, 0);
      e = hipMallocAsync(&p, size, g.compute);
    }
    if (e == hipSuccess) {
      g.mempool_allocs++;
      g.outstanding += size;
      return p;
    }
    g.fallback_allocs++;
    // fall through to plain hipMalloc on persistent failure
  }
  auto it = g.free_list.find(size);
  if (it != g.free_list.end() && !it->second.empty()) {
    void* p = it->second.back();
    it->second.pop_back();
    return p;
  }
  void* p = nullptr;
  hipError_t e = hipMalloc(&p, size);
  if (e == hipErrorOutOfMemory) {
    // drop the cache (provenance-aware) and retry once
    flush_free_list_locked();
    e = hipMalloc(&p, size);
  }
  if (e != hipSuccess)
    throw std::runtime_error(std::string("hipMalloc failed: ") +
                             hipGetErrorString(e));
  g.plain_allocs.insert(p);
  g.outstanding += size;
  return p;
}

void pool_free(void* p, int64_t size) {
  size = round_size(size);
  std::lock_guard<std::mutex> lk(g.alloc_mu);
  g.outstanding -= size;
  if (g.pool_bytes + size <= kFreeListCapBytes) {
    g.free_list[size].push_back(p);
    g.pool_bytes += size;
    return;
  }
  release_ptr(p);
}

// memcpy into pinned staging is single-thread-bound at ~15 GB/s; split
// large chunks across a few threads (the GIL is already released here)
void parallel_memcpy(void* dst, const void* src, int64_t n) {
  constexpr int64_t kParallelCut = 4ll << 20;
  constexpr int kThreads = 4;
  if (n < kParallelCut) {
    memcpy(dst, src, (size_t)n);
    return;
  }
  int64_t piece = (n + kThreads - 1) / kThreads;
  std::thread workers[kThreads];
  for (int t = 0; t < kThreads; t++) {
    int64_t off = t * piece;
    int64_t len = std::min(piece, n - off);
    if (len <= 0) break;
    workers[t] = std::thread([=] {
      memcpy((char*)dst + off, (const char*)src + off, (size_t)len);
    });
  }
  for (int t = 0; t < kThreads; t++)
    if (workers[t].joinable()) workers[t].join();
}

void ensure_staging() {
  if (g.pin[0]) return;
  for (int i = 0; i < 2; i++) {
    HIP_CHECK(hipHostMalloc(&g.pin[i], State::kStage, hipHostMallocDefault));
    HIP_CHECK(hipEventCreateWithFlags(&g.pin_evt[i], hipEventDisableTiming));
  }
}

void ensure_reduce_scratch() {
  if (g.reduce_scratch) return;
  HIP_CHECK(hipMalloc(&g.reduce_scratch, 4096 * sizeof(double)));
}

uint64_t register_buf(void* p, int64_t size) {
  uint64_t h = g.next_handle++;
  g.bufs[h] = {p, size};
  return h;
}

DevBuf& get_buf(uint64_t h) {
  auto it = g.bufs.find(h);
  if (it == g.bufs.end()) throw std::runtime_error("invalid device handle");
  return it->second;
}

// bf16 256-tile variant: "t" (default; tr16 transpose-read, no B
// pre-transpose) or "b" (the r01 pre-transpose kernel, kept for A/B)
bool use_bf16_tr16() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("APP_BF16_256_VARIANT");
    v = (e && e[0] == 'b') ? 0 : 1;
  }
  return v == 1;
}

DType dtype_from_int(int dt) {
  if (dt == 0) return DType::F32;
  if (dt == 1) return DType::F64;
  throw std::runtime_error("unsupported dtype code");
}

// GIL-released device-work region with exception transport: a C++ throw
// inside Py_BEGIN/END_ALLOW_THREADS would skip the GIL reacquire and hit
// PyErr_SetString unlocked (undefined behavior); capture and rethrow
// after the GIL is back instead.
#define NOGIL_BEGIN           \
  std::string _nogil_err;     \
  Py_BEGIN_ALLOW_THREADS;     \
  try {
#define NOGIL_END                                               \
  }                                                             \
  catch (const std::exception& _e) { _nogil_err = _e.what(); }  \
  Py_END_ALLOW_THREADS;                                         \
  if (!_nogil_err.empty()) throw std::runtime_error(_nogil_err);

// C++ exceptions -> Python RuntimeError
#define WRAP_BEGIN try {
#define WRAP_END                                  \
  }                                               \
  catch (const std::exception& e) {               \
    PyErr_SetString(PyExc_RuntimeError, e.what()); \
    return nullptr;                               \
  }

// ---------------------------------------------------------------------------

PyObject* py_is_available(PyObject*, PyObject*) {
  int count = 0;
  hipError_t e = hipGetDeviceCount(&count);
  if (e != hipSuccess || count <= 0) Py_RETURN_FALSE;
  Py_RETURN_TRUE;
}

PyObject* py_device_count(PyObject*, PyObject*) {
  int count = 0;
  if (hipGetDeviceCount(&count) != hipSuccess) count = 0;
  return PyLong_FromLong(count);
}

PyObject* py_init(PyObject*, PyObject* args) {
  int device = 0;
  if (!PyArg_ParseTuple(args, "|i", &device)) return nullptr;
  WRAP_BEGIN
  if (g.initialized) Py_RETURN_NONE;
  NOGIL_BEGIN
  HIP_CHECK(hipSetDevice(device));
  HIP_CHECK(hipStreamCreateWithFlags(&g.compute, hipStreamNonBlocking));
  HIP_CHECK(hipStreamCreateWithFlags(&g.copy, hipStreamNonBlocking));
  // pinned staging + reduce scratch are allocated lazily on first use:
  // hipHostMalloc of 64 MB costs tens of ms and sandbox prewarm latency
  // bounds the service's sustained request throughput
  HIP_CHECK(hipMalloc(&g.scalar_dev, sizeof(double)));
  HIP_CHECK(hipHostMalloc((void**)&g.scalar_pin, sizeof(double),
                          hipHostMallocDefault));
  {
    hipMemPool_t pool = nullptr;
    if (hipDeviceGetDefaultMemPool(&pool, device) == hipSuccess &&
        pool != nullptr) {
      uint64_t threshold = UINT64_MAX;
      if (hipMemPoolSetAttribute(pool, hipMemPoolAttrReleaseThreshold,
                                 &threshold) == hipSuccess)
        g.use_mempool = true;
    }
  }
  NOGIL_END
  g.device = device;
  g.initialized = true;
  Py_RETURN_NONE;
  WRAP_END
}

PyObject* py_alloc(PyObject*, PyObject* args) {
  long long nbytes;
  if (!PyArg_ParseTuple(args, "L", &nbytes)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  void* p = nullptr;
  NOGIL_BEGIN
  p = pool_alloc(nbytes);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(p, nbytes));
  WRAP_END
}

PyObject* py_free(PyObject*, PyObject* args) {
  unsigned long long h;
  if (!PyArg_ParseTuple(args, "K", &h)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  DevBuf buf = get_buf(h);
  g.bufs.erase(h);
  pool_free(buf.ptr, buf.size);
  Py_RETURN_NONE;
  WRAP_END
}

// upload(host_buffer) -> handle
PyObject* py_upload(PyObject*, PyObject* args) {
  PyObject* obj;
  if (!PyArg_ParseTuple(args, "O", &obj)) return nullptr;
  Py_buffer view;
  if (PyObject_GetBuffer(obj, &view, PyBUF_C_CONTIGUOUS) != 0) return nullptr;
  WRAP_BEGIN
  ensure_init();
  int64_t nbytes = view.len;
  void* dev = nullptr;
  NOGIL_BEGIN
  std::lock_guard<std::mutex> stage_lk(g.stage_mu);
  ensure_staging();
  dev = pool_alloc(nbytes);
  // the allocation is stream-ordered on the compute stream; the copy
  // stream must not DMA into it before the alloc point is reached
  HIP_CHECK(hipEventRecord(g.pin_evt[0], g.compute));
  HIP_CHECK(hipStreamWaitEvent(g.copy, g.pin_evt[0], 0));
  const char* src = (const char*)view.buf;
  int64_t off = 0;
  int slot = 0;
  // double-buffered pinned staging: memcpy chunk i+1 overlaps DMA of i
  while (off < nbytes) {
    int64_t chunk = std::min(State::kStage, nbytes - off);
    HIP_CHECK(hipEventSynchronize(g.pin_evt[slot]));  // buffer free again?
    parallel_memcpy(g.pin[slot], src + off, chunk);
    HIP_CHECK(hipMemcpyAsync((char*)dev + off, g.pin[slot], chunk,
                             hipMemcpyHostToDevice, g.copy));
    HIP_CHECK(hipEventRecord(g.pin_evt[slot], g.copy));
    off += chunk;
    slot ^= 1;
  }
  // compute stream must not run ahead of the upload
  HIP_CHECK(hipEventSynchronize(g.pin_evt[0]));
  HIP_CHECK(hipEventSynchronize(g.pin_evt[1]));
  NOGIL_END
  PyBuffer_Release(&view);
  return PyLong_FromUnsignedLongLong(register_buf(dev, nbytes));
  WRAP_END
}

// download(handle, writable_host_buffer)
// download_slice(h, byte_offset, nbytes) -> bytes. Small-window reads
// (x[i], head/tail peeks) must not pull the whole buffer across PCIe.
PyObject* py_download_slice(PyObject*, PyObject* args) {
  unsigned long long h;
  long long off, nbytes;
  if (!PyArg_ParseTuple(args, "KLL", &h, &off, &nbytes)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  DevBuf& buf = get_buf(h);
  if (off < 0 || nbytes < 0 || off + nbytes > buf.size)
    throw std::runtime_error("download_slice out of range");
  PyObject* out = PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)nbytes);
  if (!out) throw std::bad_alloc();
  char* dst = PyBytes_AS_STRING(out);
  NOGIL_BEGIN
  std::lock_guard<std::mutex> stage_lk(g.stage_mu);
  HIP_CHECK(hipStreamSynchronize(g.compute));
  HIP_CHECK(hipMemcpy(dst, (char*)buf.ptr + off, (size_t)nbytes,
                      hipMemcpyDeviceToHost));
  NOGIL_END
  return out;
  WRAP_END
}

// download_strided(h, byte_offset, stride_bytes, elem_bytes, count)
// -> bytes. One pitched D2H copy: extracts a column of a row-major
// matrix (row-quantile picks after the 2-D sort) without pulling the
// whole buffer.
PyObject* py_download_strided(PyObject*, PyObject* args) {
  unsigned long long h;
  long long off, stride, esz, count;
  if (!PyArg_ParseTuple(args, "KLLLL", &h, &off, &stride, &esz, &count))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  DevBuf& buf = get_buf(h);
  if (off < 0 || stride < esz || esz <= 0 || count < 0 ||
      (count > 0 && off + (count - 1) * stride + esz > buf.size))
    throw std::runtime_error("download_strided out of range");
  PyObject* out = PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)(esz * count));
  if (!out) throw std::bad_alloc();
  char* dst = PyBytes_AS_STRING(out);
  NOGIL_BEGIN
  std::lock_guard<std::mutex> stage_lk(g.stage_mu);
  HIP_CHECK(hipStreamSynchronize(g.compute));
  HIP_CHECK(hipMemcpy2D(dst, (size_t)esz, (char*)buf.ptr + off,
                        (size_t)stride, (size_t)esz, (size_t)count,
                        hipMemcpyDeviceToHost));
  NOGIL_END
  return out;
  WRAP_END
}

PyObject* py_download(PyObject*, PyObject* args) {
  unsigned long long h;
  PyObject* obj;
  if (!PyArg_ParseTuple(args, "KO", &h, &obj)) return nullptr;
  Py_buffer view;
  if (PyObject_GetBuffer(obj, &view, PyBUF_C_CONTIGUOUS | PyBUF_WRITABLE) != 0)
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  DevBuf& buf = get_buf(h);
  if (view.len < buf.size) {
    PyBuffer_Release(&view);
    throw std::runtime_error("download target too small");
  }
  NOGIL_BEGIN
  std::lock_guard<std::mutex> stage_lk(g.stage_mu);
  ensure_staging();
  HIP_CHECK(hipStreamSynchronize(g.compute));
  char* dst = (char*)view.buf;
  int64_t off = 0;
  int slot = 0;
  int64_t pending_off[2] = {-1, -1};
  int64_t pending_len[2] = {0, 0};
  while (off < buf.size) {
    int64_t chunk = std::min(State::kStage, buf.size - off);
    if (pending_off[slot] >= 0) {
      HIP_CHECK(hipEventSynchronize(g.pin_evt[slot]));
      parallel_memcpy(dst + pending_off[slot], g.pin[slot], pending_len[slot]);
    }
    HIP_CHECK(hipMemcpyAsync(g.pin[slot], (char*)buf.ptr + off, chunk,
                             hipMemcpyDeviceToHost, g.copy));
    HIP_CHECK(hipEventRecord(g.pin_evt[slot], g.copy));
    pending_off[slot] = off;
    pending_len[slot] = chunk;
    off += chunk;
    slot ^= 1;
  }
  for (int i = 0; i < 2; i++) {
    int s = slot ^ i ^ 1;  // drain in issue order
    if (pending_off[s] >= 0) {
      HIP_CHECK(hipEventSynchronize(g.pin_evt[s]));
      parallel_memcpy(dst + pending_off[s], g.pin[s], pending_len[s]);
    }
  }
  NOGIL_END
  PyBuffer_Release(&view);
  Py_RETURN_NONE;
  WRAP_END
}

PyObject* py_rand(PyObject*, PyObject* args) {
  long long n;
  int dt;
  unsigned long long seed;
  if (!PyArg_ParseTuple(args, "LiK", &n, &dt, &seed)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType dtype = dtype_from_int(dt);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  void* dev = nullptr;
  uint64_t off = g.rand_offset.fetch_add((uint64_t)n);  // exclusive range
  NOGIL_BEGIN
  dev = pool_alloc(n * esize);
  launch_rand_uniform(dtype, dev, n, seed, off, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(dev, n * esize));
  WRAP_END
}

// randn(n, seed, mu, sigma) -> handle (float64 N(mu, sigma^2))
PyObject* py_randn(PyObject*, PyObject* args) {
  long long n;
  unsigned long long seed;
  double mu, sigma;
  if (!PyArg_ParseTuple(args, "LKdd", &n, &seed, &mu, &sigma)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  void* dev = nullptr;
  uint64_t off = g.rand_offset.fetch_add((uint64_t)n);
  NOGIL_BEGIN
  dev = pool_alloc(n * 8);
  launch_rand_normal(dev, n, seed, off, mu, sigma, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(dev, n * 8));
  WRAP_END
}

// convert(h, src_dt, dst_dt, n) -> handle (f64 <-> f32 cast)
PyObject* py_convert(PyObject*, PyObject* args) {
  unsigned long long h;
  int src, dst;
  long long n;
  if (!PyArg_ParseTuple(args, "KiiL", &h, &src, &dst, &n)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType sdt = dtype_from_int(src), ddt = dtype_from_int(dst);
  DevBuf& in = get_buf(h);
  int64_t out_size = n * (ddt == DType::F64 ? 8 : 4);
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(out_size);
  launch_convert(sdt, ddt, in.ptr, out, n, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, out_size));
  WRAP_END
}

PyObject* py_unary(PyObject*, PyObject* args) {
  unsigned long long h;
  int op, dt;
  long long n;
  if (!PyArg_ParseTuple(args, "KiiL", &h, &op, &dt, &n)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(in.size);
  launch_unary(dtype, (UnaryOp)op, in.ptr, out, n, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, in.size));
  WRAP_END
}

PyObject* py_binary(PyObject*, PyObject* args) {
  unsigned long long ha, hb;
  int op, dt;
  long long n;
  if (!PyArg_ParseTuple(args, "KKiiL", &ha, &hb, &op, &dt, &n)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType dtype = dtype_from_int(dt);
  DevBuf& a = get_buf(ha);
  DevBuf& b = get_buf(hb);
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(a.size);
  launch_binary(dtype, (BinOp)op, a.ptr, b.ptr, out, n, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, a.size));
  WRAP_END
}

PyObject* py_binary_scalar(PyObject*, PyObject* args) {
  unsigned long long ha;
  double scalar;
  int op, dt;
  long long n;
  if (!PyArg_ParseTuple(args, "KdiiL", &ha, &scalar, &op, &dt, &n))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType dtype = dtype_from_int(dt);
  DevBuf& a = get_buf(ha);
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(a.size);
  launch_binary_scalar(dtype, (BinOp)op, a.ptr, scalar, out, n, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, a.size));
  WRAP_END
}

// sum(handle, dtype, n, mode:int) -> float
// mode: 0 = sum, 1 = sum of squares, 2 = max, 3 = min (ReduceOp)
PyObject* py_sum(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt, mode;
  long long n;
  if (!PyArg_ParseTuple(args, "KiLi", &h, &dt, &n, &mode)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (mode < 0 || mode > 3) throw std::runtime_error("bad reduce mode");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  double result = 0;
  NOGIL_BEGIN
  std::lock_guard<std::mutex> reduce_lk(g.reduce_mu);
  ensure_reduce_scratch();
  launch_sum(dtype, (ReduceOp)mode, in.ptr, g.reduce_scratch, g.scalar_dev, n,
             g.compute);
  HIP_CHECK(hipMemcpyAsync(g.scalar_pin, g.scalar_dev, 8,
                           hipMemcpyDeviceToHost, g.compute));
  HIP_CHECK(hipStreamSynchronize(g.compute));
  if (dtype == DType::F64)
    result = *g.scalar_pin;
  else
    result = (double)*(float*)g.scalar_pin;
  NOGIL_END
  return PyFloat_FromDouble(result);
  WRAP_END
}

// sort(h, dtype, n, want_idx) -> handle | (handle, idx_handle)
PyObject* py_sort(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt, want_idx;
  long long n;
  if (!PyArg_ParseTuple(args, "KiLi", &h, &dt, &n, &want_idx))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (n < 1 || n > (1ll << 31)) throw std::runtime_error("bad sort length");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  int64_t ksize = esize;  // key width == element width
  int64_t nchunks = radix_sort_nchunks(n);
  int64_t nseg = (nchunks + 255) / 256;
  void* out = nullptr;
  void* idx_res = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(n * esize);
  void* keys_a = pool_alloc(n * ksize);
  void* keys_b = pool_alloc(n * ksize);
  void* counts = pool_alloc(nchunks * 256 * 4);
  void* dig = pool_alloc(256 * 8 + nseg * 256 * 4);
  void* idx_a = nullptr;
  void* idx_b = nullptr;
  if (want_idx) {
    idx_a = pool_alloc(n * 8);
    idx_b = pool_alloc(n * 8);
  }
  launch_radix_sort(dtype, in.ptr, out, idx_a, keys_a, keys_b, idx_a, idx_b,
                    counts, dig, n, want_idx != 0, g.compute);
  pool_free(keys_a, n * ksize);
  pool_free(keys_b, n * ksize);
  pool_free(counts, nchunks * 256 * 4);
  pool_free(dig, 256 * 8 + nseg * 256 * 4);
  if (want_idx) {
    idx_res = idx_a;  // result indices live in idx_a after even passes
    pool_free(idx_b, n * 8);
  }
  NOGIL_END
  unsigned long long hout = register_buf(out, n * esize);
  if (!want_idx) return PyLong_FromUnsignedLongLong(hout);
  unsigned long long hidx = register_buf(idx_res, n * 8);
  return Py_BuildValue("(KK)", hout, hidx);
  WRAP_END
}

// sort2d(h, dtype, rows, cols, want_idx): independent stable sort of
// every row (np.sort/np.argsort axis=-1 on 2-D). Returns handle |
// (handle, int64 per-row-position handle).
PyObject* py_sort2d(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt, want_idx;
  long long rows, cols;
  if (!PyArg_ParseTuple(args, "KiLLi", &h, &dt, &rows, &cols, &want_idx))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (rows < 1 || cols < 1 || rows * cols > (1ll << 31))
    throw std::runtime_error("bad sort2d shape");
  const int64_t n = rows * cols;
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  int64_t nchunks = radix_sort_nchunks(n);
  int64_t nseg = (nchunks + 255) / 256;
  void* out = nullptr;
  void* idx_res = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(n * esize);
  void* keys_a = pool_alloc(n * esize);
  void* keys_b = pool_alloc(n * esize);
  void* idx_a = pool_alloc(n * 8);
  void* idx_b = pool_alloc(n * 8);
  void* counts = pool_alloc(nchunks * 256 * 4);
  void* dig = pool_alloc(256 * 8 + nseg * 256 * 4);
  if (want_idx) idx_res = pool_alloc(n * 8);
  launch_radix_sort_rows(dtype, in.ptr, out, idx_res, keys_a, keys_b, idx_a,
                         idx_b, counts, dig, rows, cols, want_idx != 0,
                         g.compute);
  pool_free(keys_a, n * esize);
  pool_free(keys_b, n * esize);
  pool_free(idx_a, n * 8);
  pool_free(idx_b, n * 8);
  pool_free(counts, nchunks * 256 * 4);
  pool_free(dig, 256 * 8 + nseg * 256 * 4);
  NOGIL_END
  unsigned long long hout = register_buf(out, n * esize);
  if (!want_idx) return PyLong_FromUnsignedLongLong(hout);
  unsigned long long hidx = register_buf(idx_res, n * 8);
  return Py_BuildValue("(KK)", hout, hidx);
  WRAP_END
}

// searchsorted(ha, n, hv, m, dtype, right) -> int64 handle (m indices)
PyObject* py_searchsorted(PyObject*, PyObject* args) {
  unsigned long long ha, hv;
  int dt, right;
  long long n, m;
  if (!PyArg_ParseTuple(args, "KLKLii", &ha, &n, &hv, &m, &dt, &right))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (n < 0 || m < 1) throw std::runtime_error("bad searchsorted sizes");
  DType dtype = dtype_from_int(dt);
  DevBuf& a = get_buf(ha);
  DevBuf& v = get_buf(hv);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  if (n * esize > a.size || m * esize > v.size)
    throw std::runtime_error("searchsorted oob");
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(m * 8);
  launch_searchsorted(dtype, a.ptr, n, v.ptr, m, right, out, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, m * 8));
  WRAP_END
}

// diff(h, dtype, outer, inner) -> handle (outer x (inner-1))
PyObject* py_diff(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt;
  long long outer, inner;
  if (!PyArg_ParseTuple(args, "KiLL", &h, &dt, &outer, &inner)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (outer < 1 || inner < 2) throw std::runtime_error("bad diff shape");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  if (outer * inner * esize > in.size) throw std::runtime_error("diff oob");
  int64_t n_out = outer * (inner - 1);
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(n_out * esize);
  launch_diff(dtype, in.ptr, out, outer, inner, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, n_out * esize));
  WRAP_END
}

// cumsum2d(h, dtype, rows, cols) -> handle (row-wise inclusive scan)
PyObject* py_cumsum2d(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt;
  long long rows, cols;
  if (!PyArg_ParseTuple(args, "KiLL", &h, &dt, &rows, &cols)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (rows < 1 || cols < 1) throw std::runtime_error("bad cumsum2d shape");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  int64_t n = rows * cols;
  if (n * esize > in.size) throw std::runtime_error("cumsum2d oob");
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(n * esize);
  launch_cumsum_rows(dtype, in.ptr, out, rows, cols, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, n * esize));
  WRAP_END
}

// copy_d2d(dst, dst_off, src, src_off, nbytes): device-to-device copy
// window (np.concatenate/stack building blocks)
PyObject* py_copy_d2d(PyObject*, PyObject* args) {
  unsigned long long hd, hs;
  long long doff, soff, nbytes;
  if (!PyArg_ParseTuple(args, "KLKLL", &hd, &doff, &hs, &soff, &nbytes))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  DevBuf& dst = get_buf(hd);
  DevBuf& src = get_buf(hs);
  if (doff < 0 || soff < 0 || nbytes < 0 || doff + nbytes > dst.size ||
      soff + nbytes > src.size)
    throw std::runtime_error("copy_d2d out of range");
  NOGIL_BEGIN
  HIP_CHECK(hipMemcpyAsync((char*)dst.ptr + doff, (char*)src.ptr + soff,
                           (size_t)nbytes, hipMemcpyDeviceToDevice,
                           g.compute));
  NOGIL_END
  Py_RETURN_NONE;
  WRAP_END
}

// transpose(h, dtype, rows, cols) -> handle ([cols][rows] result)
PyObject* py_transpose(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt;
  long long rows, cols;
  if (!PyArg_ParseTuple(args, "KiLL", &h, &dt, &rows, &cols)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (rows < 1 || cols < 1) throw std::runtime_error("bad transpose shape");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  int64_t n = rows * cols;
  if (n * esize > in.size) throw std::runtime_error("transpose oob");
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(n * esize);
  launch_transpose(dtype, in.ptr, out, rows, cols, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, n * esize));
  WRAP_END
}

// cumsum(h, dtype, n) -> handle (same dtype)
PyObject* py_cumsum(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt;
  long long n;
  if (!PyArg_ParseTuple(args, "KiL", &h, &dt, &n)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  void* out = nullptr;
  NOGIL_BEGIN
  std::lock_guard<std::mutex> reduce_lk(g.reduce_mu);
  ensure_reduce_scratch();
  out = pool_alloc(n * esize);
  launch_cumsum(dtype, in.ptr, out, g.reduce_scratch, n, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, n * esize));
  WRAP_END
}

// mask_logic(ha, hb_or_0, n, op) -> u8 mask handle
PyObject* py_mask_logic(PyObject*, PyObject* args) {
  unsigned long long ha, hb;
  long long n;
  int op;
  if (!PyArg_ParseTuple(args, "KKLi", &ha, &hb, &n, &op)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (op < 0 || op > 4) throw std::runtime_error("bad mask-logic op");
  DevBuf& a = get_buf(ha);
  void* bptr = hb ? get_buf(hb).ptr : nullptr;
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(n);
  launch_mask_logic(a.ptr, bptr, out, n, op, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, n));
  WRAP_END
}

// histogram(h, dtype, n, lo, hi, bins) -> bytes((bins+3) * u64):
// [bin counts..., nan_count, below_count, above_count]
PyObject* py_histogram(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt, bins, exact = 0;
  long long n;
  double lo, hi;
  if (!PyArg_ParseTuple(args, "KiLddi|i", &h, &dt, &n, &lo, &hi, &bins,
                        &exact))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (bins < 1 || bins > 4096) throw std::runtime_error("bad bin count");
  if (!(hi > lo)) throw std::runtime_error("bad histogram range");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  std::vector<unsigned long long> host((size_t)bins + 3, 0);
  double inv_width = (double)bins / (hi - lo);
  NOGIL_BEGIN
  void* counts = pool_alloc((int64_t)(bins + 3) * 8);
  HIP_CHECK(hipMemsetAsync(counts, 0, (int64_t)(bins + 3) * 8, g.compute));
  void* extra = (char*)counts + (int64_t)bins * 8;
  launch_hist_range(dtype, in.ptr, n, lo, hi, inv_width, bins, exact,
                    counts, extra, g.compute);
  HIP_CHECK(hipMemcpyAsync(host.data(), counts, (int64_t)(bins + 3) * 8,
                           hipMemcpyDeviceToHost, g.compute));
  HIP_CHECK(hipStreamSynchronize(g.compute));
  pool_free(counts, (int64_t)(bins + 3) * 8);
  NOGIL_END
  return PyBytes_FromStringAndSize((const char*)host.data(),
                                   (Py_ssize_t)((bins + 3) * 8));
  WRAP_END
}

// extract_range(h, dtype, n, lo, hi, cap) -> (count, bytes(f64 values))
PyObject* py_extract_range(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt;
  long long n, cap;
  double lo, hi;
  if (!PyArg_ParseTuple(args, "KiLddL", &h, &dt, &n, &lo, &hi, &cap))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (cap < 1 || cap > (1 << 22)) throw std::runtime_error("bad cap");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  unsigned long long found = 0;
  std::vector<double> host;
  NOGIL_BEGIN
  void* out = pool_alloc(cap * 8 + 8);
  void* counter = (char*)out + cap * 8;
  HIP_CHECK(hipMemsetAsync(counter, 0, 8, g.compute));
  launch_extract_range(dtype, in.ptr, n, lo, hi, out, counter, cap,
                       g.compute);
  HIP_CHECK(hipMemcpyAsync(&found, counter, 8, hipMemcpyDeviceToHost,
                           g.compute));
  HIP_CHECK(hipStreamSynchronize(g.compute));
  unsigned long long take = found < (unsigned long long)cap
                                ? found
                                : (unsigned long long)cap;
  host.resize((size_t)take);
  if (take) {
    HIP_CHECK(hipMemcpyAsync(host.data(), out, (int64_t)take * 8,
                             hipMemcpyDeviceToHost, g.compute));
    HIP_CHECK(hipStreamSynchronize(g.compute));
  }
  pool_free(out, cap * 8 + 8);
  NOGIL_END
  PyObject* bytes = PyBytes_FromStringAndSize(
      (const char*)host.data(), (Py_ssize_t)(host.size() * 8));
  if (!bytes) return nullptr;
  PyObject* tuple = Py_BuildValue("(KN)", found, bytes);
  return tuple;
  WRAP_END
}

// compare(h, dtype, n, cmp_op, hb_or_0, scalar) -> u8 mask handle
PyObject* py_compare(PyObject*, PyObject* args) {
  unsigned long long h, hb;
  int dt, op;
  long long n;
  double scalar;
  if (!PyArg_ParseTuple(args, "KiLiKd", &h, &dt, &n, &op, &hb, &scalar))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (op < 0 || op > 5) throw std::runtime_error("bad compare op");
  DType dtype = dtype_from_int(dt);
  DevBuf& a = get_buf(h);
  void* bptr = hb ? get_buf(hb).ptr : nullptr;
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(n);
  launch_compare(dtype, op, a.ptr, bptr, scalar, out, n, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, n));
  WRAP_END
}

// where(hmask, dtype, n, ha_or_0, sa, hb_or_0, sb) -> handle
PyObject* py_where(PyObject*, PyObject* args) {
  unsigned long long hm, ha, hb;
  int dt;
  long long n;
  double sa, sb;
  if (!PyArg_ParseTuple(args, "KiLKdKd", &hm, &dt, &n, &ha, &sa, &hb, &sb))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType dtype = dtype_from_int(dt);
  DevBuf& m = get_buf(hm);
  void* pa = ha ? get_buf(ha).ptr : nullptr;
  void* pb = hb ? get_buf(hb).ptr : nullptr;
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(n * esize);
  launch_where(dtype, m.ptr, pa, sa, pb, sb, out, n, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, n * esize));
  WRAP_END
}

// masked_fill(h, hmask, dtype, n, value): in-place on h's buffer
PyObject* py_masked_fill(PyObject*, PyObject* args) {
  unsigned long long h, hm;
  int dt;
  long long n;
  double value;
  if (!PyArg_ParseTuple(args, "KKiLd", &h, &hm, &dt, &n, &value))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType dtype = dtype_from_int(dt);
  DevBuf& a = get_buf(h);
  DevBuf& m = get_buf(hm);
  NOGIL_BEGIN
  launch_masked_fill(dtype, a.ptr, m.ptr, value, n, g.compute);
  NOGIL_END
  Py_RETURN_NONE;
  WRAP_END
}

// mask_count(hmask, n) -> int64 popcount
PyObject* py_mask_count(PyObject*, PyObject* args) {
  unsigned long long hm;
  long long n;
  if (!PyArg_ParseTuple(args, "KL", &hm, &n)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  DevBuf& m = get_buf(hm);
  long long result = 0;
  NOGIL_BEGIN
  std::lock_guard<std::mutex> reduce_lk(g.reduce_mu);
  ensure_reduce_scratch();
  launch_mask_count(m.ptr, g.reduce_scratch, g.scalar_dev, n, g.compute);
  HIP_CHECK(hipMemcpyAsync(g.scalar_pin, g.scalar_dev, 8,
                           hipMemcpyDeviceToHost, g.compute));
  HIP_CHECK(hipStreamSynchronize(g.compute));
  result = *(int64_t*)g.scalar_pin;
  NOGIL_END
  return PyLong_FromLongLong(result);
  WRAP_END
}

// binary_bcast(ha, hb, bop, dtype, outer, inner, mode) -> handle
PyObject* py_binary_bcast(PyObject*, PyObject* args) {
  unsigned long long ha, hb;
  int op, dt, mode;
  long long outer, inner;
  if (!PyArg_ParseTuple(args, "KKiiLLi", &ha, &hb, &op, &dt, &outer, &inner,
                        &mode))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  DType dtype = dtype_from_int(dt);
  DevBuf& a = get_buf(ha);
  DevBuf& b = get_buf(hb);
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(a.size);
  launch_binary_bcast(dtype, (BinOp)op, mode, a.ptr, b.ptr, out, outer,
                      inner, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, a.size));
  WRAP_END
}

// argminmax(h, dtype, n, maxop) -> int64 index (numpy tie/NaN semantics)
PyObject* py_argminmax(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt, maxop;
  long long n;
  if (!PyArg_ParseTuple(args, "KiLi", &h, &dt, &n, &maxop)) return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (n <= 0) throw std::runtime_error("argminmax of empty array");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  long long result = 0;
  NOGIL_BEGIN
  std::lock_guard<std::mutex> reduce_lk(g.reduce_mu);
  ensure_reduce_scratch();
  // reuse the pinned scalar as the int64 result landing slot
  launch_argminmax(dtype, maxop != 0, in.ptr, g.reduce_scratch, g.scalar_dev,
                   n, g.compute);
  HIP_CHECK(hipMemcpyAsync(g.scalar_pin, g.scalar_dev, 8,
                           hipMemcpyDeviceToHost, g.compute));
  HIP_CHECK(hipStreamSynchronize(g.compute));
  result = *(int64_t*)g.scalar_pin;
  NOGIL_END
  return PyLong_FromLongLong(result);
  WRAP_END
}

// reduce_axis(h, dtype, outer, red, inner, mode) -> handle
// contiguous [outer][red][inner] reduced over the middle axis
PyObject* py_reduce_axis(PyObject*, PyObject* args) {
  unsigned long long h;
  int dt, mode;
  long long outer, red, inner;
  if (!PyArg_ParseTuple(args, "KiLLLi", &h, &dt, &outer, &red, &inner, &mode))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (mode < 0 || mode > 3) throw std::runtime_error("bad reduce mode");
  DType dtype = dtype_from_int(dt);
  DevBuf& in = get_buf(h);
  int64_t esize = dtype == DType::F64 ? 8 : 4;
  int64_t out_size = outer * inner * esize;
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(out_size);
  launch_reduce_axis(dtype, (ReduceOp)mode, in.ptr, out, outer, red, inner,
                     g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, out_size));
  WRAP_END
}

// gemm_batched(hA, hB, batch, m, n, k, dtype) -> handle
// C[b] = A[b] @ B[b] for contiguous [batch][m][k] x [batch][k][n];
// per-batch kernel launches enqueue back-to-back on the compute stream
PyObject* py_gemm_batched(PyObject*, PyObject* args) {
  unsigned long long ha, hb;
  int batch, m, n, k, dt;
  if (!PyArg_ParseTuple(args, "KKiiiii", &ha, &hb, &batch, &m, &n, &k, &dt))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  if (dt != 0 && dt != 1)
    throw std::runtime_error("gemm_batched supports f32/f64");
  DevBuf& a = get_buf(ha);
  DevBuf& b = get_buf(hb);
  int64_t esize = dt == 1 ? 8 : 4;
  int64_t out_size = (int64_t)batch * m * n * esize;
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc(out_size);
  for (int i = 0; i < batch; i++) {
    const char* pa = (const char*)a.ptr + (int64_t)i * m * k * esize;
    const char* pb = (const char*)b.ptr + (int64_t)i * k * n * esize;
    char* pc = (char*)out + (int64_t)i * m * n * esize;
    if (dt == 1)
      launch_gemm_f64((const double*)pa, (const double*)pb, (double*)pc, m,
                      n, k, g.compute);
    else
      launch_gemm_f32((const float*)pa, (const float*)pb, (float*)pc, m, n,
                      k, g.compute);
  }
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, out_size));
  WRAP_END
}

// gemm(hA, hB, m, n, k, dtype) -> handle  (row-major C = A @ B)
PyObject* py_gemm(PyObject*, PyObject* args) {
  unsigned long long ha, hb;
  int m, n, k, dt;
  if (!PyArg_ParseTuple(args, "KKiiii", &ha, &hb, &m, &n, &k, &dt))
    return nullptr;
  WRAP_BEGIN
  ensure_init();
  DevBuf& a = get_buf(ha);
  DevBuf& b = get_buf(hb);
  int64_t esize = dt == 1 ? 8 : (dt == 0 ? 4 : 2);
  void* out = nullptr;
  NOGIL_BEGIN
  out = pool_alloc((int64_t)m * n * esize);
  if (dt == 1)
    launch_gemm_f64((const double*)a.ptr, (const double*)b.ptr, (double*)out,
                    m, n, k, g.compute);
  else if (dt == 0)
    launch_gemm_f32((const float*)a.ptr, (const float*)b.ptr, (float*)out, m,
                    n, k, g.compute);
  else if (gemm_bf16_256_supported(m, n, k)) {
    if (use_bf16_tr16()) {
      // fast path: B consumed native [K][N] through the tr16 image
      launch_gemm_bf16_256t((const uint16_t*)a.ptr, (const uint16_t*)b.ptr,
                            (uint16_t*)out, m, n, k, g.compute);
    } else {
      // r01 path: pre-transpose B to [N][K] so both operands stream
      // K-contiguous through glds
      void* bt = pool_alloc((int64_t)n * k * 2);
      launch_transpose_bf16((const uint16_t*)b.ptr, (uint16_t*)bt, k, n,
                            g.compute);
      launch_gemm_bf16_256((const uint16_t*)a.ptr, (const uint16_t*)bt,
                           (uint16_t*)out, m, n, k, g.compute);
      pool_free(bt, (int64_t)n * k * 2);
    }
  } else if (2.0 * m * n * k >= 4e9) {
    // large non-aligned shape: zero-pad to the 256-tile fast path
    // (measured: the general 128-tile kernel runs ~120-160 TF vs ~1130
    // for the 256 path at 4k-class shapes; padding costs three pitched
    // device copies + two memsets at HBM rate). Zero rows/cols
    // contribute exact zeros, so the extracted C block is bit-identical
    // to the unpadded computation.
    int mp = (m + 255) & ~255;
    int np2 = (n + 255) & ~255;
    int kp = (k + 127) & ~127;
    void* ap = pool_alloc((int64_t)mp * kp * 2);
    void* bp = pool_alloc((int64_t)kp * np2 * 2);
    void* cp = pool_alloc((int64_t)mp * np2 * 2);
    HIP_CHECK(hipMemsetAsync(ap, 0, (int64_t)mp * kp * 2, g.compute));
    HIP_CHECK(hipMemsetAsync(bp, 0, (int64_t)kp * np2 * 2, g.compute));
    HIP_CHECK(hipMemcpy2DAsync(ap, (size_t)kp * 2, a.ptr, (size_t)k * 2,
                               (size_t)k * 2, m, hipMemcpyDeviceToDevice,
                               g.compute));
    HIP_CHECK(hipMemcpy2DAsync(bp, (size_t)np2 * 2, b.ptr, (size_t)n * 2,
                               (size_t)n * 2, k, hipMemcpyDeviceToDevice,
                               g.compute));
    if (use_bf16_tr16()) {
      launch_gemm_bf16_256t((const uint16_t*)ap, (const uint16_t*)bp,
                            (uint16_t*)cp, mp, np2, kp, g.compute);
    } else {
      void* bt = pool_alloc((int64_t)np2 * kp * 2);
      launch_transpose_bf16((const uint16_t*)bp, (uint16_t*)bt, kp, np2,
                            g.compute);
      launch_gemm_bf16_256((const uint16_t*)ap, (const uint16_t*)bt,
                           (uint16_t*)cp, mp, np2, kp, g.compute);
      pool_free(bt, (int64_t)np2 * kp * 2);
    }
    HIP_CHECK(hipMemcpy2DAsync(out, (size_t)n * 2, cp, (size_t)np2 * 2,
                               (size_t)n * 2, m, hipMemcpyDeviceToDevice,
                               g.compute));
    pool_free(ap, (int64_t)mp * kp * 2);
    pool_free(bp, (int64_t)kp * np2 * 2);
    pool_free(cp, (int64_t)mp * np2 * 2);
  } else
    launch_gemm_bf16((const uint16_t*)a.ptr, (const uint16_t*)b.ptr,
                     (uint16_t*)out, m, n, k, g.compute);
  NOGIL_END
  return PyLong_FromUnsignedLongLong(register_buf(out, (int64_t)m * n * esize));
  WRAP_END
}

// torch interop: raw-pointer GEMM on the CALLER's stream and context.
// The sandbox torch hook (ops/hiptorch.py) passes torch-allocated device
// buffers and torch's current HIP stream, so the hand-written MFMA
// kernels run inside torch's own context with torch's stream ordering --
// no _hipops state (init, pools, staging) is touched. dtype codes:
// 0=f32, 1=f64, 2=bf16. For bf16 the caller passes scratch for the
// B pre-transpose when the 256^2-tile fast path applies (bt != 0).
PyObject* py_gemm_raw(PyObject*, PyObject* args) {
  unsigned long long pa, pb, pc, pbt, stream;
  int m, n, k, dt;
  if (!PyArg_ParseTuple(args, "KKKKiiiiK", &pa, &pb, &pc, &pbt, &m, &n, &k,
                        &dt, &stream))
    return nullptr;
  WRAP_BEGIN
  if (dt < 0 || dt > 2) throw std::runtime_error("gemm_raw: bad dtype code");
  hipStream_t s = (hipStream_t)stream;
  NOGIL_BEGIN
  if (dt == 1)
    launch_gemm_f64((const double*)pa, (const double*)pb, (double*)pc, m, n,
                    k, s);
  else if (dt == 0)
    launch_gemm_f32((const float*)pa, (const float*)pb, (float*)pc, m, n, k,
                    s);
  else if (dt == 2) {
    if (gemm_bf16_256_supported(m, n, k) && use_bf16_tr16()) {
      launch_gemm_bf16_256t((const uint16_t*)pa, (const uint16_t*)pb,
                            (uint16_t*)pc, m, n, k, s);
    } else if (pbt && gemm_bf16_256_supported(m, n, k)) {
      launch_transpose_bf16((const uint16_t*)pb, (uint16_t*)pbt, k, n, s);
      launch_gemm_bf16_256((const uint16_t*)pa, (const uint16_t*)pbt,
                           (uint16_t*)pc, m, n, k, s);
    } else {
      launch_gemm_bf16((const uint16_t*)pa, (const uint16_t*)pb,
                       (uint16_t*)pc, m, n, k, s);
    }
  }
  NOGIL_END
  Py_RETURN_NONE;
  WRAP_END
}

PyObject* py_gemm_bf16_256_ok(PyObject*, PyObject* args) {
  int m, n, k;
  if (!PyArg_ParseTuple(args, "iii", &m, &n, &k)) return nullptr;
  if (gemm_bf16_256_supported(m, n, k)) Py_RETURN_TRUE;
  Py_RETURN_FALSE;
}

PyObject* py_synchronize(PyObject*, PyObject*) {
  WRAP_BEGIN
  ensure_init();
  NOGIL_BEGIN
  HIP_CHECK(hipStreamSynchronize(g.compute));
  HIP_CHECK(hipStreamSynchronize(g.copy));
  NOGIL_END
  Py_RETURN_NONE;
  WRAP_END
}

PyObject* py_mem_info(PyObject*, PyObject*) {
  WRAP_BEGIN
  ensure_init();
  size_t free_b = 0, total_b = 0;
  HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
  return Py_BuildValue("(KKLlll)", (unsigned long long)free_b,
                       (unsigned long long)total_b,
                       (long long)g.outstanding, g.mempool_allocs,
                       g.fallback_allocs, g.oom_trims);
  WRAP_END
}

PyMethodDef methods[] = {
    {"is_available", py_is_available, METH_NOARGS, "GPU present?"},
    {"device_count", py_device_count, METH_NOARGS, "visible GPU count"},
    {"init", py_init, METH_VARARGS, "init(device=0)"},
    {"alloc", py_alloc, METH_VARARGS, "alloc(nbytes) -> handle"},
    {"free", py_free, METH_VARARGS, "free(handle)"},
    {"upload", py_upload, METH_VARARGS, "upload(buffer) -> handle"},
    {"download", py_download, METH_VARARGS, "download(handle, buffer)"},
    {"download_strided", py_download_strided, METH_VARARGS,
     "download_strided(handle, off, stride, elem_bytes, count) -> bytes"},
    {"download_slice", py_download_slice, METH_VARARGS,
     "download_slice(handle, byte_offset, nbytes) -> bytes"},
    {"rand", py_rand, METH_VARARGS, "rand(n, dtype, seed) -> handle"},
    {"randn", py_randn, METH_VARARGS,
     "randn(n, seed, mu, sigma) -> handle (f64 normal)"},
    {"unary", py_unary, METH_VARARGS, "unary(h, op, dtype, n) -> handle"},
    {"convert", py_convert, METH_VARARGS,
     "convert(h, src_dt, dst_dt, n) -> handle (f64<->f32)"},
    {"binary", py_binary, METH_VARARGS, "binary(ha, hb, op, dtype, n) -> handle"},
    {"binary_scalar", py_binary_scalar, METH_VARARGS,
     "binary_scalar(h, scalar, op, dtype, n) -> handle"},
    {"sum", py_sum, METH_VARARGS, "sum(h, dtype, n, mode) -> float (mode 0=sum 1=sumsq 2=max 3=min)"},
    {"gemm", py_gemm, METH_VARARGS, "gemm(ha, hb, m, n, k, dtype) -> handle"},
    {"cumsum", py_cumsum, METH_VARARGS, "cumsum(h, dtype, n) -> handle"},
    {"searchsorted", py_searchsorted, METH_VARARGS,
     "searchsorted(ha, n, hv, m, dtype, right) -> int64 handle"},
    {"diff", py_diff, METH_VARARGS,
     "diff(h, dtype, outer, inner) -> handle"},
    {"cumsum2d", py_cumsum2d, METH_VARARGS,
     "cumsum2d(h, dtype, rows, cols) -> handle (row-wise scan)"},
    {"copy_d2d", py_copy_d2d, METH_VARARGS,
     "copy_d2d(dst, dst_off, src, src_off, nbytes)"},
    {"transpose", py_transpose, METH_VARARGS,
     "transpose(h, dtype, rows, cols) -> handle"},
    {"sort2d", py_sort2d, METH_VARARGS,
     "sort2d(h, dtype, rows, cols, want_idx) -> handle | (handle, idx)"},
    {"sort", py_sort, METH_VARARGS,
     "sort(h, dtype, n, want_idx) -> handle | (handle, int64 idx handle)"},
    {"mask_logic", py_mask_logic, METH_VARARGS,
     "mask_logic(ha, hb_or_0, n, op) -> u8 handle (0 and 1 or 2 xor 3 andnot 4 not)"},
    {"histogram", py_histogram, METH_VARARGS,
     "histogram(h, dtype, n, lo, hi, bins) -> bytes of u64 counts"},
    {"extract_range", py_extract_range, METH_VARARGS,
     "extract_range(h, dtype, n, lo, hi, cap) -> (count, f64 bytes)"},
    {"compare", py_compare, METH_VARARGS,
     "compare(h, dtype, n, cmp_op, hb_or_0, scalar) -> u8 mask handle"},
    {"where", py_where, METH_VARARGS,
     "where(hmask, dtype, n, ha_or_0, sa, hb_or_0, sb) -> handle"},
    {"masked_fill", py_masked_fill, METH_VARARGS,
     "masked_fill(h, hmask, dtype, n, value): in place"},
    {"mask_count", py_mask_count, METH_VARARGS,
     "mask_count(hmask, n) -> int64"},
    {"binary_bcast", py_binary_bcast, METH_VARARGS,
     "binary_bcast(ha, hb, op, dtype, outer, inner, mode) -> handle"},
    {"argminmax", py_argminmax, METH_VARARGS,
     "argminmax(h, dtype, n, maxop) -> int64 index"},
    {"reduce_axis", py_reduce_axis, METH_VARARGS,
     "reduce_axis(h, dtype, outer, red, inner, mode) -> handle"},
    {"gemm_batched", py_gemm_batched, METH_VARARGS,
     "gemm_batched(ha, hb, batch, m, n, k, dtype) -> handle"},
    {"gemm_raw", py_gemm_raw, METH_VARARGS,
     "gemm_raw(pa, pb, pc, pbt, m, n, k, dtype, stream): raw-pointer GEMM "
     "on the caller's stream (torch interop; dtype 0=f32 1=f64 2=bf16)"},
    {"gemm_bf16_256_ok", py_gemm_bf16_256_ok, METH_VARARGS,
     "gemm_bf16_256_ok(m, n, k) -> bool (256-tile fast path applies?)"},
    {"synchronize", py_synchronize, METH_NOARGS, "sync all streams"},
    {"mem_info", py_mem_info, METH_NOARGS, "(free, total) bytes"},
    {nullptr, nullptr, 0, nullptr},
};

struct PyModuleDef module_def = {
    PyModuleDef_HEAD_INIT, "_hipops",
    "gfx950 HIP kernel library (MFMA GEMM, elementwise, reduction, RNG, "
    "pinned staging)",
    -1, methods,
};

}  // namespace

PyMODINIT_FUNC PyInit__hipops(void) { return PyModule_Create(&module_def); }
