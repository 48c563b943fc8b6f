// _hipgemm: the torch-interop slice of the gfx950 kernel library as its
// OWN extension module.
//
// Why a second module: sandbox children are forked from a zygote that
// pre-imported _hipops (the numpy path's CFS-throttle fix -- the HIP
// userspace stack loads once, pre-fork). A fat-binary registered in the
// PARENT is not usable after the CHILD re-initializes the HIP runtime
// (torch's CUDA init): launching such a kernel segfaults (measured:
// scripts/torch_case.py --steps; zygote child dies at gemm_raw, cold
// child is fine). The torch routing layer therefore loads THIS module,
// which the zygote never imports, so every torch-using child dlopens a
// fresh copy and registration happens against the child's live runtime.
//
// Surface: stateless raw-pointer entry points only -- the caller (torch)
// owns memory and streams.

#define PY_SSIZE_T_CLEAN
#include <Python.h>

#include <string>

#include "common.h"

namespace {

#define NOGIL_BEGIN           \
  std::string _nogil_err;     \
  Py_BEGIN_ALLOW_THREADS;     \
  try {
#define NOGIL_END                                               \
  }                                                             \
  catch (const std::exception& _e) { _nogil_err = _e.what(); }  \
  Py_END_ALLOW_THREADS;                                         \
  if (!_nogil_err.empty()) throw std::runtime_error(_nogil_err);

#define WRAP_BEGIN try {
#define WRAP_END                                  \
  }                                               \
  catch (const std::exception& e) {               \
    PyErr_SetString(PyExc_RuntimeError, e.what()); \
    return nullptr;                               \
  }

PyObject* py_is_available(PyObject*, PyObject*) {
  int count = 0;
  hipError_t e = hipGetDeviceCount(&count);
  if (e != hipSuccess || count <= 0) Py_RETURN_FALSE;
  Py_RETURN_TRUE;
}

// gemm_raw(pa, pb, pc, pbt, m, n, k, dtype, stream): dtype 0=f32 1=f64
// 2=bf16; pbt = scratch for the variant-b bf16 pre-transpose (unused by
// the default tr16 variant)
PyObject* py_gemm_raw(PyObject*, PyObject* args) {
  unsigned long long pa, pb, pc, pbt, stream;
  int m, n, k, dt;
  if (!PyArg_ParseTuple(args, "KKKKiiiiK", &pa, &pb, &pc, &pbt, &m, &n, &k,
                        &dt, &stream))
    return nullptr;
  WRAP_BEGIN
  if (dt < 0 || dt > 2) throw std::runtime_error("gemm_raw: bad dtype code");
  const char* variant = getenv("APP_BF16_256_VARIANT");
  bool tr16 = !(variant && variant[0] == 'b');
  hipStream_t s = (hipStream_t)stream;
  NOGIL_BEGIN
  if (dt == 1)
    launch_gemm_f64((const double*)pa, (const double*)pb, (double*)pc, m, n,
                    k, s);
  else if (dt == 0)
    launch_gemm_f32((const float*)pa, (const float*)pb, (float*)pc, m, n, k,
                    s);
  else {
    if (gemm_bf16_256_supported(m, n, k) && tr16) {
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

// gemm_raw_nt(pa, pw, pc, m, n, k, stream): bf16 C[M,N] = A[M,K] @ W^T
// where W is [N][K] row-major -- torch's nn.Linear weight layout IS the
// 256-tile kernel's pre-transposed B operand, so linear routes with
// ZERO transpose work. 256-supported shapes only (caller checks).
PyObject* py_gemm_raw_nt(PyObject*, PyObject* args) {
  unsigned long long pa, pw, pc, stream;
  int m, n, k;
  if (!PyArg_ParseTuple(args, "KKKiiiK", &pa, &pw, &pc, &m, &n, &k, &stream))
    return nullptr;
  WRAP_BEGIN
  if (!gemm_bf16_256_supported(m, n, k))
    throw std::runtime_error("gemm_raw_nt: shape not 256-tile supported");
  hipStream_t s = (hipStream_t)stream;
  NOGIL_BEGIN
  launch_gemm_bf16_256((const uint16_t*)pa, (const uint16_t*)pw,
                       (uint16_t*)pc, m, n, k, s);
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

PyMethodDef methods[] = {
    {"is_available", py_is_available, METH_NOARGS, "GPU present?"},
    {"gemm_raw", py_gemm_raw, METH_VARARGS,
     "gemm_raw(pa, pb, pc, pbt, m, n, k, dtype, stream)"},
    {"gemm_raw_nt", py_gemm_raw_nt, METH_VARARGS,
     "gemm_raw_nt(pa, pw, pc, m, n, k, stream): bf16 A @ W^T, W=[N][K]"},
    {"gemm_bf16_256_ok", py_gemm_bf16_256_ok, METH_VARARGS,
     "gemm_bf16_256_ok(m, n, k) -> bool"},
    {nullptr, nullptr, 0, nullptr},
};

struct PyModuleDef module_def = {
    PyModuleDef_HEAD_INIT, "_hipgemm",
    "gfx950 MFMA GEMM kernels for the torch routing layer (raw pointers, "
    "caller's stream)",
    -1, methods,
};

}  // namespace

PyMODINIT_FUNC PyInit__hipgemm(void) { return PyModule_Create(&module_def); }
