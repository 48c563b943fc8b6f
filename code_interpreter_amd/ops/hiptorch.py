"""torch -> gfx950 hand-written kernel routing for sandboxed user code.

The MI355X mandate routes user *numpy and torch* compute to the
hand-written CDNA4 kernels (BASELINE.json north star; reference analog:
the user-compute path of /v1/execute-custom-tool,
/root/reference/src/code_interpreter/services/custom_tool_executor.py:157-195).
hipnp.py covers numpy; this module covers torch:

- a global ``TorchFunctionMode`` intercepts the matmul family
  (``torch.matmul``, ``torch.mm``, ``a @ b``) and runs eligible 2-D
  GEMMs through ``_hipgemm.gemm_raw`` -- the same MFMA kernels rocprof
  shows on the numpy path (gemm_bf16_256b / gemm_f32 / gemm_f64) --
  ON TORCH'S OWN current HIP stream and memory (``data_ptr``), so
  stream ordering and the caching allocator behave exactly as for a
  native torch op;
- everything else (batched/odd shapes, other dtypes, CPU tensors,
  autograd graphs needing grad) falls through to torch untouched.

Install is idempotent and cheap; ``mode="require"`` raises when no GPU
or extension is present (GPU boxes must not silently fall back to
hipBLASLt). Imported standalone inside sandbox children (sys.path), like
hipnp; importing does NOT import torch (that costs ~1.5 s) -- install()
is called from the sandbox runtime's import hook after the user imports
torch.
"""

import os
import sys

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
if _OPS_DIR not in sys.path:
    sys.path.insert(0, _OPS_DIR)

# _hipgemm (the GEMM-kernel extension) is imported LAZILY, inside
# install(), strictly AFTER `import torch`: torch ships its own
# libamdhip64 (SONAME libamdhip64.so.7) and the dynamic loader binds
# _hipgemm's soname dependency to the FIRST matching instance -- with
# torch loaded first that is torch's own runtime, so our kernels
# register with and launch on exactly the runtime whose streams torch
# hands us. Binding to any other instance is fragile (and an instance
# loaded before the zygote fork segfaults at launch -- isolated by
# scripts/torch_case.py --steps). The zygote never imports this module.
_hipgemm = None


def _ensure_hipgemm():
    global _hipgemm
    if _hipgemm is None:
        import torch  # noqa: F401  (load torch's HIP runtime first)

        import _hipgemm as m

        _hipgemm = m
    return _hipgemm

# route only when the kernel launch is worth more than its overhead;
# env-tunable like hipnp's thresholds
MIN_MM_FLOPS = float(os.environ.get("APP_HIP_TORCH_MIN_FLOPS", 5e7))

STATS = {"mm_routed": 0, "mm_fallback": 0, "linear_routed": 0,
         "bmm_routed": 0}

_state = {"mode_obj": None, "torch": None}


def available() -> bool:
    try:
        return _ensure_hipgemm().is_available()
    except ImportError:
        return False


def _dtype_code(torch, dtype):
    if dtype == torch.float32:
        return 0
    if dtype == torch.float64:
        return 1
    if dtype == torch.bfloat16:
        return 2
    return None


def _try_mm(torch, a, b):
    """Route one 2-D matmul to the hand-written kernels; None = not
    eligible (caller falls back to torch)."""
    if not (isinstance(a, torch.Tensor) and isinstance(b, torch.Tensor)):
        return None
    if not (a.is_cuda and b.is_cuda) or a.device != b.device:
        return None
    if a.dtype != b.dtype:
        return None
    dt = _dtype_code(torch, a.dtype)
    if dt is None:
        return None
    if a.dim() != 2 or b.dim() != 2 or a.shape[1] != b.shape[0]:
        return None
    if a.requires_grad or b.requires_grad:
        # autograd needs the aten graph; only inference work is routed
        if torch.is_grad_enabled():
            return None
    m, k = a.shape
    n = b.shape[1]
    if m == 0 or n == 0 or k == 0 or 2.0 * m * n * k < MIN_MM_FLOPS:
        return None
    if (
        dt == 2
        and not _hipgemm.gemm_bf16_256_ok(m, n, k)
        and 2.0 * m * n * k >= 4e9
    ):
        # large non-aligned bf16: zero-pad to the 256-tile fast path
        # (~9x the general kernel; zero padding is exact)
        mp, np_, kp = (m + 255) & ~255, (n + 255) & ~255, (k + 127) & ~127
        a_p = torch.zeros((mp, kp), dtype=a.dtype, device=a.device)
        a_p[:m, :k] = a
        b_p = torch.zeros((kp, np_), dtype=b.dtype, device=b.device)
        b_p[:k, :n] = b
        c_p = _try_mm(torch, a_p, b_p)
        if c_p is not None:
            return c_p[:m, :n]
    a = a.contiguous()
    b = b.contiguous()
    c = torch.empty((m, n), dtype=a.dtype, device=a.device)
    bt_ptr = 0
    bt = None
    if dt == 2 and _hipgemm.gemm_bf16_256_ok(m, n, k):
        # scratch for the B pre-transpose (the 256^2-tile kernel consumes
        # both operands K-contiguous); torch-allocated on the same stream,
        # so the caching allocator's stream ordering keeps it live until
        # the GEMM that reads it has retired
        bt = torch.empty((n, k), dtype=a.dtype, device=a.device)
        bt_ptr = bt.data_ptr()
    stream = torch.cuda.current_stream(a.device).cuda_stream
    _hipgemm.gemm_raw(
        a.data_ptr(), b.data_ptr(), c.data_ptr(), bt_ptr, m, n, k, dt, stream
    )
    del bt
    STATS["mm_routed"] += 1
    return c


def _try_linear(torch, x, w, bias=None):
    """Route F.linear: y = x @ w.T (+ bias). torch's weight layout
    [out,in] row-major IS the 256-tile kernel's pre-transposed B
    operand, so eligible bf16 linears run with zero transpose work."""
    if not (isinstance(x, torch.Tensor) and isinstance(w, torch.Tensor)):
        return None
    if not (x.is_cuda and w.is_cuda) or x.dtype != w.dtype:
        return None
    if x.dtype != torch.bfloat16 or w.dim() != 2 or x.dim() < 2:
        return None
    if torch.is_grad_enabled() and (
        x.requires_grad or w.requires_grad
        or (bias is not None and bias.requires_grad)
    ):
        return None
    n, k = w.shape
    if x.shape[-1] != k:
        return None
    lead = x.shape[:-1]
    m = 1
    for d in lead:
        m *= d
    if 2.0 * m * n * k < MIN_MM_FLOPS:
        return None
    if not _hipgemm.gemm_bf16_256_ok(m, n, k):
        return None
    x2 = x.reshape(m, k).contiguous()
    w = w.contiguous()
    c = torch.empty((m, n), dtype=x.dtype, device=x.device)
    stream = torch.cuda.current_stream(x.device).cuda_stream
    _hipgemm.gemm_raw_nt(x2.data_ptr(), w.data_ptr(), c.data_ptr(), m, n, k,
                         stream)
    if bias is not None:
        c = c + bias
    STATS["linear_routed"] += 1
    return c.reshape(*lead, n)


def _try_bmm(torch, a, b):
    """Route equal-batch 3-D matmuls as per-batch GEMM launches enqueued
    back-to-back on torch's stream."""
    if not (isinstance(a, torch.Tensor) and isinstance(b, torch.Tensor)):
        return None
    if not (a.is_cuda and b.is_cuda) or a.dtype != b.dtype:
        return None
    dt = _dtype_code(torch, a.dtype)
    if dt is None or a.dim() != 3 or b.dim() != 3:
        return None
    if a.shape[0] != b.shape[0] or a.shape[2] != b.shape[1]:
        return None
    if torch.is_grad_enabled() and (a.requires_grad or b.requires_grad):
        return None
    batch, m, k = a.shape
    n = b.shape[2]
    if batch == 0 or 2.0 * m * n * k < MIN_MM_FLOPS:
        return None
    a = a.contiguous()
    b = b.contiguous()
    c = torch.empty((batch, m, n), dtype=a.dtype, device=a.device)
    es = a.element_size()
    stream = torch.cuda.current_stream(a.device).cuda_stream
    for i in range(batch):
        bt_ptr = 0
        bt = None
        if dt == 2 and _hipgemm.gemm_bf16_256_ok(m, n, k):
            bt = torch.empty((n, k), dtype=a.dtype, device=a.device)
            bt_ptr = bt.data_ptr()
        _hipgemm.gemm_raw(
            a.data_ptr() + i * m * k * es,
            b.data_ptr() + i * k * n * es,
            c.data_ptr() + i * m * n * es,
            bt_ptr, m, n, k, dt, stream,
        )
        del bt
    STATS["bmm_routed"] += 1
    return c


def _make_mode(torch):
    from torch.overrides import TorchFunctionMode

    mm_funcs = {
        torch.matmul,
        torch.mm,
        torch.Tensor.matmul,
        torch.Tensor.__matmul__,
    }
    bmm_funcs = {torch.bmm, torch.Tensor.bmm}
    linear_funcs = {torch.nn.functional.linear}

    def _guarded(fn, *args):
        try:
            return fn(torch, *args)
        except Exception:
            if os.environ.get("APP_HIP_TORCH", "auto").lower() == "require":
                raise
            return None

    class HipMatmulMode(TorchFunctionMode):
        """Global interception of the matmul family; self-excluded while
        handling (torch guarantees a mode is not re-entered from its own
        __torch_function__), so internal empty/contiguous calls are
        ordinary torch."""

        def __torch_function__(self, func, types, args=(), kwargs=None):
            kwargs = kwargs or {}
            r = None
            if func in mm_funcs and len(args) == 2 and not kwargs:
                a, b = args
                if (
                    isinstance(a, torch.Tensor) and isinstance(b, torch.Tensor)
                    and a.dim() == 3 and b.dim() == 3
                ):
                    r = _guarded(_try_bmm, a, b)
                else:
                    r = _guarded(_try_mm, a, b)
            elif func in bmm_funcs and len(args) == 2 and not kwargs:
                r = _guarded(_try_bmm, args[0], args[1])
            elif func in linear_funcs and 2 <= len(args) <= 3 and set(
                kwargs
            ) <= {"bias"}:
                bias = args[2] if len(args) == 3 else kwargs.get("bias")
                r = _guarded(_try_linear, args[0], args[1], bias)
            if r is not None:
                return r
            if func in mm_funcs or func in bmm_funcs or func in linear_funcs:
                STATS["mm_fallback"] += 1
            return func(*args, **kwargs)

    return HipMatmulMode()


def install(mode: str = "auto") -> bool:
    """Enter the routing mode globally for this process. Returns True if
    installed; mode="require" raises when the HIP path is unusable."""
    if _state["mode_obj"] is not None:
        return True
    if mode == "off":
        return False
    import torch

    if not (torch.cuda.is_available() and _ensure_hipgemm().is_available()):
        if mode == "require":
            raise RuntimeError("APP_HIP_TORCH=require but no AMD GPU is visible")
        return False
    m = _make_mode(torch)
    m.__enter__()  # permanent: routes for the rest of the process
    _state["mode_obj"] = m
    _state["torch"] = torch
    return True


def uninstall() -> None:
    m = _state["mode_obj"]
    if m is not None:
        m.__exit__(None, None, None)
        _state["mode_obj"] = None
