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

STATS = {"mm_routed": 0, "mm_fallback": 0}

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


def _make_mode(torch):
    from torch.overrides import TorchFunctionMode

    mm_funcs = {
        torch.matmul,
        torch.mm,
        torch.Tensor.matmul,
        torch.Tensor.__matmul__,
    }

    class HipMatmulMode(TorchFunctionMode):
        """Global interception of the matmul family; self-excluded while
        handling (torch guarantees a mode is not re-entered from its own
        __torch_function__), so internal empty/contiguous calls are
        ordinary torch."""

        def __torch_function__(self, func, types, args=(), kwargs=None):
            kwargs = kwargs or {}
            if func in mm_funcs and len(args) == 2 and not kwargs:
                try:
                    r = _try_mm(torch, args[0], args[1])
                except Exception:
                    if os.environ.get("APP_HIP_TORCH", "auto").lower() == "require":
                        raise
                    r = None
                if r is not None:
                    return r
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
