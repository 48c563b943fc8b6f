"""Same-box A/B: torch-native (hipBLASLt) vs hand-written MFMA kernel on
the acceptance shape (bf16 8192^3), plus f32/f64 checks. Run on a GPU box:

    python scripts/torch_mm_ab.py [size]

Prints both paths' ms and TF/s from one process, back to back, so the
numbers share clocks/board state (the standing rule from r01: no claim
without a same-box A/B).
"""

import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO / "code_interpreter_amd" / "ops"))

import torch  # noqa: E402

import hiptorch  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    size = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    assert torch.cuda.is_available()
    dev = "cuda"
    for dtype, name in (
        (torch.bfloat16, "bf16"),
        (torch.float32, "f32"),
        (torch.float64, "f64"),
    ):
        a = torch.randn(size, size, dtype=dtype, device=dev)
        b = torch.randn(size, size, dtype=dtype, device=dev)
        flops = 2 * size**3

        t_ref = bench(lambda: a @ b)
        assert hiptorch._state["mode_obj"] is None
        ref = (a @ b) if size <= 8192 else None

        assert hiptorch.install(mode="require")
        try:
            routed0 = hiptorch.STATS["mm_routed"]
            t_ours = bench(lambda: a @ b)
            assert hiptorch.STATS["mm_routed"] > routed0, "not routed!"
            if ref is not None:
                got = a @ b
                err = (got.float() - ref.float()).abs().max().item()
                scale = ref.float().abs().max().item()
                print(f"{name} {size}^3 relerr(max) = {err / scale:.2e}")
        finally:
            hiptorch.uninstall()

        print(
            f"{name} {size}^3: torch/hipBLASLt {t_ref*1e3:.3f} ms "
            f"({flops/t_ref/1e12:.0f} TF)  |  hand-written "
            f"{t_ours*1e3:.3f} ms ({flops/t_ours/1e12:.0f} TF)  "
            f"ratio={t_ref/t_ours:.3f}x"
        )


if __name__ == "__main__":
    main()
