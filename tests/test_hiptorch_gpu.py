"""GPU tests for the torch->HIP matmul routing (ops/hiptorch.py).

In their own file ON PURPOSE: this file sorts before test_ops_gpu.py, so
torch's CUDA runtime initializes BEFORE any _hipops context exists in
the pytest process -- the reverse order makes torch.cuda.is_available()
flaky on some boxes (two libamdhip64 instances; see profiles/NOTES.md
"HIP runtime binding rules"), which silently skipped these tests.
torch-first is the same order every sandbox child sees.
"""

import sys
from pathlib import Path

import numpy as np  # noqa: F401
import pytest

OPS_DIR = Path(__file__).resolve().parent.parent / "code_interpreter_amd" / "ops"
sys.path.insert(0, str(OPS_DIR))

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hip_torch():
    # torch initializes FIRST in this process (no _hipops dependency):
    # the order every sandbox child sees
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("torch sees no GPU")
    import hiptorch

    assert hiptorch.install(mode="require")
    yield torch, hiptorch
    hiptorch.uninstall()


@pytest.mark.parametrize(
    "dtype,shape,tol",
    [
        # shapes sit ABOVE MIN_MM_FLOPS (5e7) so routing must engage
        ("float32", (512, 384, 256), 2e-5),
        ("float64", (512, 384, 256), 1e-12),
        ("bfloat16", (512, 512, 256), 3e-2),   # 256-tile fast path
        ("bfloat16", (520, 360, 192), 3e-2),   # general bf16 kernel
    ],
)
def test_torch_mm_routed_matches_torch(hip_torch, dtype, shape, tol):
    torch, hiptorch = hip_torch
    dt = getattr(torch, dtype)
    m, n, k = shape
    torch.manual_seed(0)
    a = torch.randn(m, k, dtype=dt, device="cuda")
    b = torch.randn(k, n, dtype=dt, device="cuda")
    before = hiptorch.STATS["mm_routed"]
    c = a @ b
    assert hiptorch.STATS["mm_routed"] == before + 1, "matmul not routed"
    # plain-torch fp64 CPU reference (same inputs, library-free path)
    ref = a.double().cpu() @ b.double().cpu()
    err = (c.double().cpu() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-9
    assert err / scale < tol, (err, scale)


def test_torch_mm_transpose_detecting(hip_torch):
    torch, hiptorch = hip_torch
    # A = I with asymmetric B: catches operand/output transposes
    n = 512
    a = torch.eye(n, dtype=torch.bfloat16, device="cuda")
    b = (torch.arange(n * n, dtype=torch.float32, device="cuda")
         .reshape(n, n) / (n * n)).to(torch.bfloat16)
    c = torch.matmul(a, b)
    torch.testing.assert_close(c, b, rtol=0, atol=0)


def test_torch_small_and_batched_fall_back(hip_torch):
    torch, hiptorch = hip_torch
    before = hiptorch.STATS["mm_routed"]
    small = torch.randn(8, 8, dtype=torch.float32, device="cuda")
    _ = small @ small  # below MIN_MM_FLOPS
    a3 = torch.randn(2, 64, 64, dtype=torch.float32, device="cuda")
    _ = torch.matmul(a3, a3)  # batched: aten path
    assert hiptorch.STATS["mm_routed"] == before


def test_torch_grad_path_untouched(hip_torch):
    torch, _ = hip_torch
    a = torch.randn(256, 256, device="cuda", requires_grad=True)
    b = torch.randn(256, 256, device="cuda")
    c = (a @ b).sum()
    c.backward()  # must not break autograd
    assert a.grad is not None and torch.isfinite(a.grad).all()




def test_torch_linear_routed_zero_transpose(hip_torch):
    torch, hiptorch = hip_torch
    import torch.nn.functional as F

    x = torch.randn(512, 1024, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(768, 1024, dtype=torch.bfloat16, device="cuda")
    bias = torch.randn(768, dtype=torch.bfloat16, device="cuda")
    before = hiptorch.STATS["linear_routed"]
    y = F.linear(x, w, bias)
    assert hiptorch.STATS["linear_routed"] == before + 1, "linear not routed"
    ref = (x.double().cpu() @ w.double().cpu().T) + bias.double().cpu()
    err = (y.double().cpu() - ref).abs().max().item()
    assert err / (ref.abs().max().item() + 1e-9) < 3e-2
    # 3-D input flattens through the same path
    x3 = torch.randn(4, 128, 1024, dtype=torch.bfloat16, device="cuda")
    y3 = F.linear(x3, w)
    assert y3.shape == (4, 128, 768)
    assert hiptorch.STATS["linear_routed"] == before + 2


def test_torch_bmm_routed(hip_torch):
    torch, hiptorch = hip_torch
    a = torch.randn(6, 512, 256, dtype=torch.float32, device="cuda")
    b = torch.randn(6, 256, 384, dtype=torch.float32, device="cuda")
    before = hiptorch.STATS["bmm_routed"]
    c1 = torch.bmm(a, b)
    c2 = torch.matmul(a, b)
    assert hiptorch.STATS["bmm_routed"] == before + 2, "bmm not routed"
    ref = torch.matmul(a.double().cpu(), b.double().cpu())
    for c in (c1, c2):
        err = (c.double().cpu() - ref).abs().max().item()
        assert err / (ref.abs().max().item() + 1e-9) < 2e-4




def test_torch_bf16_nonaligned_pad_path(hip_torch):
    torch, hiptorch = hip_torch
    # 4000^3 misses the 256/128 divisibility: the routed path zero-pads
    # to the fast kernel (bit-identical by construction)
    a = torch.randn(4000, 4000, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(4000, 4000, dtype=torch.bfloat16, device="cuda")
    before = hiptorch.STATS["mm_routed"]
    c = a @ b
    assert hiptorch.STATS["mm_routed"] == before + 1
    assert c.shape == (4000, 4000)
    ref = a.double().cpu() @ b.double().cpu()
    err = (c.double().cpu() - ref).abs().max().item()
    assert err / (ref.abs().max().item() + 1e-9) < 3e-2
