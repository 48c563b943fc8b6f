"""CPU tests for the torch->HIP matmul routing layer (ops/hiptorch.py):
mode interception mechanics, eligibility gates, and install() policy are
all verifiable without a GPU (the GPU suite checks numerics against the
real kernels in tests/test_ops_gpu.py)."""

import sys
from pathlib import Path

import pytest

torch = pytest.importorskip("torch")

OPS_DIR = Path(__file__).resolve().parent.parent / "code_interpreter_amd" / "ops"
sys.path.insert(0, str(OPS_DIR))

import hiptorch  # noqa: E402


@pytest.fixture
def mode():
    m = hiptorch._make_mode(torch)
    m.__enter__()
    yield m
    m.__exit__(None, None, None)


def test_install_policy_without_gpu():
    if torch.cuda.is_available():
        pytest.skip("GPU visible: policy tested in the gpu suite")
    assert hiptorch.install(mode="auto") is False
    assert hiptorch.install(mode="off") is False
    with pytest.raises(RuntimeError):
        hiptorch.install(mode="require")


def test_dtype_codes():
    assert hiptorch._dtype_code(torch, torch.float32) == 0
    assert hiptorch._dtype_code(torch, torch.float64) == 1
    assert hiptorch._dtype_code(torch, torch.bfloat16) == 2
    assert hiptorch._dtype_code(torch, torch.int64) is None
    assert hiptorch._dtype_code(torch, torch.float16) is None


def test_try_mm_rejects_cpu_and_mismatches():
    a = torch.randn(8, 8)
    b = torch.randn(8, 8)
    assert hiptorch._try_mm(torch, a, b) is None  # cpu tensors
    assert hiptorch._try_mm(torch, a, 3) is None  # not a tensor
    assert hiptorch._try_mm(torch, a, b.to(torch.float64)) is None


def test_mode_intercepts_matmul_family(mode, monkeypatch):
    calls = []

    def fake_try_mm(t, a, b):
        calls.append((tuple(a.shape), tuple(b.shape)))
        return t.ones(a.shape[0], b.shape[1]) * 7.0

    monkeypatch.setattr(hiptorch, "_try_mm", fake_try_mm)
    a = torch.randn(4, 5)
    b = torch.randn(5, 6)
    for result in (torch.matmul(a, b), torch.mm(a, b), a @ b):
        assert float(result[0, 0]) == 7.0
    assert calls == [((4, 5), (5, 6))] * 3


def test_mode_falls_through_when_not_eligible(mode, monkeypatch):
    monkeypatch.setattr(hiptorch, "_try_mm", lambda t, a, b: None)
    a = torch.randn(4, 5)
    b = torch.randn(5, 6)
    torch.testing.assert_close(torch.matmul(a, b), a.mm(b))
    # kwargs (out=) bypass routing entirely
    out = torch.empty(4, 6)
    torch.matmul(a, b, out=out)
    torch.testing.assert_close(out, a.mm(b))
    # batched matmul is not intercepted by the 2-arg mm path
    a3 = torch.randn(2, 4, 5)
    b3 = torch.randn(2, 5, 6)
    torch.testing.assert_close(torch.matmul(a3, b3), torch.bmm(a3, b3))


def test_mode_leaves_other_ops_alone(mode):
    a = torch.randn(16, 16)
    torch.testing.assert_close(torch.add(a, a), a * 2)
    assert torch.sum(a).dim() == 0
    # linear algebra that decomposes to mm inside aten is untouched
    torch.testing.assert_close(torch.einsum("ij,jk->ik", a, a), a.mm(a))


def test_require_mode_surfaces_routing_errors(mode, monkeypatch):
    def boom(t, a, b):
        raise RuntimeError("kernel launch failed")

    monkeypatch.setattr(hiptorch, "_try_mm", boom)
    a = torch.randn(4, 4)
    # auto: swallow and fall back
    monkeypatch.delenv("APP_HIP_TORCH", raising=False)
    torch.testing.assert_close(torch.matmul(a, a), a.mm(a))
    # require: raise loudly (no silent hipBLASLt fallback on GPU boxes)
    monkeypatch.setenv("APP_HIP_TORCH", "require")
    with pytest.raises(RuntimeError, match="kernel launch failed"):
        torch.matmul(a, a)


def test_linear_and_bmm_eligibility_cpu():
    # CPU tensors: never routed, plain fallbacks intact under the mode
    m = hiptorch._make_mode(torch)
    with m:
        import torch.nn.functional as F

        x = torch.randn(8, 16)
        w = torch.randn(4, 16)
        torch.testing.assert_close(F.linear(x, w), x @ w.T)
        a = torch.randn(3, 8, 8)
        torch.testing.assert_close(torch.bmm(a, a), torch.matmul(a, a))
    assert hiptorch._try_linear(torch, x, w) is None
    assert hiptorch._try_bmm(torch, a, a) is None
