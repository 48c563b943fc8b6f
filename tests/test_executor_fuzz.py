"""Property-based fuzz of the C++ executor server's JSON/HTTP handling:
arbitrary JSON documents (and JSON-adjacent garbage) posted to /execute
must yield 200 or 400 — never a crash, never a hang (the hand-written
parser in executor/server.cpp is the riskiest native surface)."""

import json

import pytest
from hypothesis import HealthCheck, given, settings, strategies as st

from tests.test_executor_server import RawExecutor


@pytest.fixture(scope="module")
def fuzz_executor(tmp_path_factory, executor_bin):
    ex = RawExecutor(tmp_path_factory.mktemp("fuzz"), executor_bin)
    yield ex
    ex.close()


JSON_VALUES = st.recursive(
    st.one_of(
        st.none(),
        st.booleans(),
        st.integers(min_value=-(2**63), max_value=2**63),
        st.floats(allow_nan=False, allow_infinity=False),
        st.text(max_size=40),
    ),
    lambda inner: st.one_of(
        st.lists(inner, max_size=4),
        st.dictionaries(st.text(max_size=10), inner, max_size=4),
    ),
    max_leaves=8,
)


@given(JSON_VALUES)
@settings(
    max_examples=40,
    deadline=None,
    suppress_health_check=[HealthCheck.function_scoped_fixture, HealthCheck.too_slow],
)
def test_arbitrary_json_bodies(fuzz_executor, doc):
    body = json.dumps(doc).encode()
    r = fuzz_executor.client.post(
        "/execute", content=body, headers={"content-type": "application/json"}
    )
    assert r.status_code in (200, 400), r.status_code
    assert fuzz_executor.client.get("/healthz").status_code == 200


@given(st.binary(max_size=300))
@settings(
    max_examples=40,
    deadline=None,
    suppress_health_check=[HealthCheck.function_scoped_fixture, HealthCheck.too_slow],
)
def test_garbage_bodies(fuzz_executor, blob):
    r = fuzz_executor.client.post(
        "/execute", content=blob, headers={"content-type": "application/json"}
    )
    assert r.status_code in (200, 400), r.status_code
    assert fuzz_executor.client.get("/healthz").status_code == 200


@given(st.text(max_size=60))
@settings(
    max_examples=30,
    deadline=None,
    suppress_health_check=[HealthCheck.function_scoped_fixture, HealthCheck.too_slow],
)
def test_fuzzed_workspace_paths(fuzz_executor, rel):
    """Random workspace paths never escape or crash; uploads either land
    (2xx) or are rejected (4xx)."""
    import urllib.parse

    quoted = urllib.parse.quote(rel, safe="")
    r = fuzz_executor.client.put(f"/workspace/{quoted}", content=b"z")
    assert r.status_code in (204, 200, 400, 404), (rel, r.status_code)
    # nothing may appear outside the workspace root
    outside = fuzz_executor.root / "escaped"
    assert not outside.exists()
    assert fuzz_executor.client.get("/healthz").status_code == 200
