"""HTTP API behavior oracle.

Covers the same behaviors the reference's e2e suite pins
(test/e2e/test_http.py): science-stack imports, file round-trips through
the {path: hash} map, env passing, custom-tool parse golden schemas,
custom-tool execution, and error shapes.
"""

import json

import httpx
import pytest

USING_IMPORTS = """
import numpy as np
from scipy.stats import ttest_ind

np.random.seed(42)
control = np.random.normal(loc=10, scale=2, size=100)
experimental = np.random.normal(loc=12, scale=2, size=100)
t_statistic, p_value = ttest_ind(control, experimental)
print("T-Statistic:", t_statistic)
print("P-Value:", p_value)
"""


def test_imports(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/execute", json={"source_code": USING_IMPORTS, "files": {}}
    )
    assert resp.status_code == 200
    body = resp.json()
    assert body["exit_code"] == 0, body["stderr"]
    assert "P-Value" in body["stdout"]


def test_create_file_in_interpreter(http_client: httpx.Client):
    file_content = "Hello, World!"
    resp = http_client.post(
        "/v1/execute",
        json={
            "source_code": "with open('file.txt', 'w') as f:\n"
            f"    f.write({file_content!r})\n",
            "files": {},
        },
    )
    assert resp.status_code == 200
    body = resp.json()
    assert body["exit_code"] == 0
    assert body["files"].keys() == {"/workspace/file.txt"}

    resp = http_client.post(
        "/v1/execute",
        json={
            "source_code": "with open('file.txt', 'r') as f:\n    print(f.read())\n",
            "files": {"/workspace/file.txt": body["files"]["/workspace/file.txt"]},
        },
    )
    assert resp.status_code == 200
    body = resp.json()
    assert body["exit_code"] == 0
    assert body["stdout"] == file_content + "\n"
    # a pure read must not report changed files
    assert not body["files"]


def test_execute_with_env(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/execute",
        json={
            "source_code": "import os\nprint('Hello ' + os.environ['MY_NAME'])",
            "files": {},
            "env": {"MY_NAME": "John Doe"},
        },
    )
    assert resp.status_code == 200
    assert resp.json()["stdout"].strip() == "Hello John Doe"


def test_execute_error_exit_code(http_client: httpx.Client):
    resp = http_client.post("/v1/execute", json={"source_code": "1/0"})
    assert resp.status_code == 200
    body = resp.json()
    assert body["exit_code"] == 1
    assert "ZeroDivisionError" in body["stderr"]


def test_invalid_file_hash_rejected(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/execute",
        json={
            "source_code": "print(1)",
            "files": {"/workspace/a.txt": "not/a/valid/hash!"},
        },
    )
    assert resp.status_code == 422  # pydantic pattern validation


def test_parse_custom_tool_success(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/parse-custom-tool",
        json={
            "tool_source_code": '''
import typing
import typing as banana
from typing import Optional
from typing import Union as Onion

def my_tool(a: int, b: typing.Tuple[Optional[str], str] = ("hello", "world"), *, c: Onion[list[str], dict[str, banana.Optional[float]]]) -> int:
    """
    This tool is really really cool.
    Very toolish experience:
    - Toolable.
    - Toolastic.
    - Toolicious.
    :param a: something cool
    (very cool indeed)
    :param b: something nice
    :return: something great
    :param c: something awful
    """
    return 1 + 1
                '''
        },
    )
    assert resp.status_code == 200
    body = resp.json()
    assert body["tool_name"] == "my_tool"
    assert (
        body["tool_description"]
        == "This tool is really really cool.\nVery toolish experience:\n- Toolable.\n- Toolastic.\n- Toolicious.\n\nReturns: int -- something great"
    )
    assert json.loads(body["tool_input_schema_json"]) == {
        "$schema": "http://json-schema.org/draft-07/schema#",
        "type": "object",
        "title": "my_tool",
        "properties": {
            "a": {
                "type": "integer",
                "description": "something cool\n(very cool indeed)",
            },
            "b": {
                "type": "array",
                "minItems": 2,
                "items": [
                    {"anyOf": [{"type": "string"}, {"type": "null"}]},
                    {"type": "string"},
                ],
                "additionalItems": False,
                "description": "something nice",
            },
            "c": {
                "anyOf": [
                    {"type": "array", "items": {"type": "string"}},
                    {
                        "type": "object",
                        "additionalProperties": {
                            "anyOf": [{"type": "number"}, {"type": "null"}]
                        },
                    },
                ],
                "description": "something awful",
            },
        },
        "required": ["a", "c"],
        "additionalProperties": False,
    }


def test_parse_custom_tool_weather(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/parse-custom-tool",
        json={
            "tool_source_code": '''
import typing
import requests

def current_weather(lat: float, lon: float):
    """
    Get the current weather at a location.

    :param lat: A latitude.
    :param lon: A longitude.
    :return: A dictionary with the current weather.
    """
    url = "https://fake-api.com/weather?lat=" + str(lat) + "&lon=" + str(lon)
    response = requests.get(url)
    response.raise_for_status()
    return response.json()'''
        },
    )
    assert resp.status_code == 200
    body = resp.json()
    assert body["tool_name"] == "current_weather"
    assert (
        body["tool_description"]
        == "Get the current weather at a location.\n\nReturns: A dictionary with the current weather."
    )
    assert json.loads(body["tool_input_schema_json"]) == {
        "$schema": "http://json-schema.org/draft-07/schema#",
        "type": "object",
        "title": "current_weather",
        "properties": {
            "lat": {"type": "number", "description": "A latitude."},
            "lon": {"type": "number", "description": "A longitude."},
        },
        "required": ["lat", "lon"],
        "additionalProperties": False,
    }


def test_parse_custom_tool_error(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/parse-custom-tool",
        json={
            "tool_source_code": "def my_tool(a, /, b, *args, **kwargs) -> int:\n  return 1 + 1"
        },
    )
    assert resp.status_code == 400
    assert set(resp.json()["error_messages"]) == {
        "The tool function must not have positional-only arguments",
        "The tool function must not have *args",
        "The tool function must not have **kwargs",
        "The tool function arguments must have type annotations",
    }


def test_execute_custom_tool_success(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/execute-custom-tool",
        json={
            "tool_source_code": "def adding_tool(a: int, b: int) -> int:\n  return a + b",
            "tool_input_json": '{"a": 1, "b": 2}',
        },
    )
    assert resp.status_code == 200
    assert json.loads(resp.json()["tool_output_json"]) == 3


def test_execute_custom_tool_datetime_coercion(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/execute-custom-tool",
        json={
            "tool_source_code": "\nimport datetime\n\ndef date_tool(a: datetime.datetime) -> str:\n    return f\"The year is {a.year}\"\n",
            "tool_input_json": '{"a": "2000-01-01T00:00:00"}',
        },
    )
    assert resp.status_code == 200
    assert json.loads(resp.json()["tool_output_json"]) == "The year is 2000"


def test_execute_custom_tool_stdout_suppressed(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/execute-custom-tool",
        json={
            "tool_source_code": "def noisy(a: int) -> int:\n  print('SIDE EFFECT')\n  return a",
            "tool_input_json": '{"a": 5}',
        },
    )
    assert resp.status_code == 200
    assert json.loads(resp.json()["tool_output_json"]) == 5


def test_execute_custom_tool_error(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/execute-custom-tool",
        json={
            "tool_source_code": "def division_tool(a: int, b: int) -> int:\n  return a / b",
            "tool_input_json": '{"a": 0, "b": 0}',
        },
    )
    assert resp.status_code == 400
    assert "division by zero" in resp.json()["stderr"]


def test_execute_custom_tool_with_env(http_client: httpx.Client):
    resp = http_client.post(
        "/v1/execute-custom-tool",
        json={
            "tool_source_code": "import os\ndef greet() -> str:\n  return 'Hello ' + os.environ['MY_NAME']",
            "tool_input_json": "{}",
            "env": {"MY_NAME": "John Doe"},
        },
    )
    assert resp.status_code == 200
    assert json.loads(resp.json()["tool_output_json"]) == "Hello John Doe"


@pytest.mark.slow
def test_ad_hoc_install_from_wheelhouse(tmp_path_factory, executor_bin, wheelhouse):
    """On-the-fly dependency install (reference: upm+pip; ours: AST scan +
    pip) against a local wheelhouse -- no network needed."""
    from tests.conftest import ServiceUnderTest

    svc = ServiceUnderTest(
        tmp_path_factory.mktemp("whl"),
        pip_extra_args=f"--no-index --find-links {wheelhouse}",
    ).start()
    try:
        with httpx.Client(base_url=svc.base_url, timeout=180.0) as client:
            resp = client.post(
                "/v1/execute",
                json={
                    "source_code": "import mootool\nmootool.moo('Hello World')",
                    "files": {},
                },
            )
            assert resp.status_code == 200
            body = resp.json()
            assert body["exit_code"] == 0, body["stderr"]
            assert "moo says: Hello World" in body["stdout"]
    finally:
        svc.stop()


def test_metrics_endpoint(http_client):
    """/metrics (beyond-reference observability): request counters and
    latency histograms in Prometheus text format."""
    r = http_client.post("/v1/execute", json={"source_code": "print('m')"})
    assert r.status_code == 200
    m = http_client.get("/metrics")
    assert m.status_code == 200
    body = m.text
    assert "code_interpreter_requests_total" in body
    assert '/v1/execute",status="200"' in body.replace("route=", "")
    assert "code_interpreter_request_seconds_bucket" in body


def test_healthz(http_client):
    r = http_client.get("/healthz")
    assert r.status_code == 200 and r.json() == {"status": "ok"}
