"""gRPC API behavior oracle (mirrors reference test/e2e/test_grpc.py):
Execute + file round-trip, oneof success/error on the tool RPCs, and the
Execute-has-no-env quirk."""

import asyncio
import json
import threading

import grpc
import pytest

from code_interpreter_amd.grpc_api import descriptors as pb
from code_interpreter_amd.grpc_api.client import CodeInterpreterClient


class GrpcServiceUnderTest:
    """grpc.aio server running in a dedicated thread with its own loop."""

    def __init__(self, service):
        # reuse the HTTP test service's executors & storage
        self.ctx = service.ctx
        self.port = None
        self._loop = None
        self._started = threading.Event()
        self._stop = None

    def start(self):
        def run():
            self._loop = asyncio.new_event_loop()
            asyncio.set_event_loop(self._loop)
            self._loop.run_until_complete(self._main())

        self._thread = threading.Thread(target=run, daemon=True)
        self._thread.start()
        assert self._started.wait(15), "grpc server did not start"
        return self

    async def _main(self):
        from code_interpreter_amd.services.grpc_server import GrpcServer

        server = GrpcServer(
            code_executor=self.ctx.code_executor,
            custom_tool_executor=self.ctx.custom_tool_executor,
            request_id_context_var=self.ctx.request_id_context_var,
        )
        self.port = server.server.add_insecure_port("127.0.0.1:0")
        await server.server.start()
        self._started.set()
        self._stop = asyncio.Event()
        await self._stop.wait()
        await server.server.stop(grace=1)

    def stop(self):
        if self._loop and self._stop:
            self._loop.call_soon_threadsafe(self._stop.set)
        self._thread.join(timeout=10)


@pytest.fixture(scope="module")
def grpc_client(service):
    svc = GrpcServiceUnderTest(service).start()
    channel = grpc.insecure_channel(f"127.0.0.1:{svc.port}")
    yield CodeInterpreterClient(channel)
    channel.close()
    svc.stop()


def test_execute(grpc_client):
    response = grpc_client.Execute(
        pb.ExecuteRequest(source_code="print(21 * 2)"), timeout=120
    )
    assert response.stdout == "42\n"
    assert response.exit_code == 0


def test_execute_file_roundtrip(grpc_client):
    response = grpc_client.Execute(
        pb.ExecuteRequest(
            source_code="with open('file.txt', 'w') as f:\n    f.write('Hello, World!')\n"
        ),
        timeout=120,
    )
    assert response.exit_code == 0
    assert set(response.files.keys()) == {"/workspace/file.txt"}

    response = grpc_client.Execute(
        pb.ExecuteRequest(
            source_code="with open('file.txt') as f:\n    print(f.read())\n",
            files={"/workspace/file.txt": response.files["/workspace/file.txt"]},
        ),
        timeout=120,
    )
    assert response.exit_code == 0
    assert response.stdout == "Hello, World!\n"
    assert not response.files


def test_execute_invalid_file_hash(grpc_client):
    with pytest.raises(grpc.RpcError) as ei:
        grpc_client.Execute(
            pb.ExecuteRequest(
                source_code="print(1)", files={"/workspace/a": "bad hash!"}
            ),
            timeout=120,
        )
    assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT


def test_parse_custom_tool_success_oneof(grpc_client):
    response = grpc_client.ParseCustomTool(
        pb.ParseCustomToolRequest(
            tool_source_code='def my_tool(a: int) -> int:\n    """\n    Adds one.\n    :param a: the number\n    :return: the result\n    """\n    return a + 1'
        ),
        timeout=60,
    )
    assert response.WhichOneof("response") == "success"
    assert response.success.tool_name == "my_tool"
    assert response.success.tool_description == "Adds one.\n\nReturns: int -- the result"
    schema = json.loads(response.success.tool_input_schema_json)
    assert schema["properties"]["a"] == {
        "type": "integer",
        "description": "the number",
    }


def test_parse_custom_tool_error_oneof(grpc_client):
    response = grpc_client.ParseCustomTool(
        pb.ParseCustomToolRequest(
            tool_source_code="def my_tool(a, /, b, *args, **kwargs) -> int:\n  return 1 + 1"
        ),
        timeout=60,
    )
    assert response.WhichOneof("response") == "error"
    assert set(response.error.error_messages) == {
        "The tool function must not have positional-only arguments",
        "The tool function must not have *args",
        "The tool function must not have **kwargs",
        "The tool function arguments must have type annotations",
    }


def test_execute_custom_tool_success_oneof(grpc_client):
    result = grpc_client.ExecuteCustomTool(
        pb.ExecuteCustomToolRequest(
            tool_source_code="def adding_tool(a: int, b: int) -> int:\n  return a + b",
            tool_input_json='{"a": 1, "b": 2}',
        ),
        timeout=120,
    )
    assert result.WhichOneof("response") == "success"
    assert result.success.tool_output_json == "3"


def test_execute_custom_tool_error_oneof(grpc_client):
    result = grpc_client.ExecuteCustomTool(
        pb.ExecuteCustomToolRequest(
            tool_source_code="def division_tool(a: int, b: int) -> int:\n  return a / b",
            tool_input_json='{"a": 0, "b": 0}',
        ),
        timeout=120,
    )
    assert result.WhichOneof("response") == "error"
    assert "division by zero" in result.error.stderr


def test_wire_format_roundtrip():
    """Messages built from runtime descriptors serialize/parse correctly."""
    msg = pb.ExecuteRequest(source_code="x", files={"/workspace/a": "h" * 64})
    data = msg.SerializeToString()
    back = pb.ExecuteRequest.FromString(data)
    assert back.source_code == "x"
    assert dict(back.files) == {"/workspace/a": "h" * 64}


def test_cli_client_execute(service, tmp_path, capsys):
    """The curl-style CLI (python -m code_interpreter_amd.grpc_api.client)
    executes source against a live service and propagates the exit code."""
    from code_interpreter_amd.grpc_api.client import main as cli_main

    svc = GrpcServiceUnderTest(service).start()
    try:
        addr = f"127.0.0.1:{svc.port}"
        rc = cli_main(["--addr", addr, "--source", "print(6 * 7)"])
        out = capsys.readouterr()
        assert rc == 0
        assert "42" in out.out

        script = tmp_path / "s.py"
        script.write_text("import sys\nprint('boom')\nsys.exit(3)\n")
        rc = cli_main(["--addr", addr, "--file", str(script)])
        out = capsys.readouterr()
        assert rc == 3
        assert "boom" in out.out
    finally:
        svc.stop()
