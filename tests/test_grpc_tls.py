"""gRPC TLS: server credentials from APP_GRPC_TLS_* material
(reference parity: application_context.py:102-110, config.py:56-62)."""

import asyncio
import subprocess
import threading

import grpc
import pytest

from code_interpreter_amd.grpc_api import descriptors as pb
from code_interpreter_amd.grpc_api.client import CodeInterpreterClient


@pytest.fixture(scope="module")
def tls_material(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("tls")
    key = tmp / "server.key"
    cert = tmp / "server.crt"
    subprocess.run(
        [
            "openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
            "-keyout", str(key), "-out", str(cert), "-days", "1",
            "-subj", "/CN=localhost",
            "-addext", "subjectAltName=DNS:localhost,IP:127.0.0.1",
        ],
        check=True,
        capture_output=True,
    )
    return cert.read_bytes(), key.read_bytes()


def test_grpc_tls_roundtrip(service, tls_material):
    cert, key = tls_material

    holder = {}
    started = threading.Event()

    def run():
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)

        async def main():
            from code_interpreter_amd.services.grpc_server import GrpcServer

            server = GrpcServer(
                code_executor=service.ctx.code_executor,
                custom_tool_executor=service.ctx.custom_tool_executor,
                request_id_context_var=service.ctx.request_id_context_var,
                tls_cert=cert,
                tls_cert_key=key,
            )
            holder["port"] = server.server.add_secure_port(
                "127.0.0.1:0", server.server_credentials
            )
            await server.server.start()
            started.set()
            holder["stop"] = asyncio.Event()
            await holder["stop"].wait()
            await server.server.stop(grace=1)

        holder["loop"] = loop
        loop.run_until_complete(main())

    thread = threading.Thread(target=run, daemon=True)
    thread.start()
    assert started.wait(15)

    creds = grpc.ssl_channel_credentials(root_certificates=cert)
    with grpc.secure_channel(f"127.0.0.1:{holder['port']}", creds) as channel:
        client = CodeInterpreterClient(channel)
        resp = client.Execute(pb.ExecuteRequest(source_code="print(6 * 7)"), timeout=120)
        assert resp.stdout == "42\n"

    holder["loop"].call_soon_threadsafe(holder["stop"].set)
    thread.join(timeout=10)


def test_insecure_client_rejected_by_tls_server(service, tls_material):
    cert, key = tls_material
    # handshake failure surfaces as RpcError on call
    creds = grpc.ssl_channel_credentials(root_certificates=None)  # system roots
    with grpc.secure_channel("127.0.0.1:1", creds) as channel:
        client = CodeInterpreterClient(channel)
        with pytest.raises(grpc.RpcError):
            client.Execute(pb.ExecuteRequest(source_code="print(1)"), timeout=2)
