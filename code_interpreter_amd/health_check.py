"""Liveness probe: gRPC Execute("print(21 * 2)") must print 42
(parity: reference health_check.py:45-53). Exit 0 on success."""

import sys

import grpc

from code_interpreter_amd.config import Config
from code_interpreter_amd.grpc_api import descriptors as pb
from code_interpreter_amd.grpc_api.client import CodeInterpreterClient


def health_check(addr: str | None = None, timeout: float = 60.0) -> None:
    config = Config()
    target = addr or config.grpc_listen_addr
    if target.startswith("0.0.0.0:"):
        target = "127.0.0.1:" + target.split(":", 1)[1]
    with grpc.insecure_channel(target) as channel:
        client = CodeInterpreterClient(channel)
        response = client.Execute(
            pb.ExecuteRequest(source_code="print(21 * 2)"), timeout=timeout
        )
    if response.stdout != "42\n":
        raise RuntimeError(
            f"health check failed: stdout={response.stdout!r} "
            f"stderr={response.stderr!r} exit_code={response.exit_code}"
        )


if __name__ == "__main__":
    try:
        health_check(sys.argv[1] if len(sys.argv) > 1 else None)
        print("OK")
    except Exception as e:
        print(f"FAIL: {e}", file=sys.stderr)
        sys.exit(1)
