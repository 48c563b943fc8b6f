"""Synchronous client stub for CodeInterpreterService (what
CodeInterpreterServiceStub from generated code would provide)."""

import grpc

from code_interpreter_amd.grpc_api import descriptors as pb


class CodeInterpreterClient:
    def __init__(self, channel: grpc.Channel):
        prefix = f"/{pb.SERVICE_NAME}/"
        self.Execute = channel.unary_unary(
            prefix + "Execute",
            request_serializer=pb.ExecuteRequest.SerializeToString,
            response_deserializer=pb.ExecuteResponse.FromString,
        )
        self.ParseCustomTool = channel.unary_unary(
            prefix + "ParseCustomTool",
            request_serializer=pb.ParseCustomToolRequest.SerializeToString,
            response_deserializer=pb.ParseCustomToolResponse.FromString,
        )
        self.ExecuteCustomTool = channel.unary_unary(
            prefix + "ExecuteCustomTool",
            request_serializer=pb.ExecuteCustomToolRequest.SerializeToString,
            response_deserializer=pb.ExecuteCustomToolResponse.FromString,
        )


def main(argv=None) -> int:
    """Tiny CLI against a running service (the gRPC analog of curl):

        python -m code_interpreter_amd.grpc_api.client \
            --addr localhost:50051 --source 'print(21 * 2)'
        python -m code_interpreter_amd.grpc_api.client \
            --addr localhost:50051 --file script.py

    Prints stdout/stderr and exits with the execution's exit code.
    """
    import argparse
    import sys

    parser = argparse.ArgumentParser(description=main.__doc__)
    parser.add_argument("--addr", default="localhost:50051")
    parser.add_argument("--source", help="inline source code")
    parser.add_argument("--file", help="read source from a file")
    parser.add_argument(
        "--timeout", type=float, default=120.0, help="RPC timeout (s)"
    )
    args = parser.parse_args(argv)
    if bool(args.source) == bool(args.file):
        parser.error("exactly one of --source / --file is required")
    source = args.source if args.source else open(args.file).read()

    channel = grpc.insecure_channel(args.addr)
    client = CodeInterpreterClient(channel)
    resp = client.Execute(
        pb.ExecuteRequest(source_code=source), timeout=args.timeout
    )
    if resp.stdout:
        sys.stdout.write(resp.stdout)
    if resp.stderr:
        sys.stderr.write(resp.stderr)
    return resp.exit_code


if __name__ == "__main__":
    raise SystemExit(main())
