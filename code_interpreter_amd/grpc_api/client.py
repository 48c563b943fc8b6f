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
