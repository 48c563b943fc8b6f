"""CodeInterpreterService servicer (async).

Behavior parity with the reference servicer
(grpc_servicers/code_interpreter_servicer.py:44-135):
- requests validated (file map patterns) -> INVALID_ARGUMENT;
- Execute does NOT forward env (the gRPC API has no env field -- quirk
  preserved);
- tool RPCs return oneof success/error instead of raising.
"""

import json
import logging
import re
import uuid
from contextvars import ContextVar

import grpc

from code_interpreter_amd.grpc_api import descriptors as pb
from code_interpreter_amd.services.custom_tool_executor import (
    CustomToolExecuteError,
    CustomToolExecutor,
    CustomToolParseError,
)
from code_interpreter_amd.utils.validation import (
    ABSOLUTE_PATH_PATTERN,
    HASH_PATTERN,
)

logger = logging.getLogger("code_interpreter_servicer")

_HASH_RE = re.compile(HASH_PATTERN)
_PATH_RE = re.compile(ABSOLUTE_PATH_PATTERN)


class CodeInterpreterServicer:
    def __init__(
        self,
        code_executor,
        custom_tool_executor: CustomToolExecutor,
        request_id_context_var: ContextVar,
    ):
        self.code_executor = code_executor
        self.custom_tool_executor = custom_tool_executor
        self.request_id_context_var = request_id_context_var

    async def _validate_files(self, files, context) -> None:
        for path, object_hash in files.items():
            if not _PATH_RE.match(path) or not _HASH_RE.match(object_hash):
                await context.abort(
                    grpc.StatusCode.INVALID_ARGUMENT,
                    f"invalid file entry: {path!r}: {object_hash!r}",
                )

    async def Execute(self, request, context):
        self.request_id_context_var.set(str(uuid.uuid4()))
        logger.info("Executing code with %d files", len(request.files))
        await self._validate_files(request.files, context)
        result = await self.code_executor.execute(
            source_code=request.source_code,
            files=dict(request.files),
            # env intentionally not part of the gRPC Execute API
        )
        return pb.ExecuteResponse(
            stdout=result.stdout,
            stderr=result.stderr,
            exit_code=result.exit_code,
            files=dict(result.files),
        )

    async def ParseCustomTool(self, request, context):
        self.request_id_context_var.set(str(uuid.uuid4()))
        logger.info("Parsing custom tool")
        try:
            tool = self.custom_tool_executor.parse(
                tool_source_code=request.tool_source_code
            )
        except CustomToolParseError as e:
            logger.warning("Invalid custom tool: %s", e.errors)
            return pb.ParseCustomToolResponse(
                error=pb.ParseCustomToolError(error_messages=e.errors)
            )
        return pb.ParseCustomToolResponse(
            success=pb.ParseCustomToolSuccess(
                tool_name=tool.name,
                tool_input_schema_json=json.dumps(tool.input_schema),
                tool_description=tool.description,
            )
        )

    async def ExecuteCustomTool(self, request, context):
        self.request_id_context_var.set(str(uuid.uuid4()))
        logger.info("Executing custom tool")
        try:
            result = await self.custom_tool_executor.execute(
                tool_source_code=request.tool_source_code,
                tool_input_json=request.tool_input_json,
            )
        except CustomToolExecuteError as e:
            logger.warning("Error executing custom tool: %s", e)
            return pb.ExecuteCustomToolResponse(
                error=pb.ExecuteCustomToolError(stderr=e.stderr)
            )
        return pb.ExecuteCustomToolResponse(
            success=pb.ExecuteCustomToolSuccess(tool_output_json=json.dumps(result))
        )

    def method_handlers(self) -> dict:
        return {
            "Execute": grpc.unary_unary_rpc_method_handler(
                self.Execute,
                request_deserializer=pb.ExecuteRequest.FromString,
                response_serializer=pb.ExecuteResponse.SerializeToString,
            ),
            "ParseCustomTool": grpc.unary_unary_rpc_method_handler(
                self.ParseCustomTool,
                request_deserializer=pb.ParseCustomToolRequest.FromString,
                response_serializer=pb.ParseCustomToolResponse.SerializeToString,
            ),
            "ExecuteCustomTool": grpc.unary_unary_rpc_method_handler(
                self.ExecuteCustomTool,
                request_deserializer=pb.ExecuteCustomToolRequest.FromString,
                response_serializer=pb.ExecuteCustomToolResponse.SerializeToString,
            ),
        }
