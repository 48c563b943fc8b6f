"""HTTP API frontend.

Wire-contract parity with the reference (http_server.py:36-162; see
SURVEY.md section 1 for the byte-level contracts):

  POST /v1/execute             {source_code, files?, env?} ->
                               {stdout, stderr, exit_code, files}
                               errors -> 500 {detail}
  POST /v1/parse-custom-tool   {tool_source_code} ->
                               {tool_name, tool_input_schema_json,
                                tool_description}
                               parse errors -> 400 {error_messages}
  POST /v1/execute-custom-tool {tool_source_code, tool_input_json, env?} ->
                               {tool_output_json}
                               tool errors -> 400 {stderr}
"""

import json
import logging
import uuid
from contextvars import ContextVar
from typing import Dict, List

from fastapi import Depends, FastAPI, HTTPException, status
from fastapi.responses import JSONResponse
from pydantic import BaseModel

from code_interpreter_amd.services.custom_tool_executor import (
    CustomToolExecuteError,
    CustomToolExecutor,
    CustomToolParseError,
)
from code_interpreter_amd.utils.validation import AbsolutePath, Hash

logger = logging.getLogger("code_interpreter_service")


class ExecuteRequest(BaseModel):
    source_code: str
    files: Dict[AbsolutePath, Hash] = {}
    env: Dict[str, str] = {}


class ExecuteResponse(BaseModel):
    stdout: str
    stderr: str
    exit_code: int
    files: Dict[AbsolutePath, Hash]


class ParseCustomToolRequest(BaseModel):
    tool_source_code: str


class ParseCustomToolResponse(BaseModel):
    tool_name: str
    tool_input_schema_json: str
    tool_description: str


class ParseCustomToolErrorResponse(BaseModel):
    error_messages: List[str]


class ExecuteCustomToolRequest(BaseModel):
    tool_source_code: str
    tool_input_json: str
    env: Dict[str, str] = {}


class ExecuteCustomToolResponse(BaseModel):
    tool_output_json: str


class ExecuteCustomToolErrorResponse(BaseModel):
    stderr: str


def _make_metrics():
    """Prometheus counters/histograms when prometheus_client is present
    (it is in the service image); observability beyond the reference,
    which has no metrics endpoint. Returns
    (observe(route, status, seconds), response_factory|None)."""
    try:
        from prometheus_client import (
            CollectorRegistry,
            Counter,
            Histogram,
            generate_latest,
            CONTENT_TYPE_LATEST,
        )
    except ImportError:
        return (lambda route, status_code, seconds: None), None

    registry = CollectorRegistry()
    requests_total = Counter(
        "code_interpreter_requests_total",
        "API requests by route and status",
        ["route", "status"],
        registry=registry,
    )
    latency = Histogram(
        "code_interpreter_request_seconds",
        "End-to-end request latency by route",
        ["route"],
        buckets=(0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1, 2.5, 5, 10, 30, 60),
        registry=registry,
    )

    def observe(route: str, status_code: int, seconds: float) -> None:
        requests_total.labels(route=route, status=str(status_code)).inc()
        latency.labels(route=route).observe(seconds)

    def render():
        from fastapi import Response

        return Response(generate_latest(registry), media_type=CONTENT_TYPE_LATEST)

    return observe, render


def create_http_server(
    code_executor,
    custom_tool_executor: CustomToolExecutor,
    request_id_context_var: ContextVar,
) -> FastAPI:
    app = FastAPI(title="code-interpreter-amd")
    observe, metrics_response = _make_metrics()

    @app.middleware("http")
    async def record_metrics(request, call_next):
        import time

        t0 = time.perf_counter()
        response = await call_next(request)
        observe(request.url.path, response.status_code, time.perf_counter() - t0)
        return response

    if metrics_response is not None:

        @app.get("/metrics")
        async def metrics():
            return metrics_response()

    @app.get("/healthz")
    async def healthz():
        """Cheap liveness for HTTP-only probes (the reference's liveness
        is the gRPC health_check; both are wired in k8s/)."""
        return {"status": "ok"}

    def set_request_id() -> str:
        request_id = str(uuid.uuid4())
        request_id_context_var.set(request_id)
        from code_interpreter_amd.services.code_executor import REQUEST_ID

        REQUEST_ID.set(request_id)
        return request_id

    @app.post("/v1/execute", response_model=ExecuteResponse)
    async def execute(request: ExecuteRequest, request_id: str = Depends(set_request_id)):
        logger.debug(
            "execute: %d input file(s), %d source bytes",
            len(request.files), len(request.source_code),
        )
        try:
            result = await code_executor.execute(
                source_code=request.source_code,
                files=request.files,
                env=request.env,
            )
        except Exception as e:
            logger.exception("execute failed (request %s)", request_id)
            raise HTTPException(status_code=500, detail=str(e))
        logger.debug("execute done: exit=%d, %d changed file(s)",
                     result.exit_code, len(result.files))
        return ExecuteResponse(
            stdout=result.stdout,
            stderr=result.stderr,
            exit_code=result.exit_code,
            files=dict(result.files),
        )

    @app.post("/v1/parse-custom-tool", response_model=ParseCustomToolResponse)
    async def parse_custom_tool(
        request: ParseCustomToolRequest, request_id: str = Depends(set_request_id)
    ):
        logger.info("parse-custom-tool (request %s)", request_id)
        tool = custom_tool_executor.parse(tool_source_code=request.tool_source_code)
        return ParseCustomToolResponse(
            tool_name=tool.name,
            tool_input_schema_json=json.dumps(tool.input_schema),
            tool_description=tool.description,
        )

    @app.exception_handler(CustomToolParseError)
    async def parse_error_handler(request, e: CustomToolParseError):
        logger.warning("custom tool rejected by parser: %s", e.errors)
        return JSONResponse(
            status_code=status.HTTP_400_BAD_REQUEST,
            content=ParseCustomToolErrorResponse(error_messages=e.errors).model_dump(),
        )

    @app.post("/v1/execute-custom-tool", response_model=ExecuteCustomToolResponse)
    async def execute_custom_tool(
        request: ExecuteCustomToolRequest, request_id: str = Depends(set_request_id)
    ):
        logger.info("execute-custom-tool (request %s)", request_id)
        result = await custom_tool_executor.execute(
            tool_source_code=request.tool_source_code,
            tool_input_json=request.tool_input_json,
            env=request.env,
        )
        return ExecuteCustomToolResponse(tool_output_json=json.dumps(result))

    @app.exception_handler(CustomToolExecuteError)
    async def execute_error_handler(request, e: CustomToolExecuteError):
        logger.warning("custom tool execution returned nonzero: %s", e)
        return JSONResponse(
            status_code=status.HTTP_400_BAD_REQUEST,
            content=ExecuteCustomToolErrorResponse(stderr=e.stderr).model_dump(),
        )

    return app
