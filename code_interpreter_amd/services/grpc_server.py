"""gRPC frontend: grpc.aio server hosting CodeInterpreterService
(parity: reference grpc_server.py:20-71; reflection is enabled when the
grpc_reflection package is present, skipped otherwise)."""

import logging
from contextvars import ContextVar
from typing import Optional

import grpc

from code_interpreter_amd.grpc_api import descriptors as pb
from code_interpreter_amd.grpc_api.servicer import CodeInterpreterServicer

logger = logging.getLogger("grpc_server")


class GrpcServer:
    def __init__(
        self,
        code_executor,
        custom_tool_executor,
        request_id_context_var: ContextVar,
        tls_cert: Optional[bytes] = None,
        tls_cert_key: Optional[bytes] = None,
        tls_ca_cert: Optional[bytes] = None,
    ):
        self.server = grpc.aio.server()
        self.servicer = CodeInterpreterServicer(
            code_executor=code_executor,
            custom_tool_executor=custom_tool_executor,
            request_id_context_var=request_id_context_var,
        )
        handler = grpc.method_handlers_generic_handler(
            pb.SERVICE_NAME, self.servicer.method_handlers()
        )
        self.server.add_generic_rpc_handlers((handler,))
        self._enable_reflection()

        self.server_credentials = None
        if tls_cert and tls_cert_key:
            self.server_credentials = grpc.ssl_server_credentials(
                private_key_certificate_chain_pairs=[(tls_cert_key, tls_cert)],
                root_certificates=tls_ca_cert,
                require_client_auth=tls_ca_cert is not None,
            )

    def _enable_reflection(self) -> None:
        try:
            from grpc_reflection.v1alpha import reflection
        except ImportError:
            logger.info("grpc_reflection not installed; reflection disabled")
            return
        reflection.enable_server_reflection(
            [pb.SERVICE_NAME, reflection.SERVICE_NAME], self.server
        )

    async def serve(self, listen_addr: str) -> None:
        if self.server_credentials is None:
            logger.info("Starting gRPC server on insecure port %s", listen_addr)
            self.server.add_insecure_port(listen_addr)
        else:
            logger.info("Starting gRPC server on secure port %s", listen_addr)
            self.server.add_secure_port(listen_addr, self.server_credentials)
        try:
            await self.server.start()
            await self.server.wait_for_termination()
        finally:
            await self.server.stop(grace=5)
