"""Lazy service wiring + request-scoped logging.

Parity with the reference's ApplicationContext (application_context.py:36-125):
cached-property DI, a request-id ContextVar injected into every log line,
and warm-pool prefill kicked off as a background task.
"""

import asyncio
import logging
import logging.config
from contextvars import ContextVar
from functools import cached_property

from code_interpreter_amd.config import Config


class _RequestIdFilter(logging.Filter):
    def __init__(self, request_id_context_var: ContextVar):
        super().__init__()
        self.request_id_context_var = request_id_context_var

    def filter(self, record: logging.LogRecord) -> bool:
        record.request_id = self.request_id_context_var.get()
        return True


class ApplicationContext:
    def __init__(self, config: Config | None = None):
        self.config = config or Config()
        self._setup_logging()

    def _setup_logging(self) -> None:
        logging.config.dictConfig(self.config.logging_config)
        log_filter = _RequestIdFilter(self.request_id_context_var)
        for handler in logging.getLogger().handlers:
            handler.addFilter(log_filter)

    @cached_property
    def request_id_context_var(self) -> ContextVar:
        return ContextVar("request_id", default="-")

    @cached_property
    def file_storage(self):
        from code_interpreter_amd.services.storage import Storage

        return Storage(storage_path=self.config.file_storage_path)

    @cached_property
    def code_executor(self):
        if self.config.executor_backend == "kubernetes":
            executor = self._kubernetes_executor()
        else:
            executor = self._local_executor()
        # warm-pool prefill off the startup path
        try:
            asyncio.get_running_loop().create_task(executor.fill_pool())
        except RuntimeError:
            pass  # no loop yet (tests construct the context synchronously)
        return executor

    def _local_executor(self):
        from code_interpreter_amd.services.local_executor import LocalPoolExecutor

        cfg = self.config
        return LocalPoolExecutor(
            file_storage=self.file_storage,
            pool_target_length=cfg.executor_pool_target_length,
            engines_per_gpu=cfg.engines_per_gpu,
            gpu_count=cfg.gpu_count,
            gpu_pinning=cfg.gpu_pinning,
            executor_root=cfg.executor_root,
            server_bin=cfg.executor_server_bin,
            execute_timeout=cfg.execute_timeout,
            zygote_enabled=cfg.zygote_enabled,
            scan_recursive=cfg.scan_recursive,
            dep_install=cfg.dep_install,
            pip_extra_args=cfg.pip_extra_args,
            hip_numpy=cfg.hip_numpy,
            max_inflight_per_engine=cfg.max_inflight_per_engine,
        )

    def _kubernetes_executor(self):
        from code_interpreter_amd.services.kubectl import Kubectl
        from code_interpreter_amd.services.pod_executor import PodExecutor

        cfg = self.config
        return PodExecutor(
            kubectl=Kubectl(),
            file_storage=self.file_storage,
            executor_image=cfg.executor_image,
            container_resources=cfg.executor_container_resources,
            pod_spec_extra=cfg.executor_pod_spec_extra,
            pod_queue_target_length=cfg.executor_pod_queue_target_length,
            pod_name_prefix=cfg.executor_pod_name_prefix,
            gpu_count=cfg.gpu_count,
            gpu_pinning=cfg.gpu_pinning,
        )

    @cached_property
    def custom_tool_executor(self):
        from code_interpreter_amd.services.custom_tool_executor import (
            CustomToolExecutor,
        )

        return CustomToolExecutor(code_executor=self.code_executor)

    @cached_property
    def http_server(self):
        from code_interpreter_amd.services.http_server import create_http_server

        return create_http_server(
            code_executor=self.code_executor,
            custom_tool_executor=self.custom_tool_executor,
            request_id_context_var=self.request_id_context_var,
        )

    @cached_property
    def grpc_server(self):
        from code_interpreter_amd.services.grpc_server import GrpcServer

        return GrpcServer(
            code_executor=self.code_executor,
            custom_tool_executor=self.custom_tool_executor,
            request_id_context_var=self.request_id_context_var,
            tls_cert=self.config.grpc_tls_cert,
            tls_cert_key=self.config.grpc_tls_cert_key,
            tls_ca_cert=self.config.grpc_tls_ca_cert,
        )
