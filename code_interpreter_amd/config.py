"""Service configuration.

All fields are overridable via APP_-prefixed environment variables
(reference parity: config.py:18-80). New MI355X-specific knobs are grouped
at the bottom: executor backend selection, per-GPU warm pools, HIP-numpy
routing, and the in-pod zygote.
"""

from typing import Optional

from code_interpreter_amd.utils.envsettings import EnvSettings

DEFAULT_LOGGING_CONFIG: dict = {
    "version": 1,
    "disable_existing_loggers": False,
    "handlers": {
        "console": {
            "class": "logging.StreamHandler",
            "formatter": "standard",
        },
    },
    "formatters": {
        "standard": {
            "format": "[%(levelname)s] [%(request_id)s] %(name)s: %(message)s",
        },
    },
    "root": {
        "level": "WARNING",
        "handlers": ["console"],
        "propagate": True,
    },
    "loggers": {
        "kubectl": {"level": "INFO"},
        "grpc_server": {"level": "INFO"},
        "code_interpreter_servicer": {"level": "INFO"},
        "code_executor": {"level": "INFO"},
    },
}


class Config(EnvSettings):
    logging_config: dict = DEFAULT_LOGGING_CONFIG

    # listen addresses (reference config.py:50-53)
    grpc_listen_addr: str = "0.0.0.0:50051"
    http_listen_addr: str = "0.0.0.0:50081"

    # optional gRPC TLS material: certificate / key / CA cert file CONTENTS
    grpc_tls_cert: Optional[bytes] = None
    grpc_tls_cert_key: Optional[bytes] = None
    grpc_tls_ca_cert: Optional[bytes] = None

    # object storage for workspace files ({path: hash} round-trips)
    file_storage_path: str = "./.tmp/files"

    # ---- executor backend -------------------------------------------------
    # "local": single-use local executor-server processes from a warm pool
    #          (dev mode + single-node GPU serving; no cluster needed)
    # "kubernetes": single-use executor pods managed via kubectl
    executor_backend: str = "local"

    # user-code execution timeout, seconds (reference server.rs:151)
    execute_timeout: float = 60.0

    # ---- kubernetes backend (reference config.py:65-80) -------------------
    executor_image: str = "localhost/code-interpreter-amd-executor:local"
    executor_container_resources: dict = {}
    executor_pod_spec_extra: dict = {}
    executor_pod_queue_target_length: int = 5
    executor_pod_name_prefix: str = "code-executor-"

    # ---- local backend ----------------------------------------------------
    # warm pool of pre-forked single-use interpreters, per engine
    executor_pool_target_length: int = 2
    # engines (executor-server + zygote + GPU daemon) per GPU: raises the
    # sandbox-management parallelism of one device
    engines_per_gpu: int = 1
    # backpressure: queue (not reject) requests beyond this many in-flight
    # sandboxes per engine; 0 disables
    max_inflight_per_engine: int = 8
    # root dir for per-executor workspaces + unix sockets (tmpdir if empty)
    executor_root: str = ""
    # path to the executor-server binary ("" = bundled build)
    executor_server_bin: str = ""

    # ---- GPU scheduling ---------------------------------------------------
    # number of MI355X devices to fan executor pools across.
    # -1 = autodetect (rocm-smi / torch), 0 = CPU-only executors.
    gpu_count: int = -1
    # pin each executor to one GPU via HIP_VISIBLE_DEVICES (local backend)
    # / amd.com/gpu resource + env (kubernetes backend)
    gpu_pinning: bool = True

    # ---- sandbox runtime --------------------------------------------------
    # route large numpy ops to the gfx950 HIP kernels inside the sandbox:
    # "auto" (use when a GPU is visible), "require" (fail loudly if the HIP
    # extension is unavailable), "off"
    hip_numpy: str = "auto"
    # pre-forked interpreter pool: per-request cost is fork(), not cold start
    zygote_enabled: bool = True
    # changed-file scan: reference parity is a non-recursive /workspace scan
    # (server.rs:98-118); recursive scan is the fixed behavior behind a flag
    scan_recursive: bool = False
    # auto-install of missing imports before execution (reference upm+pip,
    # server.rs:126-147); extra args let air-gapped deploys point pip at a
    # wheelhouse, e.g. "--no-index --find-links /wheels"
    dep_install: bool = True
    pip_extra_args: str = ""
