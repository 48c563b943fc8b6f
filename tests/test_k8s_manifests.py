"""Static consistency checks: every APP_* env var used in deployment
artifacts must be a real Config field, and the manifests must be valid
YAML with the expected GPU scheduling knobs."""

import re
from pathlib import Path

import yaml

from code_interpreter_amd.config import Config

REPO = Path(__file__).resolve().parent.parent


def _app_envs(text: str):
    return set(re.findall(r"APP_[A-Z0-9_]+", text))


KNOWN_NON_CONFIG = {
    # executor/sandbox-process envs (server.cpp / zygote / hipnp), not
    # control-plane Config fields
    "APP_LISTEN_ADDR", "APP_LISTEN_UNIX", "APP_WORKSPACE", "APP_PYTHON",
    "APP_RUNTIME_DIR", "APP_OPS_DIR", "APP_ZYGOTE", "APP_WARM_CHILDREN",
    "APP_SESSIONS_DIR", "APP_HIP_DAEMON", "APP_GPU_SERVICE",
    "APP_GPU_SERVICE_WAIT", "APP_HIP_NUMPY_MIN_ELEMS",
    "APP_HIP_NUMPY_MIN_MATMUL_FLOPS", "APP_MAX_BODY_BYTES",
}


def test_manifest_envs_are_config_fields():
    fields = {f"APP_{name.upper()}" for name in Config.model_fields}
    for manifest in (REPO / "k8s").glob("*.yaml"):
        for env in _app_envs(manifest.read_text()):
            assert env in fields | KNOWN_NON_CONFIG, f"{manifest.name}: {env}"


def test_manifests_parse_and_pin_gpus():
    for name in ("local.yaml", "pull.yaml"):
        docs = list(yaml.safe_load_all((REPO / "k8s" / name).read_text()))
        kinds = [d["kind"] for d in docs]
        assert {"ServiceAccount", "Role", "RoleBinding", "Pod"} <= set(kinds)
        pod = [d for d in docs if d["kind"] == "Pod"][0]
        env = {
            e["name"]: e.get("value")
            for e in pod["spec"]["containers"][0]["env"]
        }
        assert env["APP_EXECUTOR_BACKEND"] == "kubernetes"
        assert env["APP_GPU_COUNT"] == "8"


def test_readme_env_vars_exist():
    fields = {f"APP_{name.upper()}" for name in Config.model_fields}
    for doc in (REPO / "README.md", REPO / "docs" / "ARCHITECTURE.md"):
        for env in _app_envs(doc.read_text()):
            assert env in fields | KNOWN_NON_CONFIG, f"{doc.name}: {env}"
