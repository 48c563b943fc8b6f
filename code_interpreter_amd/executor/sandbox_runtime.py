"""Sandbox-side runtime: import hooks, dependency auto-install, HIP numpy
routing, and user-script execution.

This is the MI355X analog of the reference's executor/sitecustomize.py:1-31
(headless artifact capture) fused with its upm+pip dependency guessing
(executor/server.rs:126-147), plus the new part: routing large numpy ops to
the gfx950 HIP kernel library (ops/hipnp.py) so user compute lands on the
MI355X matrix cores instead of host BLAS.

Imported by zygote.py in the zygote process (pre-fork, CPU-only work) and
used post-fork in each sandbox child (HIP init happens only there --
the HIP runtime must never be initialized before fork).
"""

import builtins
import importlib.util
import os
import subprocess
import sys

_original_import = builtins.__import__
_hooks_installed = False

# import-name -> pip package name, for modules whose names differ.
# The reference ships replit/upm with its full pypi import->package
# sqlite map (executor/Dockerfile:32-38); this is the equivalent
# knowledge for the import names that actually differ from their
# distribution names (identical names need no entry: the scan falls
# through to pip install <import name>).
PIP_NAME_ALIASES = {
    # imaging / media
    "cv2": "opencv-python-headless",
    "PIL": "pillow",
    "fitz": "pymupdf",
    "ffmpeg": "ffmpeg-python",
    "moviepy": "moviepy",
    "skimage": "scikit-image",
    "OpenGL": "PyOpenGL",
    "cairo": "pycairo",
    "cairosvg": "CairoSVG",
    # science / ML
    "sklearn": "scikit-learn",
    "Bio": "biopython",
    "community": "python-louvain",
    "igraph": "python-igraph",
    "Levenshtein": "python-Levenshtein",
    "speech_recognition": "SpeechRecognition",
    # parsing / documents
    "yaml": "pyyaml",
    "bs4": "beautifulsoup4",
    "docx": "python-docx",
    "pptx": "python-pptx",
    "odf": "odfpy",
    "pdfminer": "pdfminer.six",
    "PyPDF2": "pypdf2",
    "markdown": "Markdown",
    "slugify": "python-slugify",
    "magic": "python-magic",
    "ruamel": "ruamel.yaml",
    # crypto / auth
    "Crypto": "pycryptodome",
    "Cryptodome": "pycryptodomex",
    "OpenSSL": "pyopenssl",
    "jwt": "PyJWT",
    "jose": "python-jose",
    "nacl": "pynacl",
    "socks": "PySocks",
    # datetime / locale
    "dateutil": "python-dateutil",
    "babel": "Babel",
    # config / env
    "dotenv": "python-dotenv",
    # web / network / APIs
    "websocket": "websocket-client",
    "zmq": "pyzmq",
    "kafka": "kafka-python",
    "memcache": "python-memcached",
    "MySQLdb": "mysqlclient",
    "psycopg2": "psycopg2-binary",
    "github": "PyGithub",
    "gitlab": "python-gitlab",
    "googleapiclient": "google-api-python-client",
    "apiclient": "google-api-python-client",
    "telegram": "python-telegram-bot",
    "discord": "discord.py",
    "rest_framework": "djangorestframework",
    "flask_sqlalchemy": "Flask-SQLAlchemy",
    "flask_cors": "Flask-Cors",
    "flask_login": "Flask-Login",
    "flask_wtf": "Flask-WTF",
    "fake_useragent": "fake-useragent",
    "user_agents": "user-agents",
    # hardware / misc
    "serial": "pyserial",
    "usb": "pyusb",
    "git": "GitPython",
    "faker": "Faker",
    "dns": "dnspython",
    "attr": "attrs",
    "mpl_toolkits": "matplotlib",
    "pylab": "matplotlib",
}


def _env_flag(name: str, default: str = "1") -> bool:
    return os.environ.get(name, default) not in ("0", "false", "off", "")


# ---------------------------------------------------------------------------
# headless artifact capture (parity with reference sitecustomize.py:9-26)
# ---------------------------------------------------------------------------
def _patched_import(name, globals=None, locals=None, fromlist=(), level=0):
    module = _original_import(name, globals, locals, fromlist, level)

    if name == "matplotlib.pyplot":
        plt = sys.modules.get("matplotlib.pyplot")
        if plt is not None:
            plt.show = lambda *a, **k: plt.savefig("plot.png")
    elif name == "PIL":
        _original_import("PIL.ImageShow", globals, locals, fromlist, level)
        image_show = sys.modules.get("PIL.ImageShow")
        if image_show is not None:
            image_show.show = lambda img, *a, **k: img.save("image.png")
    elif name == "moviepy.editor" or name == "moviepy":
        editor = sys.modules.get("moviepy.editor") or sys.modules.get("moviepy")
        clip = getattr(editor, "VideoClip", None)
        if clip is not None and not getattr(clip, "_ci_amd_quiet", False):
            original = clip.write_videofile

            def quiet_write(self, *a, **k):
                k.setdefault("logger", None)
                return original(self, *a, **k)

            clip.write_videofile = quiet_write
            clip._ci_amd_quiet = True
    elif name == "torch" or name.startswith("torch."):
        caller = ""
        if isinstance(globals, dict):
            caller = globals.get("__name__") or ""
        if not caller.startswith("torch"):
            _maybe_install_hip_torch()
    elif name == "numpy" or name.startswith("numpy."):
        # only act on imports from OUTSIDE numpy: by the time a non-numpy
        # caller's `import numpy` returns, numpy is fully initialized
        # (numpy-internal imports fire this hook mid-init, and numpy 2.x
        # lazy-loads submodules via module __getattr__, so attribute
        # probing from inside the hook would re-enter the importer)
        caller = ""
        if isinstance(globals, dict):
            caller = globals.get("__name__") or ""
        if not caller.startswith("numpy"):
            _maybe_install_hip_numpy()

    return module


def install_import_hooks() -> None:
    global _hooks_installed
    if _hooks_installed:
        return
    builtins.__import__ = _patched_import
    _hooks_installed = True


# ---------------------------------------------------------------------------
# HIP numpy routing
# ---------------------------------------------------------------------------
_hipnp_state = {"installed": False, "attempted": False}


def _maybe_install_hip_numpy() -> None:
    """Patch numpy's hot entry points to dispatch to the HIP kernels.

    Modes (APP_HIP_NUMPY): "auto" (route when a GPU + the extension are
    available, silently stay on CPU otherwise), "require" (raise loudly if
    the HIP path is unavailable -- used on GPU boxes so a silent eager
    fallback cannot masquerade as the native path), "off".
    """
    if _hipnp_state["attempted"]:
        return
    mode = os.environ.get("APP_HIP_NUMPY", "auto").lower()
    if mode == "off":
        return
    # the import hook also fires for numpy-internal imports while numpy is
    # still initializing (e.g. `from numpy import dtypes`); installing then
    # would import a partially initialized numpy into hipnp
    np_module = sys.modules.get("numpy")
    # __dict__ lookups only (hasattr would trigger numpy's lazy submodule
    # __getattr__ re-entrantly). "test" is assigned at the tail of
    # numpy/__init__.py, so its presence means numpy finished initializing
    # (C extensions calling import_array() re-enter the import hook with
    # globals=None mid-init, which the caller filter cannot catch).
    if np_module is None or not (
        "ndarray" in np_module.__dict__ and "test" in np_module.__dict__
    ):
        return
    _hipnp_state["attempted"] = True
    ops_dir = os.environ.get("APP_OPS_DIR")
    if ops_dir and ops_dir not in sys.path:
        sys.path.insert(0, ops_dir)
    try:
        import hipnp

        hipnp.install(sys.modules["numpy"], mode=mode)
        _hipnp_state["installed"] = True
    except Exception:
        if mode == "require":
            raise
        # auto mode: CPU numpy is the documented fallback


_hiptorch_state = {"attempted": False, "installed": False}


def _maybe_install_hip_torch() -> None:
    """Route the torch matmul family to the hand-written MFMA kernels
    (ops/hiptorch.py) once the user's `import torch` completes.

    Modes (APP_HIP_TORCH): "auto" (route when a GPU + the extension are
    available), "require" (raise if the HIP path is unusable), "off".
    """
    if _hiptorch_state["attempted"]:
        return
    mode = os.environ.get("APP_HIP_TORCH", "auto").lower()
    if mode == "off":
        return
    torch_module = sys.modules.get("torch")
    # same late-init guard as numpy: torch-internal imports fire this hook
    # mid-init; "classes" is assigned near the tail of torch/__init__.py
    if torch_module is None or not (
        "Tensor" in torch_module.__dict__ and "classes" in torch_module.__dict__
    ):
        return
    _hiptorch_state["attempted"] = True
    ops_dir = os.environ.get("APP_OPS_DIR")
    if ops_dir and ops_dir not in sys.path:
        sys.path.insert(0, ops_dir)
    try:
        import hiptorch

        _hiptorch_state["installed"] = hiptorch.install(mode=mode)
    except Exception:
        if mode == "require":
            raise


def preload() -> None:
    """Called in the ZYGOTE before any fork: import everything heavy
    (numpy, the hooks) so children get it copy-on-write.

    The HIP stack is deliberately NOT touched here -- not as an
    extension import and not as a bare library load. Measured facts
    (r01 + scripts/torch_case.py isolation this round):
    - r01 preloaded _hipops in the zygote to avoid per-child ~0.3-CPU-s
      HIP userspace reloads bursting the container's CFS quota;
    - but ANY libamdhip64 instance loaded pre-fork cannot launch kernels
      post-fork (segfault at first launch). torch survives because it
      always loads its own bundled runtime post-fork; our extensions
      bind by soname to the pre-fork instance and die.
    The resolution is in ops/hipnp.py (_load_hipops): daemon-RPC
    children never import _hipops at all (no mapping cost, beating the
    r01 COW trick), and processes that really launch local kernels
    (LocalBackend, hiptorch's _hipgemm) load the runtime fresh in
    themselves, post-fork.
    """
    install_import_hooks()
    try:
        import numpy  # noqa: F401
    except ImportError:
        return
    ops_dir = os.environ.get("APP_OPS_DIR")
    if ops_dir and ops_dir not in sys.path:
        sys.path.insert(0, ops_dir)


def prewarm() -> None:
    """Expensive init in the pre-forked warm child, before any request:
    import numpy, install hooks, and bring up the HIP runtime (device
    context + pinned staging buffers) so request latency excludes it."""
    install_import_hooks()
    try:
        import numpy  # noqa: F401
    except ImportError:
        return
    _maybe_install_hip_numpy()
    if _hipnp_state["installed"]:
        try:
            import hipnp

            hipnp.warmup()
        except Exception:
            if os.environ.get("APP_HIP_NUMPY", "auto").lower() == "require":
                raise


# ---------------------------------------------------------------------------
# dependency auto-install (reference: upm guess + pip, server.rs:126-147)
# ---------------------------------------------------------------------------
def scan_missing_imports(source: str) -> list:
    """AST scan of top-level imported module names; returns pip requirement
    names for those not importable in this environment."""
    import ast

    try:
        tree = ast.parse(source)
    except SyntaxError:
        return []
    roots = set()
    for node in ast.walk(tree):
        if isinstance(node, ast.Import):
            for alias in node.names:
                roots.add(alias.name.split(".")[0])
        elif isinstance(node, ast.ImportFrom):
            if node.level == 0 and node.module:
                roots.add(node.module.split(".")[0])
    stdlib = getattr(sys, "stdlib_module_names", frozenset())
    missing = []
    for root in sorted(roots):
        if not root or root in stdlib or root in sys.modules:
            continue
        try:
            spec = importlib.util.find_spec(root)
        except (ImportError, ValueError):
            spec = None
        if spec is None:
            missing.append(PIP_NAME_ALIASES.get(root, root))
    return missing


def install_missing_deps(source: str) -> None:
    if not _env_flag("APP_DEP_INSTALL"):
        return
    missing = scan_missing_imports(source)
    if not missing:
        return
    cmd = [sys.executable, "-m", "pip", "install", "--no-cache-dir"]
    extra = os.environ.get("APP_PIP_EXTRA_ARGS", "")
    if extra:
        cmd += extra.split()
    cmd += missing
    # Failures are non-fatal (reference parity: pip's exit status is
    # ignored, server.rs:140-147); the user script will raise ImportError.
    try:
        subprocess.run(
            cmd,
            stdout=subprocess.DEVNULL,
            stderr=subprocess.DEVNULL,
            timeout=120,
        )
        importlib.invalidate_caches()
    except Exception:
        pass


# ---------------------------------------------------------------------------
# shell escapes: `!cmd` lines (the reference runs scripts under xonsh,
# server.rs:152-165, whose headline capability is shell escapes; full
# xonsh is a Python-superset shell we deliberately do not ship -- see
# MIGRATION.md -- but `!cmd` lines are restored here)
# ---------------------------------------------------------------------------
import re as _re

_SHELL_LINE = _re.compile(r"^(\s*)!(?!=)(.+)$")
# $(cmd) capture expressions (xonsh returns the command's stdout as a
# string); conservative: no nested parens/newlines inside
_SHELL_CAPTURE = _re.compile(r"\$\(([^()\n]+)\)")


def transform_shell_escapes(source: str):
    """Rewrite xonsh-style shell escapes: `<indent>!cmd` lines into
    `<indent>__ci_shell__('cmd')`, and `$(cmd)` expressions into
    `__ci_shell_capture('cmd')` (returning the command's stdout like
    xonsh). Returns the transformed source, or None if nothing matched.
    Only applied when the ORIGINAL source fails to compile as python
    (neither form is valid python syntax), so pure-python scripts --
    including ones with strings containing `!` lines or `$(...)` -- are
    never touched; and the transformed source must itself compile or
    the user sees the original SyntaxError."""
    changed = False
    out = []
    for line in source.splitlines():
        m = _SHELL_LINE.match(line)
        if m:
            indent, cmd = m.group(1), m.group(2).strip()
            out.append(f"{indent}__ci_shell__({cmd!r})")
            changed = True
        else:
            new_line, n = _SHELL_CAPTURE.subn(
                lambda mm: f"__ci_shell_capture__({mm.group(1).strip()!r})",
                line,
            )
            if n:
                changed = True
                out.append(new_line)
            else:
                out.append(line)
    return "\n".join(out) + "\n" if changed else None


def _ci_shell_capture(cmd: str) -> str:
    """$(cmd): run and return stdout as a string (xonsh capture
    semantics; stderr passes through)."""
    import subprocess

    sys.stdout.flush()
    sys.stderr.flush()
    r = subprocess.run(cmd, shell=True, capture_output=True, text=True)
    if r.stderr:
        sys.stderr.write(r.stderr)
    return r.stdout


def _ci_shell(cmd: str) -> int:
    """Run one shell escape: inherit stdout/stderr, return the exit code
    (xonsh parity: a failing command does not abort the script)."""
    import subprocess

    sys.stdout.flush()
    sys.stderr.flush()
    return subprocess.run(cmd, shell=True).returncode


# ---------------------------------------------------------------------------
# user script execution
# ---------------------------------------------------------------------------
def _apply_sandbox_rlimits() -> None:
    """Resource hygiene for the single-use sandbox process (the reference
    delegates this to the pod boundary; the local engine applies process
    rlimits): no core dumps, bounded file writes, and a CPU-time belt in
    case the wall-clock timeout is ever missed. All env-tunable; a value
    of 0 disables the limit."""
    try:
        import resource
    except ImportError:
        return

    def _set(limit, value):
        try:
            soft, hard = resource.getrlimit(limit)
            cap = value if hard == resource.RLIM_INFINITY else min(value, hard)
            resource.setrlimit(limit, (cap, hard))
        except (ValueError, OSError):
            pass

    _set(resource.RLIMIT_CORE, 0)
    fsize_mb = int(os.environ.get("APP_SANDBOX_FSIZE_MB", "4096"))
    if fsize_mb > 0:
        _set(resource.RLIMIT_FSIZE, fsize_mb << 20)
    cpu_s = int(os.environ.get("APP_SANDBOX_CPU_SECONDS", "300"))
    if cpu_s > 0:
        _set(resource.RLIMIT_CPU, cpu_s)


def run_user_script(script_path: str) -> int:
    """Run the user's script the way `python script.py` would: fresh
    __main__ globals, argv[0] = script, tracebacks to stderr, exit code 0/1
    (or SystemExit's code). Returns the exit code."""
    import time as _time

    _t0 = _time.perf_counter()
    _apply_sandbox_rlimits()
    install_import_hooks()

    with open(script_path, "r", encoding="utf-8", errors="replace") as f:
        source = f.read()

    install_missing_deps(source)

    # forked children must not share the parent's RNG stream
    if "numpy" in sys.modules:
        try:
            sys.modules["numpy"].random.seed()
        except Exception:
            pass

    _t1 = _time.perf_counter()
    sys.argv = [script_path]
    script_globals = {
        "__name__": "__main__",
        "__file__": script_path,
        "__builtins__": builtins,
        "__ci_shell__": _ci_shell,
        "__ci_shell_capture__": _ci_shell_capture,
    }
    try:
        code = compile(source, script_path, "exec")
    except SyntaxError:
        # `!cmd` shell escapes are invalid python: retry with the
        # transform (APP_SHELL_ESCAPES=0 disables)
        code = None
        if _env_flag("APP_SHELL_ESCAPES"):
            transformed = transform_shell_escapes(source)
            if transformed is not None:
                try:
                    code = compile(transformed, script_path, "exec")
                except SyntaxError:
                    code = None
        if code is None:
            import traceback

            traceback.print_exc(limit=0)
            return 1
    try:
        exec(code, script_globals)
    except SystemExit as e:
        if e.code is None:
            return 0
        return e.code if isinstance(e.code, int) else 1
    except BaseException as e:
        if type(e).__name__ == "GpuBackendLost":
            # infrastructure failure, not a user bug: exit 113 so the
            # control plane retries the whole execution in a fresh sandbox
            print("gpu backend lost; execution will be retried", file=sys.stderr)
            return 113
        import traceback

        etype, exc, tb = sys.exc_info()
        tb = tb.tb_next  # hide the runner frame: match `python script.py`
        traceback.print_exception(etype, exc, tb)
        return 1
    finally:
        _write_child_timings(_t0, _t1, _time.perf_counter())
        try:
            sys.stdout.flush()
            sys.stderr.flush()
        except Exception:
            pass
    return 0


def _write_child_timings(t0: float, t1: float, t2: float) -> None:
    path = os.environ.get("SANDBOX_TIMING_FILE")
    if not path:
        return
    try:
        rpc_ms, rpc_n = 0.0, 0
        hipnp = sys.modules.get("hipnp")
        per_op = {}
        if hipnp is not None:
            stats = getattr(hipnp, "RPC_STATS", None)
            if stats:
                rpc_ms, rpc_n = stats.get("ms", 0.0), stats.get("n", 0)
                per_op = stats.get("per_op", {})
        import json

        with open(path, "w") as f:
            json.dump(
                {
                    "child_setup_ms": round((t1 - t0) * 1000, 2),
                    "child_exec_ms": round((t2 - t1) * 1000, 2),
                    "gpu_rpc_ms": round(rpc_ms, 2),
                    "gpu_rpc_n": rpc_n,
                    **{f"rpc_{k}": v for k, v in per_op.items()},
                },
                f,
            )
    except Exception:
        pass
