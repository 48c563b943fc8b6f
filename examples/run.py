"""Submit an example payload to a running service over the HTTP API.

Usage (service from `python -m code_interpreter_amd`, default :50081):

    python examples/run.py examples/fib.py
    python examples/run.py examples/hello_world_write_file.py
    python examples/run.py examples/hello_world_read_file.py \
        --file example.txt=<hash printed by the previous run>
    python examples/run.py --url http://127.0.0.1:50081 examples/crash.py

Prints stdout/stderr, the exit code, and the hash of every file the
execution created or changed (pass those back via --file to chain
executions, as the reference's file round-trip flow does).
"""

import argparse
import sys
from pathlib import Path

import httpx


def main() -> int:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("payload", help="path to a .py payload to execute")
    ap.add_argument("--url", default="http://127.0.0.1:50081")
    ap.add_argument(
        "--file",
        action="append",
        default=[],
        metavar="PATH=HASH",
        help="stage a stored object into the workspace (repeatable)",
    )
    ap.add_argument(
        "--env",
        action="append",
        default=[],
        metavar="KEY=VALUE",
        help="environment variable for the execution (repeatable)",
    )
    args = ap.parse_args()

    files = {}
    for spec in args.file:
        path, _, digest = spec.partition("=")
        files["./" + path if not path.startswith("/") else path] = digest
    env = dict(spec.partition("=")[::2] for spec in args.env)

    body = {
        "source_code": Path(args.payload).read_text(),
        "files": files,
        "env": env,
    }
    r = httpx.post(f"{args.url}/v1/execute", json=body, timeout=120.0)
    r.raise_for_status()
    out = r.json()

    if out["stdout"]:
        sys.stdout.write(out["stdout"])
    if out["stderr"]:
        sys.stderr.write(out["stderr"])
    for path, digest in sorted(out["files"].items()):
        print(f"[file] {path} = {digest}")
    print(f"[exit] {out['exit_code']}")
    return 0 if out["exit_code"] == 0 else 1


if __name__ == "__main__":
    raise SystemExit(main())
