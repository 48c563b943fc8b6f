"""Async wrapper over the `kubectl` CLI.

Chosen over the Python kubernetes clients for the same reasons as the
reference (kubectl.py:26-28): first-class async subprocess support and
`exec`. Behavior parity with the reference's wrapper (kubectl.py:24-193):

- attribute access maps to subcommands (``kubectl.create(...)``,
  ``kubectl.get("pod", name)``, ``kubectl.wait(...)``, ...);
- JSON-output subcommands get ``--output=json`` appended and return the
  parsed object; others return stdout as str;
- kwargs become ``--key=value`` flags (underscores -> dashes);
- ``body=...`` pipes a manifest via stdin with ``-f -``;
- nonzero exit raises RuntimeError with stderr (which the executor's
  retry layer treats as retryable);
- ``exec_raw`` returns the live subprocess for streaming use.
"""

import asyncio
import json
import logging
import shutil
from typing import Any, Optional

logger = logging.getLogger("kubectl")

JSON_OUTPUT_COMMANDS = frozenset(
    {"create", "get", "apply", "patch", "replace", "run", "expose"}
)


class Kubectl:
    def __init__(self, kubectl_bin: str = "kubectl", context: Optional[str] = None):
        self.kubectl_bin = kubectl_bin
        self.context = context

    def available(self) -> bool:
        return shutil.which(self.kubectl_bin) is not None

    def __getattr__(self, command: str):
        if command.startswith("_"):
            raise AttributeError(command)

        async def run(*args: str, body: Optional[dict] = None, **kwargs: Any):
            return await self._invoke(command, *args, body=body, **kwargs)

        run.__name__ = command
        return run

    async def exec_raw(self, *args: str) -> asyncio.subprocess.Process:
        """Start `kubectl exec ...` and return the live process."""
        argv = self._argv("exec", *args)
        logger.info("kubectl %s", " ".join(argv[1:]))
        return await asyncio.create_subprocess_exec(
            *argv,
            stdin=asyncio.subprocess.PIPE,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
        )

    def _argv(self, command: str, *args: str, **kwargs: Any) -> list:
        argv = [self.kubectl_bin]
        if self.context:
            argv += ["--context", self.context]
        argv.append(command)
        argv += list(args)
        for key, value in kwargs.items():
            if value is None:
                continue
            flag = "--" + key.replace("_", "-")
            if isinstance(value, bool):
                argv.append(f"{flag}={'true' if value else 'false'}")
            else:
                argv.append(f"{flag}={value}")
        return argv

    async def _invoke(
        self, command: str, *args: str, body: Optional[dict] = None, **kwargs: Any
    ):
        json_output = command in JSON_OUTPUT_COMMANDS
        argv = self._argv(command, *args, **kwargs)
        stdin_data = None
        if body is not None:
            argv += ["-f", "-"]
            stdin_data = json.dumps(body).encode()
        if json_output:
            argv.append("--output=json")

        logger.info("kubectl %s", " ".join(argv[1:]))
        proc = await asyncio.create_subprocess_exec(
            *argv,
            stdin=asyncio.subprocess.PIPE if stdin_data else None,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
        )
        stdout, stderr = await proc.communicate(stdin_data)
        if proc.returncode != 0:
            raise RuntimeError(
                f"kubectl {command} failed ({proc.returncode}): "
                f"{stderr.decode(errors='replace').strip()}"
            )
        if json_output:
            try:
                return json.loads(stdout)
            except json.JSONDecodeError:
                return stdout.decode(errors="replace")
        return stdout.decode(errors="replace")
