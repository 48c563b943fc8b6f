"""Minimal env-var settings loader (pydantic-settings is not in this image).

Provides EnvSettings: a pydantic BaseModel whose fields are filled from
``<prefix><FIELD_UPPER>`` environment variables when not passed explicitly.
Non-string fields are parsed as JSON when possible, falling back to the raw
string (so APP_EXECUTOR_POD_QUEUE_TARGET_LENGTH=5 and
APP_EXECUTOR_POD_SPEC_EXTRA='{"nodeSelector": {...}}' both work).

Behavior parity target: pydantic-settings BaseSettings with
``env_prefix="APP_", env_ignore_empty=True`` as used by the reference's
config.py:18-19.
"""

import json
import os
from typing import Any, ClassVar

from pydantic import BaseModel


class EnvSettings(BaseModel):
    ENV_PREFIX: ClassVar[str] = "APP_"

    def __init__(self, **values: Any):
        merged = dict(self._env_values())
        merged.update(values)
        super().__init__(**merged)

    @classmethod
    def _env_values(cls) -> dict[str, Any]:
        out: dict[str, Any] = {}
        for name, field in cls.model_fields.items():
            raw = os.environ.get(cls.ENV_PREFIX + name.upper())
            if raw is None or raw == "":  # env_ignore_empty semantics
                continue
            out[name] = cls._coerce(raw, field.annotation)
        return out

    @staticmethod
    def _coerce(raw: str, annotation: Any) -> Any:
        if annotation is str:
            return raw
        if annotation is bytes:
            return raw.encode()
        try:
            return json.loads(raw)
        except (json.JSONDecodeError, ValueError):
            return raw
