"""Validated string types used across the wire API.

Wire-contract parity with the reference's utils/validation.py:19-22
(Hash and AbsolutePath regex patterns are part of the public HTTP API).
"""

from typing import Annotated

from pydantic import Field, TypeAdapter

HASH_PATTERN = r"^[0-9a-zA-Z_-]{1,255}$"
ABSOLUTE_PATH_PATTERN = r"^/[^/].*$"

Hash = Annotated[str, Field(pattern=HASH_PATTERN)]
AbsolutePath = Annotated[str, Field(pattern=ABSOLUTE_PATH_PATTERN)]

_hash_adapter = TypeAdapter(Hash)
_abs_path_adapter = TypeAdapter(AbsolutePath)


def validate_hash(value: str) -> str:
    return _hash_adapter.validate_python(value)


def validate_absolute_path(value: str) -> str:
    return _abs_path_adapter.validate_python(value)
