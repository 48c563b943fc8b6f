"""Unit tests for the custom-tool parser (no sandbox needed)."""

import pytest

from code_interpreter_amd.services.custom_tool_executor import (
    CustomToolExecutor,
    CustomToolParseError,
    parse_docstring,
)


@pytest.fixture
def parser():
    return CustomToolExecutor(code_executor=None)


def test_parse_minimal(parser):
    tool = parser.parse("def f(x: int) -> int:\n    return x")
    assert tool.name == "f"
    assert tool.description == "Returns: int"
    assert tool.input_schema["properties"] == {"x": {"type": "integer"}}
    assert tool.input_schema["required"] == ["x"]


def test_parse_no_return_annotation(parser):
    tool = parser.parse("def f(x: str):\n    return x")
    assert tool.description == ""
    assert tool.input_schema["properties"] == {"x": {"type": "string"}}


def test_defaults_not_required(parser):
    tool = parser.parse(
        "def f(a: int, b: str = 'x', *, c: float, d: bool = True):\n    return a"
    )
    assert tool.input_schema["required"] == ["a", "c"]


def test_syntax_error(parser):
    with pytest.raises(CustomToolParseError) as ei:
        parser.parse("def f(:")
    assert ei.value.errors[0].startswith("Syntax error:")


def test_not_single_function(parser):
    for bad in ("x = 1", "def f(): pass\ndef g(): pass", "import os"):
        with pytest.raises(CustomToolParseError) as ei:
            parser.parse(bad)
        assert ei.value.errors == [
            "The tool source code must only define a single function, optionally preceded by imports."
        ]


def test_unsafe_annotation_rejected(parser):
    with pytest.raises(CustomToolParseError) as ei:
        parser.parse("def f(x: __import__('os').system('true')) -> int:\n    return 1")
    assert "Invalid type annotation" in ei.value.errors[0]


def test_unsafe_import_not_in_namespace(parser):
    # os is not in the allowed module set; evaluating the annotation fails
    with pytest.raises(CustomToolParseError) as ei:
        parser.parse("import os\ndef f(x: os.PathLike) -> int:\n    return 1")
    assert "Error when parsing type" in ei.value.errors[0]


def test_allowed_module_alias(parser):
    tool = parser.parse(
        "import datetime as dt\ndef f(x: dt.date) -> str:\n    return str(x)"
    )
    assert tool.input_schema["properties"]["x"] == {
        "type": "string",
        "format": "date",
    }


def test_pathlib_annotation(parser):
    tool = parser.parse(
        "from pathlib import Path\ndef f(p: Path) -> str:\n    return str(p)"
    )
    assert tool.input_schema["properties"]["p"]["format"] == "path"


def test_docstring_parser_directives():
    desc, ret, params = parse_docstring(
        """
        Summary line.
        More summary.

        :param alpha: first thing
        continued line
        :return: the answer
        :param beta: second thing
        :raises ValueError: ignored directive
        """
    )
    assert desc == "Summary line.\nMore summary."
    assert params == {
        "alpha": "first thing\ncontinued line",
        "beta": "second thing",
    }
    assert ret == "the answer"


def test_docstring_colon_in_text_not_directive():
    desc, ret, params = parse_docstring("Uses key: value pairs.\nStill summary.")
    assert desc == "Uses key: value pairs.\nStill summary."
    assert params == {}
    assert ret == ""


def test_docstring_empty():
    assert parse_docstring("") == ("", "", {})


def test_star_import_of_safe_module(parser):
    tool = parser.parse(
        "from typing import *\n"
        "def f(x: List[int]) -> int:\n"
        "    return len(x)\n"
    )
    assert tool.input_schema["properties"]["x"]["type"] == "array"
