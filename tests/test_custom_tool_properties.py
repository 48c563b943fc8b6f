"""Property-based tests (hypothesis) for the custom-tool parser: for any
generated signature in the supported type algebra the parser must produce
a draft-07 schema whose required-set and property names match the
signature, and for arbitrary junk input it must raise CustomToolParseError
(never crash with anything else). Reference behavior under test:
custom_tool_executor.py parse (reference services/custom_tool_executor.py:48-155)."""

import json
import keyword

import pytest
from hypothesis import HealthCheck, given, settings, strategies as st

from code_interpreter_amd.services.custom_tool_executor import (
    CustomToolExecutor,
    CustomToolParseError,
)


@pytest.fixture(scope="module")
def parser():
    return CustomToolExecutor(code_executor=None)


TYPES = st.recursive(
    st.sampled_from(
        ["int", "float", "str", "bool", "bytes", "datetime.datetime", "pathlib.Path"]
    ),
    lambda inner: st.one_of(
        inner.map(lambda t: f"typing.List[{t}]"),
        inner.map(lambda t: f"typing.Optional[{t}]"),
        st.tuples(inner, inner).map(lambda p: f"typing.Dict[str, {p[1]}]"),
        st.tuples(inner, inner).map(lambda p: f"typing.Union[{p[0]}, {p[1]}]"),
        st.tuples(inner, inner).map(lambda p: f"typing.Tuple[{p[0]}, {p[1]}]"),
    ),
    max_leaves=4,
)

NAMES = st.from_regex(r"[a-z][a-z0-9_]{0,10}", fullmatch=True).filter(
    lambda s: not keyword.iskeyword(s)
)

DEFAULTS = {"int": "0", "float": "1.5", "str": "'x'", "bool": "True"}


@st.composite
def signatures(draw):
    n = draw(st.integers(min_value=0, max_value=5))
    names = draw(
        st.lists(NAMES, min_size=n, max_size=n, unique=True)
    )
    params = []
    required = []
    seen_default = False
    for name in names:
        t = draw(TYPES)
        # positional params after one with a default must also default
        give_default = seen_default or (t in DEFAULTS and draw(st.booleans()))
        if give_default and t in DEFAULTS:
            params.append(f"{name}: {t} = {DEFAULTS[t]}")
            seen_default = True
        elif not seen_default:
            params.append(f"{name}: {t}")
            required.append(name)
        else:
            # cannot legally omit the default here: give it None via Optional
            params.append(f"{name}: typing.Optional[{t}] = None")
            seen_default = True
    return names, required, ", ".join(params)


@given(signatures())
@settings(max_examples=60, deadline=None, suppress_health_check=[HealthCheck.too_slow])
def test_generated_signatures_parse(parser, sig):
    names, required, params = sig
    src = (
        "import typing\nimport datetime\nimport pathlib\n"
        f"def tool({params}) -> str:\n"
        '    """Does a thing."""\n'
        "    return 'ok'\n"
    )
    tool = parser.parse(src)
    assert tool.name == "tool"
    schema = tool.input_schema
    assert schema.get("$schema", "").endswith("draft-07/schema#")
    assert sorted(schema.get("properties", {})) == sorted(names)
    assert sorted(schema.get("required", [])) == sorted(required)
    # draft-07: no prefixItems anywhere (tuples must be rewritten)
    assert "prefixItems" not in json.dumps(schema)


@given(st.text(max_size=200))
@settings(max_examples=100, deadline=None)
def test_junk_source_never_crashes(parser, src):
    try:
        parser.parse(src)
    except CustomToolParseError:
        pass  # the only acceptable failure mode


@given(NAMES, st.sampled_from(["*args", "**kwargs", "x, /"]))
@settings(max_examples=20, deadline=None)
def test_unsupported_signatures_rejected(parser, name, bad):
    src = f"def {name}({bad}) -> int:\n    return 1\n"
    with pytest.raises(CustomToolParseError):
        parser.parse(src)


WORDS = st.from_regex(r"[a-z][a-z ]{0,24}[a-z]|[a-z]", fullmatch=True)


@given(
    st.one_of(st.just(""), WORDS),
    st.lists(
        st.tuples(st.from_regex(r"[a-z][a-z_]{0,8}", fullmatch=True), WORDS),
        max_size=4,
        unique_by=lambda t: t[0],
    ),
    st.one_of(st.none(), WORDS),
)
@settings(max_examples=60, deadline=None)
def test_docstring_roundtrip(description, params, ret):
    """Constructed ReST docstrings parse back into their parts."""
    from code_interpreter_amd.services.custom_tool_executor import parse_docstring

    lines = [description]
    for name, desc in params:
        lines.append(f":param {name}: {desc}")
    if ret is not None:
        lines.append(f":return: {ret}")
    parsed_desc, parsed_ret, parsed_params = parse_docstring("\n".join(lines))
    assert parsed_desc == description.strip()
    for name, desc in params:
        assert parsed_params.get(name) == desc.strip(), (name, parsed_params)
    if ret is not None:
        assert parsed_ret == ret.strip()
