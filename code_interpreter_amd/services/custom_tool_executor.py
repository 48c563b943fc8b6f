"""Custom-tool engine: parse one annotated Python function into a JSON-schema
callable, and execute it in the sandbox with pydantic-coerced input.

Behavior parity (asserted byte-for-byte by tests/test_custom_tool.py golden
schemas) with the reference's services/custom_tool_executor.py:48-296:

- source = imports* followed by exactly one function def;
- rejected shapes: positional-only args, *args, **kwargs, missing
  annotations (exact error strings are part of the API);
- ReST docstrings: ":param x:" / ":return:" directives, multi-line bodies;
- description = fn description + "Returns: <type> -- <return desc>";
- annotations evaluated only against builtins plus imports of
  typing/pathlib/datetime (AST-whitelisted first), then JSON-schema'd by
  pydantic with a draft-07 generator that rewrites tuple prefixItems to
  items + additionalItems: false;
- required = positional args without defaults + kw-only args without
  defaults;
- execute() synthesizes a wrapper script (imports re-emitted for dependency
  detection, user stdout suppressed, result JSON on stdout) and runs it
  through the code executor; nonzero exit raises
  CustomToolExecuteError(stderr).
"""

import ast
import inspect
import json
import re
import textwrap
import typing
from dataclasses import dataclass

import pydantic
import pydantic.json_schema

SAFE_TYPE_MODULES = frozenset({"typing", "pathlib", "datetime"})

_BUILTIN_TYPES = {
    "str": str,
    "int": int,
    "float": float,
    "bool": bool,
    "list": list,
    "dict": dict,
    "set": set,
    "tuple": tuple,
}


@dataclass
class CustomTool:
    name: str
    description: str
    input_schema: dict


@dataclass
class CustomToolParseError(Exception):
    errors: list


@dataclass
class CustomToolExecuteError(Exception):
    stderr: str

    def __str__(self) -> str:
        return self.stderr


class _Draft07Schema(pydantic.json_schema.GenerateJsonSchema):
    """Draft-07 dialect; tuples use items-array + additionalItems."""

    schema_dialect = "http://json-schema.org/draft-07/schema#"

    def tuple_schema(self, schema):
        out = super().tuple_schema(schema)
        if "prefixItems" in out:
            out["items"] = out.pop("prefixItems")
            out.pop("maxItems", None)
            out["additionalItems"] = False
        return out


def _split_source(tool_source_code: str):
    """ast-parse and split into (import nodes, function def, clean source)."""
    clean_source = textwrap.dedent(tool_source_code)
    try:
        body = ast.parse(clean_source).body
    except SyntaxError as e:
        raise CustomToolParseError([f"Syntax error: {e.msg} on line {e.lineno}"])
    except ValueError as e:
        # e.g. null bytes in the source: a client input error, not a 500
        raise CustomToolParseError([f"Invalid source: {e}"])
    if not body:
        raise CustomToolParseError(
            [
                "The tool source code must only define a single function, optionally preceded by imports."
            ]
        )
    *imports, function_def = body
    if not all(
        isinstance(n, (ast.Import, ast.ImportFrom)) for n in imports
    ) or not isinstance(function_def, ast.FunctionDef):
        raise CustomToolParseError(
            [
                "The tool source code must only define a single function, optionally preceded by imports."
            ]
        )
    return imports, function_def, clean_source


def _validate_signature(function_def: ast.FunctionDef) -> None:
    a = function_def.args
    errors = []
    if a.posonlyargs:
        errors.append("The tool function must not have positional-only arguments")
    if a.vararg:
        errors.append("The tool function must not have *args")
    if a.kwarg:
        errors.append("The tool function must not have **kwargs")
    if not all(arg.annotation for arg in (*a.args, *a.kwonlyargs)):
        errors.append("The tool function arguments must have type annotations")
    if errors:
        raise CustomToolParseError(errors)


_DIRECTIVE_SPLIT = re.compile(r"(^|\n)\s*:", flags=re.MULTILINE)
_PARAM_DIRECTIVE = re.compile(r"param ([a-z_]+): ((?:.|\n)+)", flags=re.MULTILINE)
_RETURN_DIRECTIVE = re.compile(r"return: ((?:.|\n)+)", flags=re.MULTILINE)


def parse_docstring(docstring: str):
    """ReST docstring -> (description, return description, {param: desc}).

    Directives start at a line whose first non-space char is ':'; a
    directive body runs until the next directive and keeps internal
    newlines (continuation lines).
    """
    chunks = [c.strip() for c in _DIRECTIVE_SPLIT.split(inspect.cleandoc(docstring))]
    description = chunks[0]
    params: dict[str, str] = {}
    return_description = ""
    for chunk in chunks[1:]:
        if m := _PARAM_DIRECTIVE.match(chunk):
            params[m.group(1)] = m.group(2)
        elif m := _RETURN_DIRECTIVE.match(chunk):
            return_description = m.group(1)
    return description, return_description, params


def _namespace_from_imports(imports: list) -> dict:
    """Evaluation namespace for annotations: builtin type names plus any
    imports of the SAFE_TYPE_MODULES (honoring aliases)."""
    ns: dict = dict(_BUILTIN_TYPES)
    for node in imports:
        if isinstance(node, ast.Import):
            for alias in node.names:
                if alias.name in SAFE_TYPE_MODULES:
                    ns[alias.asname or alias.name] = __import__(alias.name)
        elif isinstance(node, ast.ImportFrom):
            if node.module in SAFE_TYPE_MODULES:
                mod = __import__(node.module, fromlist=[a.name for a in node.names])
                for alias in node.names:
                    if alias.name == "*":  # star import of a safe module
                        public = getattr(mod, "__all__", None) or [
                            n for n in vars(mod) if not n.startswith("_")
                        ]
                        for n in public:
                            ns[n] = getattr(mod, n)
                    else:
                        ns[alias.asname or alias.name] = getattr(mod, alias.name)
    return ns


def _is_safe_type_ast(node: ast.AST) -> bool:
    if isinstance(node, ast.Name):
        return True
    if isinstance(node, ast.Attribute):
        return _is_safe_type_ast(node.value)
    if isinstance(node, ast.Subscript):
        return _is_safe_type_ast(node.value) and _is_safe_type_ast(node.slice)
    if isinstance(node, (ast.Tuple, ast.List)):
        return all(_is_safe_type_ast(e) for e in node.elts)
    if isinstance(node, ast.Constant):
        return isinstance(node.value, (str, int, float, bool, type(None)))
    if isinstance(node, ast.BinOp):
        return (
            isinstance(node.op, ast.BitOr)
            and _is_safe_type_ast(node.left)
            and _is_safe_type_ast(node.right)
        )
    return False


def _annotation_schema(annotation: ast.AST, namespace: dict) -> dict:
    type_str = ast.unparse(annotation)
    if not _is_safe_type_ast(annotation):
        raise CustomToolParseError([f"Invalid type annotation `{type_str}`"])
    try:
        evaluated = eval(type_str, namespace)  # whitelisted AST + namespace
        return pydantic.TypeAdapter(evaluated).json_schema(
            schema_generator=_Draft07Schema
        )
    except CustomToolParseError:
        raise
    except Exception as e:
        raise CustomToolParseError([f"Error when parsing type `{type_str}`: {e}"])


class CustomToolExecutor:
    def __init__(self, code_executor):
        self.code_executor = code_executor

    def parse(self, tool_source_code: str) -> CustomTool:
        imports, function_def, _ = _split_source(tool_source_code)
        _validate_signature(function_def)

        description, return_description, param_descriptions = parse_docstring(
            ast.get_docstring(function_def) or ""
        )
        namespace = _namespace_from_imports(imports)

        args = function_def.args
        all_args = [*args.args, *args.kwonlyargs]
        n_pos_defaults = len(args.defaults)
        required = [a.arg for a in args.args[: len(args.args) - n_pos_defaults]]
        required += [
            a.arg
            for a, default in zip(args.kwonlyargs, args.kw_defaults)
            if default is None
        ]

        properties = {}
        for arg in all_args:
            if not arg.annotation:
                continue
            prop = _annotation_schema(arg.annotation, namespace)
            if desc := param_descriptions.get(arg.arg):
                prop = {**prop, "description": desc}
            properties[arg.arg] = prop

        schema = {
            "$schema": "http://json-schema.org/draft-07/schema#",
            "type": "object",
            "title": function_def.name,
            "properties": properties,
            "required": required,
            "additionalProperties": False,
        }

        return_type = ast.unparse(function_def.returns) if function_def.returns else None
        returns_line = " -- ".join(s for s in (return_type, return_description) if s)
        full_description = "\n\n".join(
            s
            for s in (description, f"Returns: {returns_line}" if returns_line else None)
            if s
        )

        return CustomTool(
            name=function_def.name,
            description=full_description,
            input_schema=schema,
        )

    async def execute(
        self,
        tool_source_code: str,
        tool_input_json: str,
        env: typing.Mapping[str, str] = {},
    ) -> typing.Any:
        """Run the tool in the sandbox; returns the JSON-decoded result."""
        imports, function_def, clean_source = _split_source(tool_source_code)
        import_lines = "\n".join(
            ast.unparse(n)
            for n in imports
            if isinstance(n, (ast.Import, ast.ImportFrom))
        )
        # Imports re-emitted at top level so the executor's dependency
        # auto-install sees them. stdout is suppressed at the FD level
        # while the tool runs (python-level redirect misses native
        # libraries -- e.g. RCCL prints a banner straight to fd 1), so
        # stdout is exactly one JSON document. The __main__ guard keeps
        # multiprocessing-spawn children (multi-GPU tools) from re-running
        # the wrapper when they re-import it as __mp_main__.
        script = f"""# tool dependency imports (re-emitted for dependency detection)
{import_lines}

import json
import os
import sys

import pydantic

if __name__ == "__main__":
    saved_stdout_fd = os.dup(1)
    devnull_fd = os.open(os.devnull, os.O_WRONLY)
    os.dup2(devnull_fd, 1)
    try:
        tool_globals = {{}}
        exec(compile({clean_source!r}, "<custom-tool>", "exec"), tool_globals)
        tool_result = pydantic.TypeAdapter(tool_globals[{function_def.name!r}]).validate_json({tool_input_json!r})
    finally:
        sys.stdout.flush()
        os.dup2(saved_stdout_fd, 1)

    print(json.dumps(tool_result))
"""
        result = await self.code_executor.execute(source_code=script, env=dict(env))
        if result.exit_code != 0:
            raise CustomToolExecuteError(result.stderr)
        return json.loads(result.stdout)
