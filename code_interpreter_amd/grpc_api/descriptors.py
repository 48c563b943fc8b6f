"""Runtime-built protobuf message classes for CodeInterpreterService.

grpcio-tools/protoc are not available in this image, so the descriptors in
code_interpreter_service.proto are constructed programmatically with
FileDescriptorProto and registered in the default pool; message classes
come from message_factory. The wire format is identical to protoc output
for that file.
"""

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

PACKAGE = "code_interpreter.v1"
FILE_NAME = "code_interpreter/v1/code_interpreter_service.proto"
SERVICE_NAME = f"{PACKAGE}.CodeInterpreterService"

_F = descriptor_pb2.FieldDescriptorProto


def _string_field(name: str, number: int) -> descriptor_pb2.FieldDescriptorProto:
    return _F(
        name=name,
        number=number,
        label=_F.LABEL_OPTIONAL,
        type=_F.TYPE_STRING,
    )


def _int32_field(name: str, number: int) -> descriptor_pb2.FieldDescriptorProto:
    return _F(
        name=name,
        number=number,
        label=_F.LABEL_OPTIONAL,
        type=_F.TYPE_INT32,
    )


def _map_string_string_field(
    msg: descriptor_pb2.DescriptorProto, name: str, number: int
) -> None:
    """Add a map<string,string> field (nested MapEntry message + repeated
    message field), as protoc would emit it."""
    entry_name = "".join(part.capitalize() for part in name.split("_")) + "Entry"
    entry = msg.nested_type.add()
    entry.name = entry_name
    entry.options.map_entry = True
    entry.field.append(_string_field("key", 1))
    entry.field.append(_string_field("value", 2))
    field = msg.field.add()
    field.name = name
    field.number = number
    field.label = _F.LABEL_REPEATED
    field.type = _F.TYPE_MESSAGE
    field.type_name = f".{PACKAGE}.{msg.name}.{entry_name}"


def _oneof_response(
    msg: descriptor_pb2.DescriptorProto, success_type: str, error_type: str
) -> None:
    msg.oneof_decl.add().name = "response"
    for i, (fname, tname) in enumerate(
        (("success", success_type), ("error", error_type))
    ):
        field = msg.field.add()
        field.name = fname
        field.number = i + 1
        field.label = _F.LABEL_OPTIONAL
        field.type = _F.TYPE_MESSAGE
        field.type_name = f".{PACKAGE}.{tname}"
        field.oneof_index = 0


def _build_file_descriptor() -> descriptor_pb2.FileDescriptorProto:
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = FILE_NAME
    fdp.package = PACKAGE
    fdp.syntax = "proto3"

    m = fdp.message_type.add()
    m.name = "ExecuteRequest"
    m.field.append(_string_field("source_code", 1))
    _map_string_string_field(m, "files", 2)

    m = fdp.message_type.add()
    m.name = "ExecuteResponse"
    m.field.append(_string_field("stdout", 1))
    m.field.append(_string_field("stderr", 2))
    m.field.append(_int32_field("exit_code", 3))
    _map_string_string_field(m, "files", 4)

    m = fdp.message_type.add()
    m.name = "ParseCustomToolRequest"
    m.field.append(_string_field("tool_source_code", 1))

    m = fdp.message_type.add()
    m.name = "ParseCustomToolSuccess"
    m.field.append(_string_field("tool_name", 1))
    m.field.append(_string_field("tool_input_schema_json", 2))
    m.field.append(_string_field("tool_description", 3))

    m = fdp.message_type.add()
    m.name = "ParseCustomToolError"
    f = m.field.add()
    f.name = "error_messages"
    f.number = 1
    f.label = _F.LABEL_REPEATED
    f.type = _F.TYPE_STRING

    m = fdp.message_type.add()
    m.name = "ParseCustomToolResponse"
    _oneof_response(m, "ParseCustomToolSuccess", "ParseCustomToolError")

    m = fdp.message_type.add()
    m.name = "ExecuteCustomToolRequest"
    m.field.append(_string_field("tool_source_code", 1))
    m.field.append(_string_field("tool_input_json", 2))

    m = fdp.message_type.add()
    m.name = "ExecuteCustomToolSuccess"
    m.field.append(_string_field("tool_output_json", 1))

    m = fdp.message_type.add()
    m.name = "ExecuteCustomToolError"
    m.field.append(_string_field("stderr", 1))

    m = fdp.message_type.add()
    m.name = "ExecuteCustomToolResponse"
    _oneof_response(m, "ExecuteCustomToolSuccess", "ExecuteCustomToolError")

    svc = fdp.service.add()
    svc.name = "CodeInterpreterService"
    for method, req, resp in (
        ("Execute", "ExecuteRequest", "ExecuteResponse"),
        ("ParseCustomTool", "ParseCustomToolRequest", "ParseCustomToolResponse"),
        ("ExecuteCustomTool", "ExecuteCustomToolRequest", "ExecuteCustomToolResponse"),
    ):
        meth = svc.method.add()
        meth.name = method
        meth.input_type = f".{PACKAGE}.{req}"
        meth.output_type = f".{PACKAGE}.{resp}"

    return fdp


def _load():
    pool = descriptor_pool.Default()
    try:
        file_desc = pool.Add(_build_file_descriptor())
    except Exception:
        # already registered (module re-import)
        file_desc = pool.FindFileByName(FILE_NAME)
    classes = {}
    for name in (
        "ExecuteRequest",
        "ExecuteResponse",
        "ParseCustomToolRequest",
        "ParseCustomToolSuccess",
        "ParseCustomToolError",
        "ParseCustomToolResponse",
        "ExecuteCustomToolRequest",
        "ExecuteCustomToolSuccess",
        "ExecuteCustomToolError",
        "ExecuteCustomToolResponse",
    ):
        classes[name] = message_factory.GetMessageClass(
            pool.FindMessageTypeByName(f"{PACKAGE}.{name}")
        )
    return file_desc, classes


FILE_DESCRIPTOR, MESSAGES = _load()

ExecuteRequest = MESSAGES["ExecuteRequest"]
ExecuteResponse = MESSAGES["ExecuteResponse"]
ParseCustomToolRequest = MESSAGES["ParseCustomToolRequest"]
ParseCustomToolSuccess = MESSAGES["ParseCustomToolSuccess"]
ParseCustomToolError = MESSAGES["ParseCustomToolError"]
ParseCustomToolResponse = MESSAGES["ParseCustomToolResponse"]
ExecuteCustomToolRequest = MESSAGES["ExecuteCustomToolRequest"]
ExecuteCustomToolSuccess = MESSAGES["ExecuteCustomToolSuccess"]
ExecuteCustomToolError = MESSAGES["ExecuteCustomToolError"]
ExecuteCustomToolResponse = MESSAGES["ExecuteCustomToolResponse"]
