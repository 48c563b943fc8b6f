"""Wire-format sanity for the runtime-built protobuf classes: serialize /
parse round-trips (incl. map fields and oneofs) behave exactly like
protoc-generated classes would."""

from code_interpreter_amd.grpc_api import descriptors as d


def test_execute_request_roundtrip():
    msg = d.ExecuteRequest(source_code="print(1)", files={"/w/a.txt": "ab" * 32})
    data = msg.SerializeToString()
    back = d.ExecuteRequest.FromString(data)
    assert back.source_code == "print(1)"
    assert dict(back.files) == {"/w/a.txt": "ab" * 32}


def test_execute_response_fields():
    msg = d.ExecuteResponse(stdout="o", stderr="e", exit_code=-1, files={})
    back = d.ExecuteResponse.FromString(msg.SerializeToString())
    assert (back.stdout, back.stderr, back.exit_code) == ("o", "e", -1)


def test_oneof_exclusivity():
    resp = d.ParseCustomToolResponse()
    resp.success.tool_name = "t"
    assert resp.WhichOneof("response") == "success"
    resp.error.error_messages.append("boom")
    assert resp.WhichOneof("response") == "error"  # setting error clears success
    back = d.ParseCustomToolResponse.FromString(resp.SerializeToString())
    assert back.WhichOneof("response") == "error"
    assert list(back.error.error_messages) == ["boom"]


def test_unset_scalar_defaults():
    back = d.ExecuteResponse.FromString(b"")
    assert back.stdout == "" and back.exit_code == 0
