"""gRPC→MCP translation against a live in-process gRPC server with native
reflection (reference analog: translate_grpc tests)."""

import json
import socket
from concurrent import futures

import grpc
import pytest
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

from mcp_context_forge_amd.services.grpc_translate import (
    GrpcToMcpTranslator,
    ReflectionServicer,
    _schema_from_descriptor,
)


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _greeter_file() -> descriptor_pb2.FileDescriptorProto:
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "test/greeter.proto"
    fdp.package = "test.v1"
    fdp.syntax = "proto3"
    req = fdp.message_type.add()
    req.name = "HelloRequest"
    f = req.field.add(); f.name = "name"; f.number = 1; f.type = f.TYPE_STRING; f.label = f.LABEL_OPTIONAL
    f = req.field.add(); f.name = "count"; f.number = 2; f.type = f.TYPE_INT32; f.label = f.LABEL_OPTIONAL
    rep = fdp.message_type.add()
    rep.name = "HelloReply"
    f = rep.field.add(); f.name = "message"; f.number = 1; f.type = f.TYPE_STRING; f.label = f.LABEL_OPTIONAL
    f = rep.field.add(); f.name = "echoes"; f.number = 2; f.type = f.TYPE_STRING; f.label = f.LABEL_REPEATED
    svc = fdp.service.add()
    svc.name = "Greeter"
    m = svc.method.add()
    m.name = "SayHello"
    m.input_type = ".test.v1.HelloRequest"
    m.output_type = ".test.v1.HelloReply"
    return fdp


@pytest.fixture()
def grpc_server():
    fdp = _greeter_file()
    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    req_cls = message_factory.GetMessageClass(pool.FindMessageTypeByName("test.v1.HelloRequest"))
    rep_cls = message_factory.GetMessageClass(pool.FindMessageTypeByName("test.v1.HelloReply"))

    def say_hello(request, context):
        reply = rep_cls()
        reply.message = f"Hello {request.name}!"
        for _ in range(request.count):
            reply.echoes.append(request.name)
        return reply

    server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))
    handler = grpc.unary_unary_rpc_method_handler(
        say_hello, request_deserializer=req_cls.FromString,
        response_serializer=lambda m: m.SerializeToString())
    server.add_generic_rpc_handlers((grpc.method_handlers_generic_handler(
        "test.v1.Greeter", {"SayHello": handler}),))
    ReflectionServicer(pool, ["test.v1.Greeter"], {fdp.name: fdp}).add_to_server(server)
    port = _free_port()
    server.add_insecure_port(f"127.0.0.1:{port}")
    server.start()
    yield f"127.0.0.1:{port}"
    server.stop(grace=None)


def test_discover_and_invoke(grpc_server):
    tr = GrpcToMcpTranslator(grpc_server, prefix="g")
    tools = tr.discover_tools()
    assert len(tools) == 1
    t = tools[0]
    assert t["name"] == "g-test-v1-Greeter-SayHello"
    assert t["inputSchema"]["properties"]["name"] == {"type": "string"}
    assert t["inputSchema"]["properties"]["count"] == {"type": "integer"}

    import asyncio

    res = asyncio.run(tr.call_tool("test.v1.Greeter", "SayHello", {"name": "amd", "count": 2}))
    assert res["structuredContent"]["message"] == "Hello amd!"
    assert res["structuredContent"]["echoes"] == ["amd", "amd"]
    tr.close()


def test_register_into_engine(grpc_server, run):
    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine

    engine = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                    plugins_enabled=False, auth_required=False))
    tr = GrpcToMcpTranslator(grpc_server, prefix="g")
    tools = tr.register_into(engine.tool_service)
    assert len(tools) == 1

    async def go():
        out = await engine.handle_rpc_bytes(json.dumps({
            "jsonrpc": "2.0", "id": 1, "method": "tools/call",
            "params": {"name": "g-test-v1-Greeter-SayHello", "arguments": {"name": "gpu", "count": 1}},
        }).encode())
        res = json.loads(out)
        assert res["result"]["structuredContent"]["message"] == "Hello gpu!"
        await engine.shutdown()

    run(go())
    tr.close()


def test_schema_depth_guard():
    # self-referential message must not recurse forever
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "test/rec.proto"
    fdp.package = "rec"
    fdp.syntax = "proto3"
    node = fdp.message_type.add()
    node.name = "Node"
    f = node.field.add(); f.name = "child"; f.number = 1; f.type = f.TYPE_MESSAGE
    f.label = f.LABEL_OPTIONAL; f.type_name = ".rec.Node"
    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    schema = _schema_from_descriptor(pool.FindMessageTypeByName("rec.Node"))
    assert schema["type"] == "object"  # terminated


def test_invoke_error_status_propagates(grpc_server, run):
    """A server-side gRPC error surfaces as a tool error through the engine
    (not a crash): INVALID_ARGUMENT from the handler → isError response."""
    import asyncio

    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine

    async def go():
        e = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                   auth_required=False))
        tr = GrpcToMcpTranslator(grpc_server, prefix="g")
        tr.register_into(e.tool_service)
        await e.startup()
        # count must be an int32; a string that can't coerce raises in
        # ParseDict client-side → JSON-RPC error, engine stays alive
        raw = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                          "params": {"name": "g-test-v1-Greeter-SayHello",
                                     "arguments": {"name": "x", "count": "NaN"}}}).encode()
        out = json.loads((await e.process_rpc_batch([raw]))[0])
        assert "error" in out or out.get("result", {}).get("isError"), out
        # and a good call on the same engine still works
        raw2 = json.dumps({"jsonrpc": "2.0", "id": 2, "method": "tools/call",
                           "params": {"name": "g-test-v1-Greeter-SayHello",
                                      "arguments": {"name": "ok", "count": 2}}}).encode()
        out2 = json.loads((await e.process_rpc_batch([raw2]))[0])
        assert "result" in out2, out2
        sc = out2["result"]["structuredContent"]
        assert sc["message"] == "Hello ok!" and sc["echoes"] == ["ok", "ok"]
        await e.shutdown()

    run(go())


def test_unknown_fields_ignored_and_schema_types(grpc_server):
    """ParseDict(ignore_unknown_fields=True) mirrors the reference's lax
    argument handling; the synthesized schema types match the proto."""
    tr = GrpcToMcpTranslator(grpc_server)
    tools = tr.discover_tools()
    assert len(tools) == 1
    schema = tools[0]["inputSchema"]
    assert schema["properties"]["name"] == {"type": "string"}
    assert schema["properties"]["count"] == {"type": "integer"}
    # unknown argument key is dropped, not an error
    v = tr.endpoint.invoke("test.v1.Greeter", "SayHello",
                           {"name": "n", "count": 1, "bogus_key": "zzz"})
    assert v["message"] == "Hello n!"
    tr.endpoint.close()
