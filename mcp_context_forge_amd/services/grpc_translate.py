"""gRPC → MCP translation via server reflection.

Reference analogs: mcpgateway/translate_grpc.py (GrpcEndpoint :68,
reflection-based message-class synthesis :321-401, GrpcToMcpTranslator :487)
and services/grpc_service.py (descriptor DoS guards :58,137,202).

The grpc_reflection wheel is absent in this image, so the reflection
protocol (grpc.reflection.v1alpha.ServerReflection) is implemented natively:
its descriptors are constructed programmatically with protobuf
descriptor_pool (`_build_reflection_descriptors`), which also yields a
server-side servicer used by tests and by anyone exposing this gateway's
own tools over gRPC.
"""

from __future__ import annotations

import json
from typing import Any, Dict, List, Optional, Tuple

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, json_format, message_factory

# DoS guards (reference: grpc_service.py:58,137,202)
MAX_DESCRIPTOR_BYTES = 4 * 1024 * 1024
MAX_SERVICES = 512
MAX_METHODS = 10000
MAX_MESSAGE_DEPTH = 32


# ---------------------------------------------------------------------------
# Reflection proto, built programmatically (grpc/reflection/v1alpha/reflection.proto)
# ---------------------------------------------------------------------------

def _build_reflection_descriptors():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "grpc/reflection/v1alpha/reflection.proto"
    fdp.package = "grpc.reflection.v1alpha"
    fdp.syntax = "proto3"

    req = fdp.message_type.add()
    req.name = "ServerReflectionRequest"
    f = req.field.add(); f.name = "host"; f.number = 1; f.type = f.TYPE_STRING; f.label = f.LABEL_OPTIONAL
    req.oneof_decl.add().name = "message_request"
    for num, name in ((3, "file_by_filename"), (4, "file_containing_symbol"), (7, "list_services")):
        f = req.field.add(); f.name = name; f.number = num; f.type = f.TYPE_STRING
        f.label = f.LABEL_OPTIONAL; f.oneof_index = 0

    fdr = fdp.message_type.add()
    fdr.name = "FileDescriptorResponse"
    f = fdr.field.add(); f.name = "file_descriptor_proto"; f.number = 1; f.type = f.TYPE_BYTES
    f.label = f.LABEL_REPEATED

    svc_resp = fdp.message_type.add()
    svc_resp.name = "ServiceResponse"
    f = svc_resp.field.add(); f.name = "name"; f.number = 1; f.type = f.TYPE_STRING; f.label = f.LABEL_OPTIONAL

    lsr = fdp.message_type.add()
    lsr.name = "ListServiceResponse"
    f = lsr.field.add(); f.name = "service"; f.number = 1; f.type = f.TYPE_MESSAGE
    f.label = f.LABEL_REPEATED; f.type_name = ".grpc.reflection.v1alpha.ServiceResponse"

    err = fdp.message_type.add()
    err.name = "ErrorResponse"
    f = err.field.add(); f.name = "error_code"; f.number = 1; f.type = f.TYPE_INT32; f.label = f.LABEL_OPTIONAL
    f = err.field.add(); f.name = "error_message"; f.number = 2; f.type = f.TYPE_STRING; f.label = f.LABEL_OPTIONAL

    resp = fdp.message_type.add()
    resp.name = "ServerReflectionResponse"
    f = resp.field.add(); f.name = "valid_host"; f.number = 1; f.type = f.TYPE_STRING; f.label = f.LABEL_OPTIONAL
    for num, name, tn in ((4, "file_descriptor_response", ".grpc.reflection.v1alpha.FileDescriptorResponse"),
                          (6, "list_services_response", ".grpc.reflection.v1alpha.ListServiceResponse"),
                          (7, "error_response", ".grpc.reflection.v1alpha.ErrorResponse")):
        f = resp.field.add(); f.name = name; f.number = num; f.type = f.TYPE_MESSAGE
        f.label = f.LABEL_OPTIONAL; f.type_name = tn; f.oneof_index = 0
    resp.oneof_decl.add().name = "message_response"

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    req_cls = message_factory.GetMessageClass(pool.FindMessageTypeByName("grpc.reflection.v1alpha.ServerReflectionRequest"))
    resp_cls = message_factory.GetMessageClass(pool.FindMessageTypeByName("grpc.reflection.v1alpha.ServerReflectionResponse"))
    return req_cls, resp_cls


_REQ_CLS, _RESP_CLS = _build_reflection_descriptors()
REFLECTION_SERVICE = "grpc.reflection.v1alpha.ServerReflection"
REFLECTION_METHOD = f"/{REFLECTION_SERVICE}/ServerReflectionInfo"


class ReflectionServicer:
    """Server-side reflection for OUR servers/tests (grpc generic handler)."""

    def __init__(self, pool: descriptor_pool.DescriptorPool, service_names: List[str],
                 file_protos: Dict[str, descriptor_pb2.FileDescriptorProto]):
        self.pool = pool
        self.service_names = list(service_names) + [REFLECTION_SERVICE]
        self.file_protos = file_protos  # filename -> FileDescriptorProto

    def __call__(self, request_iterator, context):
        for raw in request_iterator:
            req = _REQ_CLS.FromString(raw) if isinstance(raw, bytes) else raw
            resp = _RESP_CLS()
            which = req.WhichOneof("message_request")
            if which == "list_services":
                for name in self.service_names:
                    resp.list_services_response.service.add().name = name
            elif which == "file_containing_symbol":
                sym = req.file_containing_symbol
                found = None
                for fp in self.file_protos.values():
                    if any(sym == f"{fp.package}.{svc.name}" for svc in fp.service) or \
                       any(sym == f"{fp.package}.{m.name}" for m in fp.message_type):
                        found = fp
                        break
                if found is not None:
                    resp.file_descriptor_response.file_descriptor_proto.append(found.SerializeToString())
                else:
                    resp.error_response.error_code = 5
                    resp.error_response.error_message = f"symbol {sym} not found"
            elif which == "file_by_filename":
                fp = self.file_protos.get(req.file_by_filename)
                if fp is not None:
                    resp.file_descriptor_response.file_descriptor_proto.append(fp.SerializeToString())
                else:
                    resp.error_response.error_code = 5
                    resp.error_response.error_message = "file not found"
            yield resp.SerializeToString()

    def add_to_server(self, server: grpc.Server) -> None:
        handler = grpc.stream_stream_rpc_method_handler(
            self, request_deserializer=lambda b: b, response_serializer=lambda b: b)
        generic = grpc.method_handlers_generic_handler(
            REFLECTION_SERVICE, {"ServerReflectionInfo": handler})
        server.add_generic_rpc_handlers((generic,))


class GrpcEndpoint:
    """Discovery + invocation for one upstream gRPC server
    (reference: translate_grpc.GrpcEndpoint :68)."""

    def __init__(self, target: str, timeout: float = 10.0):
        self.target = target
        self.timeout = timeout
        self.channel = grpc.insecure_channel(target)
        self.pool = descriptor_pool.DescriptorPool()
        self._known_files: set = set()
        self.services: Dict[str, Any] = {}  # fq service name -> ServiceDescriptor

    def _reflect(self, **kwargs) -> Any:
        req = _REQ_CLS(**kwargs)
        call = self.channel.stream_stream(
            REFLECTION_METHOD,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=_RESP_CLS.FromString,
        )
        responses = call(iter([req]), timeout=self.timeout)
        for resp in responses:
            return resp
        raise RuntimeError("no reflection response")

    def _add_file(self, raw: bytes, budget: List[int]) -> None:
        budget[0] -= len(raw)
        if budget[0] < 0:
            raise RuntimeError("descriptor size budget exceeded (DoS guard)")
        fp = descriptor_pb2.FileDescriptorProto.FromString(raw)
        if fp.name in self._known_files:
            return
        # resolve dependencies first
        for dep in fp.dependency:
            if dep in self._known_files:
                continue
            resp = self._reflect(file_by_filename=dep)
            if resp.WhichOneof("message_response") == "file_descriptor_response":
                for dep_raw in resp.file_descriptor_response.file_descriptor_proto:
                    self._add_file(dep_raw, budget)
        self.pool.Add(fp)
        self._known_files.add(fp.name)

    def discover(self) -> List[str]:
        """List services + load their descriptors (reflection walk)."""
        resp = self._reflect(list_services="*")
        if resp.WhichOneof("message_response") != "list_services_response":
            raise RuntimeError("reflection list_services failed")
        names = [s.name for s in resp.list_services_response.service
                 if s.name != REFLECTION_SERVICE]
        if len(names) > MAX_SERVICES:
            raise RuntimeError("too many services (DoS guard)")
        budget = [MAX_DESCRIPTOR_BYTES]
        for name in names:
            resp = self._reflect(file_containing_symbol=name)
            if resp.WhichOneof("message_response") != "file_descriptor_response":
                continue
            for raw in resp.file_descriptor_response.file_descriptor_proto:
                self._add_file(raw, budget)
            try:
                self.services[name] = self.pool.FindServiceByName(name)
            except KeyError:
                pass
        return list(self.services.keys())

    def methods(self) -> List[Tuple[str, str, Any]]:
        """[(service, method, MethodDescriptor)] across discovered services."""
        out = []
        for sname, sdesc in self.services.items():
            for m in sdesc.methods:
                out.append((sname, m.name, m))
        if len(out) > MAX_METHODS:
            raise RuntimeError("too many methods (DoS guard)")
        return out

    def invoke(self, service: str, method: str, arguments: Dict[str, Any]) -> Dict[str, Any]:
        sdesc = self.services[service]
        mdesc = sdesc.FindMethodByName(method)
        req_cls = message_factory.GetMessageClass(mdesc.input_type)
        resp_cls = message_factory.GetMessageClass(mdesc.output_type)
        request = json_format.ParseDict(arguments or {}, req_cls(), ignore_unknown_fields=True)
        call = self.channel.unary_unary(
            f"/{service}/{method}",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=resp_cls.FromString,
        )
        response = call(request, timeout=self.timeout)
        return json_format.MessageToDict(response, preserving_proto_field_name=True)

    def close(self) -> None:
        self.channel.close()


def _schema_from_descriptor(msg_desc, depth: int = 0) -> Dict[str, Any]:
    """Protobuf message descriptor → JSON schema (reference: translate_grpc
    message-class synthesis :321-401)."""
    if depth > MAX_MESSAGE_DEPTH:
        return {"type": "object"}
    from google.protobuf.descriptor import FieldDescriptor as FD

    type_map = {
        FD.TYPE_STRING: {"type": "string"}, FD.TYPE_BYTES: {"type": "string"},
        FD.TYPE_BOOL: {"type": "boolean"},
        FD.TYPE_DOUBLE: {"type": "number"}, FD.TYPE_FLOAT: {"type": "number"},
    }
    props: Dict[str, Any] = {}
    for field in msg_desc.fields:
        if field.type == FD.TYPE_MESSAGE:
            sub = _schema_from_descriptor(field.message_type, depth + 1)
        elif field.type == FD.TYPE_ENUM:
            sub = {"type": "string", "enum": [v.name for v in field.enum_type.values]}
        elif field.type in type_map:
            sub = dict(type_map[field.type])
        else:
            sub = {"type": "integer"}
        if getattr(field, "is_repeated", False) or getattr(field, "label", None) == FD.LABEL_REPEATED:
            sub = {"type": "array", "items": sub}
        props[field.json_name or field.name] = sub
    return {"type": "object", "properties": props}


class GrpcToMcpTranslator:
    """Expose a gRPC server's methods as MCP tools (reference: translate_grpc
    GrpcToMcpTranslator :487)."""

    def __init__(self, target: str, prefix: str = "grpc"):
        self.endpoint = GrpcEndpoint(target)
        self.prefix = prefix

    def discover_tools(self) -> List[Dict[str, Any]]:
        self.endpoint.discover()
        tools = []
        for service, method, mdesc in self.endpoint.methods():
            if mdesc.client_streaming or mdesc.server_streaming:
                continue  # unary only (reference limitation too)
            tools.append({
                "name": f"{self.prefix}-{service.replace('.', '-')}-{method}",
                "description": f"gRPC {service}/{method}",
                "inputSchema": _schema_from_descriptor(mdesc.input_type),
                "_grpc": (service, method),
            })
        return tools

    async def call_tool(self, service: str, method: str, arguments: Dict[str, Any]) -> Dict[str, Any]:
        import asyncio

        value = await asyncio.to_thread(self.endpoint.invoke, service, method, arguments)
        return {
            "content": [{"type": "text", "text": json.dumps(value, default=str)}],
            "structuredContent": value,
            "isError": False,
        }

    def register_into(self, tool_service, gateway_name: str = "grpc") -> List[dict]:
        """Register discovered methods as LOCAL tools backed by gRPC calls."""
        out = []
        for td in self.discover_tools():
            service, method = td.pop("_grpc")

            async def handler(args, _s=service, _m=method):
                return await self.call_tool(_s, _m, args)

            out.append(tool_service.register_local_tool(
                td["name"], handler, td["description"], input_schema=td["inputSchema"],
                annotations={"io": True}))  # real network hop → batch dispatch concurrently
        return out

    def close(self) -> None:
        self.endpoint.close()
