"""JSON-RPC 2.0 core for the MI355X-native gateway.

Implements the exact error taxonomy the reference exposes on its ``/rpc``
endpoint (reference: mcpgateway/main.py:11197-11330 `_handle_rpc_authenticated`,
mcpgateway/validation/jsonrpc.py), so clients see identical codes:

* -32700 parse error          (malformed JSON body)
* -32600 invalid request      (not a JSON-RPC 2.0 request object)
* -32601 method not found
* -32602 invalid params
* -32603 internal error
* -32000 server error         (generic downstream failure)
* -32002 server unavailable   (upstream/gateway offline)
* -32003 rate limited / forbidden by policy

This module is pure-CPU control-plane code; batched request *payloads* are
staged to the GPU by :mod:`mcp_context_forge_amd.gpu.pipeline`.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Mapping, Optional, Union

JSONRPC_VERSION = "2.0"

# Error codes (reference: mcpgateway/main.py:11227-11304)
PARSE_ERROR = -32700
INVALID_REQUEST = -32600
METHOD_NOT_FOUND = -32601
INVALID_PARAMS = -32602
INTERNAL_ERROR = -32603
SERVER_ERROR = -32000
SERVER_UNAVAILABLE = -32002
POLICY_DENIED = -32003

_ERROR_MESSAGES = {
    PARSE_ERROR: "Parse error",
    INVALID_REQUEST: "Invalid Request",
    METHOD_NOT_FOUND: "Method not found",
    INVALID_PARAMS: "Invalid params",
    INTERNAL_ERROR: "Internal error",
    SERVER_ERROR: "Server error",
    SERVER_UNAVAILABLE: "Server unavailable",
    POLICY_DENIED: "Request denied by policy",
}


class JSONRPCError(Exception):
    """JSON-RPC protocol error carrying a wire-level code."""

    def __init__(self, code: int, message: Optional[str] = None, data: Any = None):
        self.code = code
        self.message = message or _ERROR_MESSAGES.get(code, "Error")
        self.data = data
        super().__init__(f"{self.code}: {self.message}")

    def to_dict(self) -> dict:
        err: dict = {"code": self.code, "message": self.message}
        if self.data is not None:
            err["data"] = self.data
        return err


@dataclass
class JSONRPCRequest:
    """Validated JSON-RPC 2.0 request (or notification when ``id`` is None)."""

    method: str
    params: Union[dict, list, None] = None
    id: Union[str, int, None] = None
    jsonrpc: str = JSONRPC_VERSION

    @property
    def is_notification(self) -> bool:
        return self.id is None

    def to_dict(self) -> dict:
        out: dict = {"jsonrpc": self.jsonrpc, "method": self.method}
        if self.params is not None:
            out["params"] = self.params
        if self.id is not None:
            out["id"] = self.id
        return out

    @classmethod
    def from_dict(cls, obj: Any) -> "JSONRPCRequest":
        validate_request(obj)
        return cls(
            method=obj["method"],
            params=obj.get("params"),
            id=obj.get("id"),
            jsonrpc=obj.get("jsonrpc", JSONRPC_VERSION),
        )


@dataclass
class JSONRPCResponse:
    id: Union[str, int, None]
    result: Any = None
    error: Optional[JSONRPCError] = None

    def to_dict(self) -> dict:
        out: dict = {"jsonrpc": JSONRPC_VERSION, "id": self.id}
        if self.error is not None:
            out["error"] = self.error.to_dict()
        else:
            out["result"] = self.result
        return out

    def to_bytes(self) -> bytes:
        return json.dumps(self.to_dict(), separators=(",", ":")).encode()


def validate_request(obj: Any) -> None:
    """Validate the JSON-RPC envelope (reference: mcpgateway/validation/jsonrpc.py).

    Raises :class:`JSONRPCError` with -32600 on structural violations.
    """
    if not isinstance(obj, Mapping):
        raise JSONRPCError(INVALID_REQUEST, data="request must be an object")
    if obj.get("jsonrpc") != JSONRPC_VERSION:
        raise JSONRPCError(INVALID_REQUEST, data="jsonrpc must be '2.0'")
    method = obj.get("method")
    if not isinstance(method, str) or not method:
        raise JSONRPCError(INVALID_REQUEST, data="method must be a non-empty string")
    if method.startswith("rpc."):
        raise JSONRPCError(INVALID_REQUEST, data="method names starting with 'rpc.' are reserved")
    if "params" in obj and not isinstance(obj["params"], (dict, list)):
        raise JSONRPCError(INVALID_REQUEST, data="params must be an object or array")
    rid = obj.get("id")
    if rid is not None and not isinstance(rid, (str, int)):
        raise JSONRPCError(INVALID_REQUEST, data="id must be a string, number, or null")


def parse_request_bytes(raw: bytes) -> JSONRPCRequest:
    """Parse raw body bytes into a validated request.

    -32700 for malformed JSON, -32600 for envelope violations (matches the
    reference's ordering at mcpgateway/main.py:11225-11270).
    """
    try:
        # decode first: json.loads(bytes) pays a detect_encoding pass per call
        obj = json.loads(raw.decode("utf-8") if isinstance(raw, (bytes, bytearray)) else raw)
    except Exception as exc:
        raise JSONRPCError(PARSE_ERROR, data=str(exc)) from exc
    return JSONRPCRequest.from_dict(obj)


def error_response(rid: Union[str, int, None], code: int, message: Optional[str] = None, data: Any = None) -> JSONRPCResponse:
    return JSONRPCResponse(id=rid, error=JSONRPCError(code, message, data))


def result_response(rid: Union[str, int, None], result: Any) -> JSONRPCResponse:
    return JSONRPCResponse(id=rid, result=result)
