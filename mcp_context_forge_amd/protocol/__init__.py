from . import jsonrpc, mcp  # noqa: F401
