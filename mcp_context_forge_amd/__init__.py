"""MI355X-native MCP/A2A gateway (capabilities of IBM/mcp-context-forge, rebuilt GPU-first)."""

__version__ = "0.1.0"
