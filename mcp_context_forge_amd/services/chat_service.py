"""Built-in LLM chat client over the registered tool catalog.

Reference analog: services/mcp_client_chat_service.py (langchain-based chat
over registered MCP servers, routers/llmchat_router.py). Here the loop is
native: the registry's tools are advertised to an OpenAI-compatible provider
as function tools; tool_calls round-trip through ToolService.invoke_tool
(full plugin chain applies) until the model answers or `max_rounds` runs
out. The transcript is returned so clients can render intermediate steps.
"""

from __future__ import annotations

import json
import logging
from typing import Any, Dict, List, Optional

from .llm_proxy import LLMProxyError

logger = logging.getLogger(__name__)


class McpChatService:
    def __init__(self, engine, max_rounds: int = 5, max_tools: int = 128):
        self.engine = engine
        self.max_rounds = max_rounds
        self.max_tools = max_tools

    async def _tool_defs(self, server_id: Optional[str]) -> List[Dict[str, Any]]:
        tools = await self.engine.tool_service.list_tools(server_id=server_id)
        return [{
            "type": "function",
            "function": {
                "name": t["name"],
                "description": t.get("description") or "",
                "parameters": t.get("input_schema") or {"type": "object"},
            },
        } for t in tools[: self.max_tools]]

    async def chat(self, messages: List[Dict[str, Any]], model: Optional[str] = None,
                   provider: Optional[str] = None, user: Optional[str] = None,
                   server_id: Optional[str] = None, max_rounds: Optional[int] = None) -> Dict[str, Any]:
        """Run a tool-calling chat loop; returns the final assistant message
        plus the tool-call transcript."""
        tool_defs = await self._tool_defs(server_id)
        convo = list(messages)
        transcript: List[Dict[str, Any]] = []
        rounds = max_rounds or self.max_rounds
        out: Dict[str, Any] = {}
        for _ in range(rounds):
            body: Dict[str, Any] = {"messages": convo}
            if model:
                body["model"] = model
            if tool_defs:
                body["tools"] = tool_defs
            out = await self.engine.llm_proxy.chat_completions(body, provider_name=provider)
            choice = (out.get("choices") or [{}])[0]
            msg = choice.get("message") or {}
            tool_calls = msg.get("tool_calls") or []
            convo.append(msg)
            if not tool_calls:
                break
            for tc in tool_calls:
                fn = tc.get("function") or {}
                name = fn.get("name", "")
                try:
                    args = json.loads(fn.get("arguments") or "{}")
                except Exception:
                    args = {}
                try:
                    result = await self.engine.tool_service.invoke_tool(name, args, user=user,
                                                                        server_id=server_id)
                    content = json.dumps(result.get("structuredContent", result), default=str)
                    ok = True
                except Exception as exc:
                    content = f"tool error: {exc}"
                    ok = False
                transcript.append({"tool": name, "arguments": args, "ok": ok,
                                   "result": content[:2000]})
                convo.append({"role": "tool", "tool_call_id": tc.get("id", ""),
                              "name": name, "content": content})
        final = (out.get("choices") or [{}])[0].get("message", {}) if out else {}
        return {"message": final, "tool_calls": transcript, "rounds": len(transcript),
                "model": out.get("model", model or "")}
