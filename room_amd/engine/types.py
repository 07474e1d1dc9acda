"""Execution-seam types (the reference's AgentExecutionResult seam,
src/shared/agent-executor.ts:41-48 — the interface its tests mock and the
interface every engine here implements)."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, Optional


@dataclass
class ToolDef:
    """OpenAI-format tool definition (reference queen-tools.ts ToolDef)."""
    name: str
    description: str
    parameters: dict  # JSON schema

    def as_openai(self) -> dict:
        return {"type": "function",
                "function": {"name": self.name, "description": self.description,
                             "parameters": self.parameters}}


@dataclass
class ToolCall:
    name: str
    arguments: dict
    id: str = ""


@dataclass
class AgentExecutionOptions:
    prompt: str
    model: str = "stub"
    system_prompt: str = ""
    max_turns: int = 10
    session_id: Optional[str] = None
    messages: Optional[list[dict]] = None   # API-style session continuity
    tools: list[ToolDef] = field(default_factory=list)
    tool_executor: Optional[Callable[[ToolCall], str]] = None
    worker_id: Optional[int] = None
    room_id: Optional[int] = None
    max_new_tokens: int = 256
    temperature: float = 0.7
    top_p: float = 0.95
    top_k: int = 40
    on_log: Optional[Callable[[str, str], None]] = None  # (entry_type, content)


@dataclass
class AgentExecutionResult:
    text: str = ""
    success: bool = True
    error: Optional[str] = None
    session_id: Optional[str] = None
    messages: Optional[list[dict]] = None
    input_tokens: int = 0
    output_tokens: int = 0
    turns_used: int = 0
    tool_calls_executed: int = 0
    duration_ms: int = 0
