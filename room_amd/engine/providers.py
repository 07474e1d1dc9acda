"""Model dispatch + the multi-turn tool-call loop.

The reference routes model strings to external executors (CLI subprocesses /
HTTP APIs, src/shared/agent-executor.ts:91-152) and runs a multi-turn tool
loop only for API models (agent-executor.ts:378-474). Here that loop is the
one and only execution path, and the "provider" is an in-process engine:

- "stub"                  → StubEngine (deterministic; the mocked-executeAgent
                            seam the reference's tests rely on)
- "qwen3-coder-30b" /
  "local:*"               → GPU engine (room_amd.engine.llm.LocalEngine),
                            CDNA4 HIP kernels, created lazily per process
"""
from __future__ import annotations

import json
import re
import time
import uuid
from typing import Optional, Protocol

from ..core.constants import (LOCAL_MODEL_TAG, SESSION_COMPRESS_AT_MSGS,
                              SESSION_TRIM_TO_MSGS)
from .types import AgentExecutionOptions, AgentExecutionResult, ToolCall, ToolDef

TOOL_CALL_RE = re.compile(r"<tool_call>\s*(\{.*?\})\s*</tool_call>", re.S)


def parse_tool_calls(text: str) -> list[ToolCall]:
    calls = []
    for m in TOOL_CALL_RE.finditer(text):
        try:
            obj = json.loads(m.group(1))
            if isinstance(obj, dict) and "name" in obj:
                calls.append(ToolCall(name=obj["name"],
                                      arguments=obj.get("arguments", {}) or {},
                                      id=uuid.uuid4().hex[:8]))
        except (ValueError, TypeError):
            continue
    return calls


def render_tool_call(name: str, arguments: dict) -> str:
    return f'<tool_call>{json.dumps({"name": name, "arguments": arguments})}</tool_call>'


class ChatEngine(Protocol):
    """One chat turn: messages in, assistant text (+usage) out.
    Implementations: StubEngine (CPU, deterministic) and LocalEngine (GPU)."""

    def chat(self, messages: list[dict], tools: list[ToolDef],
             options: AgentExecutionOptions) -> tuple[str, int, int]:
        """Returns (assistant_text, input_tokens, output_tokens)."""
        ...


class StubEngine:
    """Deterministic scripted engine (BASELINE config 1, and the unit-test
    seam). Behavior: a queen-style prompt with an objective and no subgoals
    decomposes the objective; with pending work it saves WIP; otherwise it
    acknowledges. A scripted response queue can override."""

    def __init__(self, scripted: Optional[list[str]] = None):
        self.scripted = list(scripted) if scripted else []
        self.calls: list[list[dict]] = []

    def chat(self, messages: list[dict], tools: list[ToolDef],
             options: AgentExecutionOptions) -> tuple[str, int, int]:
        self.calls.append(messages)
        itok = sum(len(m.get("content", "")) // 4 for m in messages)
        if self.scripted:
            text = self.scripted.pop(0)
            return text, itok, len(text) // 4
        last = messages[-1]["content"] if messages else ""
        tool_names = {t.name for t in tools}
        # tool results arriving → acknowledge and stop
        if messages and messages[-1].get("role") == "tool":
            return "Done. Work recorded.", itok, 6
        if "OBJECTIVE:" in last and "room_set_goal" in tool_names \
                and "No subgoals yet" in last:
            objective = last.split("OBJECTIVE:", 1)[1].splitlines()[0].strip()
            parts = [
                render_tool_call("room_set_goal",
                                 {"description": f"Plan: {objective} — phase {i+1}"})
                for i in range(3)
            ]
            return "Decomposing the objective.\n" + "\n".join(parts), itok, 64
        if "room_save_wip" in tool_names and "CONTINUE FORWARD" in last:
            return ("Continuing WIP.\n"
                    + render_tool_call("room_save_wip", {"wip": "continued step"})), itok, 24
        return "Observed. Nothing to do this cycle.", itok, 10


# ------------------------------------------------------------------ registry

_engines: dict[str, ChatEngine] = {}


def register_engine(model: str, engine: ChatEngine) -> None:
    _engines[model] = engine


def get_model_provider(model: str) -> str:
    """Model string → provider family (model-provider.ts:31-41): cloud
    prefixes route to HTTP providers; everything else runs in-process."""
    if model.startswith("stub"):
        return "stub"
    from .cloud_providers import provider_of
    cloud = provider_of(model)
    if cloud:
        return cloud
    return "local"


def resolve_engine(model: str, ldb=None, room_id: int | None = None) -> ChatEngine:
    if model in _engines:
        return _engines[model]
    provider = get_model_provider(model)
    if provider == "stub":
        eng = StubEngine()
        _engines[model] = eng
        return eng
    if provider != "local":
        from .cloud_providers import (HttpChatEngine,
                                      resolve_api_key_for_model)
        key = (resolve_api_key_for_model(ldb, model, room_id)
               if ldb is not None else None)
        eng = HttpChatEngine(model, api_key=key)
        _engines[model] = eng
        return eng
    # lazy-build the GPU engine
    from .llm import get_local_engine
    eng = get_local_engine(model)
    _engines[model] = eng
    return eng


# ------------------------------------------------------------------ execution


def _system_with_tools(system_prompt: str, tools: list[ToolDef]) -> str:
    if not tools:
        return system_prompt
    tool_lines = "\n".join(
        f"- {t.name}: {t.description} parameters={json.dumps(t.parameters)}"
        for t in tools)
    return (f"{system_prompt}\n\n# Tools\nYou may call tools by emitting\n"
            f"<tool_call>{{\"name\": \"...\", \"arguments\": {{...}}}}</tool_call>\n"
            f"Available tools:\n{tool_lines}")


def execute_agent(options: AgentExecutionOptions) -> AgentExecutionResult:
    """The multi-turn tool-call loop (sole execution path).

    messages → engine → parse <tool_call> blocks → run each in-process via
    options.tool_executor → append tool results → repeat ≤ max_turns.
    """
    start = time.time()
    engine = resolve_engine(options.model)
    messages: list[dict] = list(options.messages or [])
    sys_msg = _system_with_tools(options.system_prompt, options.tools)
    if not messages or messages[0].get("role") != "system":
        messages.insert(0, {"role": "system", "content": sys_msg})
    else:
        messages[0] = {"role": "system", "content": sys_msg}
    messages.append({"role": "user", "content": options.prompt})

    total_in = total_out = 0
    tool_calls_executed = 0
    final_text = ""
    turns = 0
    try:
        for turns in range(1, max(1, options.max_turns) + 1):
            text, itok, otok = engine.chat(messages, options.tools, options)
            total_in += itok
            total_out += otok
            messages.append({"role": "assistant", "content": text})
            final_text = text
            if options.on_log:
                options.on_log("assistant", text)
            calls = parse_tool_calls(text)
            if not calls or options.tool_executor is None:
                break
            for call in calls:
                result = options.tool_executor(call)
                tool_calls_executed += 1
                if options.on_log:
                    options.on_log("tool", f"{call.name} → {result[:500]}")
                messages.append({"role": "tool", "name": call.name,
                                 "content": result})
        return AgentExecutionResult(
            text=final_text, success=True,
            session_id=options.session_id or uuid.uuid4().hex,
            messages=messages, input_tokens=total_in, output_tokens=total_out,
            turns_used=turns, tool_calls_executed=tool_calls_executed,
            duration_ms=int((time.time() - start) * 1000))
    except Exception as e:
        return AgentExecutionResult(
            text=final_text, success=False, error=str(e), messages=messages,
            input_tokens=total_in, output_tokens=total_out, turns_used=turns,
            tool_calls_executed=tool_calls_executed,
            duration_ms=int((time.time() - start) * 1000))


def compress_session(messages: list[dict], model: str = "stub") -> list[dict]:
    """Compress a long API-style session: summarize all but the most recent
    turns into one system note (reference: agent-executor.ts:878-948 uses a
    1-turn summarizer call; trim to 40 msgs hard cap)."""
    if len(messages) < SESSION_COMPRESS_AT_MSGS:
        return messages
    head = [m for m in messages if m.get("role") == "system"][:1]
    recent = messages[-10:]
    older = [m for m in messages[len(head):-10]]
    summary_src = "\n".join(
        f"{m.get('role')}: {str(m.get('content'))[:200]}" for m in older)
    try:
        engine = resolve_engine(model)
        opts = AgentExecutionOptions(prompt="", model=model, max_new_tokens=256)
        text, _, _ = engine.chat(
            [{"role": "system", "content": "Summarize this agent session into a "
              "compact memo of decisions, open work and learned facts."},
             {"role": "user", "content": summary_src[:8000]}], [], opts)
    except Exception:
        text = summary_src[:1500]
    compressed = head + [{"role": "system",
                          "content": f"[session summary]\n{text[:1500]}"}] + recent
    return compressed[-SESSION_TRIM_TO_MSGS:]
