"""Engine type contracts: AgentExecutionOptions/Result defaults and ToolDef
serialization (the executor seam's stable surface — agent-executor.ts:41-48).
"""
from room_amd.engine.types import (AgentExecutionOptions, AgentExecutionResult,
                                   ToolCall, ToolDef)


def test_options_defaults():
    o = AgentExecutionOptions(prompt="p", model="stub")
    assert o.max_turns >= 1
    assert o.temperature > 0 and 0 < o.top_p <= 1 and o.top_k > 0
    assert o.max_new_tokens > 0
    assert o.tools == [] or o.tools is None
    assert o.messages is None


def test_result_defaults_and_usage_fields():
    r = AgentExecutionResult(success=True, text="hi")
    assert r.error is None
    assert r.input_tokens == 0 and r.output_tokens == 0
    assert r.tool_calls_executed == 0
    assert r.turns_used == 0


def test_tooldef_openai_shape():
    t = ToolDef("room_set_goal", "Set the goal.",
                {"type": "object",
                 "properties": {"description": {"type": "string"}},
                 "required": ["description"]})
    d = t.as_openai()
    assert d["type"] == "function"
    f = d["function"]
    assert f["name"] == "room_set_goal"
    assert f["parameters"]["required"] == ["description"]


def test_toolcall_fields():
    c = ToolCall("a", {"x": 1})
    assert c.name == "a" and c.arguments == {"x": 1}
