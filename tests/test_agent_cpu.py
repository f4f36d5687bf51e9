"""Tool-calling agent loop (injected generator, no model needed)."""
from mlx_cuda_distributed_pretraining_amd.inference.agent import (
    AgentConfig, GenerationAgent,
)


def test_parse_tool_call():
    a = GenerationAgent(None, None)
    assert a.parse_tool_call("no call here") is None
    got = a.parse_tool_call('x <tool>{"name": "add", "args": {"a": 1, "b": 2}}</tool> y')
    assert got == ("add", {"a": 1, "b": 2})
    assert a.parse_tool_call("<tool>{broken json}</tool>") is None


def test_agent_loop_with_tool():
    a = GenerationAgent(None, None, AgentConfig(max_turns=4))
    a.register_tool("add", lambda a_, b_: a_ + b_, "add two numbers")
    # keyword names must match the registered fn signature
    a.tools["add"].fn = lambda a=0, b=0: a + b

    outputs = iter([
        '<tool>{"name": "add", "args": {"a": 2, "b": 3}}</tool>',
        "The answer is 5.",
    ])

    def fake_generate(model, tok, prompt, **kw):
        return next(outputs), {}

    res = a.run("what is 2+3?", generate_fn=fake_generate)
    assert res["tool_calls"] == 1
    assert res["answer"] == "The answer is 5."
    assert res["turns"][0]["tool_result"] == "5"


def test_agent_unknown_tool_and_max_turns():
    a = GenerationAgent(None, None, AgentConfig(max_turns=2))

    def fake_generate(model, tok, prompt, **kw):
        return '<tool>{"name": "nope", "args": {}}</tool>', {}

    res = a.run("hi", generate_fn=fake_generate)
    assert res["tool_calls"] == 2  # hit max_turns, each call unknown
    assert "unknown tool" in res["turns"][0]["tool_result"]


def test_system_preamble_lists_tools():
    a = GenerationAgent(None, None)
    assert a.system_preamble() == ""
    a.register_tool("search", lambda q="": "", "search the web")
    assert "search the web" in a.system_preamble()
