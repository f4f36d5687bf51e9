"""Tool-calling generation agent.

Parity surface: /root/reference/generate_agent.py (tool-call loop over a
trained model: the model emits ``<tool>{"name": ..., "args": {...}}</tool>``
blocks, the agent executes the registered python function and feeds the
result back, looping until a plain answer or max_turns).
"""
from __future__ import annotations

import json
import re
from dataclasses import dataclass
from typing import Any, Callable, Dict, List, Optional, Tuple

TOOL_RE = re.compile(r"<tool>\s*(\{.*?\})\s*</tool>", re.DOTALL)


@dataclass
class Tool:
    name: str
    fn: Callable[..., Any]
    description: str = ""


@dataclass
class AgentConfig:
    max_turns: int = 5
    max_tokens_per_turn: int = 256
    temperature: float = 0.7
    top_p: float = 0.9


class GenerationAgent:
    """Loop: prompt -> generate -> (parse tool call -> run tool -> append
    result -> generate again) until no tool call or max_turns."""

    def __init__(self, model, tokenizer, config: Optional[AgentConfig] = None):
        self.model = model
        self.tokenizer = tokenizer
        self.config = config or AgentConfig()
        self.tools: Dict[str, Tool] = {}

    def register_tool(self, name: str, fn: Callable[..., Any], description: str = "") -> None:
        self.tools[name] = Tool(name, fn, description)

    def system_preamble(self) -> str:
        if not self.tools:
            return ""
        lines = ["You can call tools by emitting <tool>{\"name\": NAME, \"args\": {...}}</tool>.",
                 "Available tools:"]
        for t in self.tools.values():
            lines.append(f"- {t.name}: {t.description}")
        return "\n".join(lines) + "\n\n"

    def parse_tool_call(self, text: str) -> Optional[Tuple[str, Dict[str, Any]]]:
        m = TOOL_RE.search(text)
        if not m:
            return None
        try:
            call = json.loads(m.group(1))
            return str(call["name"]), dict(call.get("args", {}))
        except (json.JSONDecodeError, KeyError, TypeError):
            return None

    def run_tool(self, name: str, args: Dict[str, Any]) -> str:
        tool = self.tools.get(name)
        if tool is None:
            return f"error: unknown tool '{name}'"
        try:
            return str(tool.fn(**args))
        except Exception as e:  # tool errors go back to the model as text
            return f"error: {e}"

    def run(self, prompt: str, generate_fn: Optional[Callable] = None) -> Dict[str, Any]:
        """Returns {"answer": str, "turns": [...], "tool_calls": int}.

        ``generate_fn(model, tokenizer, prompt, **kw) -> (text, stats)`` may be
        injected for testing; defaults to inference.generate.generate.
        """
        if generate_fn is None:
            from .generate import generate as generate_fn  # type: ignore

        transcript = self.system_preamble() + prompt
        turns: List[Dict[str, Any]] = []
        n_calls = 0
        text = ""
        for _ in range(self.config.max_turns):
            text, _stats = generate_fn(
                self.model, self.tokenizer, transcript,
                max_tokens=self.config.max_tokens_per_turn,
                temperature=self.config.temperature, top_p=self.config.top_p,
            )
            call = self.parse_tool_call(text)
            turns.append({"model": text, "tool_call": call})
            if call is None:
                break
            name, args = call
            result = self.run_tool(name, args)
            n_calls += 1
            turns[-1]["tool_result"] = result
            transcript = transcript + text + f"\n<tool_result>{result}</tool_result>\n"
        return {"answer": text, "turns": turns, "tool_calls": n_calls}
