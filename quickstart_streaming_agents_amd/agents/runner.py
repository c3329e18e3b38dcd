"""AI_RUN_AGENT / AI_TOOL_INVOKE: the per-record agent episode machine.

Semantics contract (LAB1-Walkthrough.md:155-181, LAB3:396-447, LAB4:330-384):
``CREATE AGENT ... USING MODEL m USING PROMPT p [USING TOOLS t] WITH
('max_iterations'='10', 'max_consecutive_failures'='2')`` and per-row
``AI_RUN_AGENT(agent, user_prompt[, key][, debug])`` -> ``(status, response)``:
a reason -> tool -> observe loop bounded by the iteration cap, aborting after
N consecutive tool failures, with the free-text response post-parsed by
REGEXP_EXTRACT (agents/parse.py).

Episodes are *generators* so thousands of concurrent per-record episodes can
share batched GPU decode: an episode yields ("llm", prompt) or
("tool", name, args) requests and is resumed with the result.  The serving
engine (models/serve.py) gangs the "llm" requests into continuous-batching
decode steps while "tool" requests run on an I/O thread pool — the GPU never
waits on MCP round trips.

Action selection is MODEL-DRIVEN by default: episodes constructed with
`tool_schemas` attach a per-turn grammar (models/grammar.py) and the
engine's constrained decode lets the model choose the action (tool call
vs finish) even under random-init weights; ``ToolCallPolicy`` parses the
emitted ``TOOL_CALL {json}`` back into actions.  Scripted policies remain
pluggable for the lab CONTENT-contract tests (labs/pipelines.py), where
the assertions are about deterministic output text (French Quarter /
Naples / verdict enums), not loop shape.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass, field
from typing import Any, Callable, Generator


@dataclass
class ToolDef:
    name: str
    description: str = ""


@dataclass
class ToolSet:
    """CREATE TOOL ... WITH ('type'='mcp','allowed_tools'='...','request_timeout'=N)."""
    name: str
    connection: str = ""
    allowed_tools: tuple[str, ...] = ()
    request_timeout_s: float = 30.0

    def allows(self, tool: str) -> bool:
        return not self.allowed_tools or tool in self.allowed_tools


@dataclass
class AgentSpec:
    """CREATE AGENT definition."""
    name: str
    model: str
    prompt: str
    tools: ToolSet | None = None
    max_iterations: int = 10
    max_consecutive_failures: int = 2
    options: dict = field(default_factory=dict)


# --- episode actions a policy can return -----------------------------------

@dataclass
class ToolCall:
    name: str
    arguments: dict


@dataclass
class Finish:
    response: str


@dataclass
class Continue:
    note: str = ""


Action = "ToolCall | Finish | Continue"


def extract_tool_call_json(text: str) -> str | None:
    """The JSON object after ``TOOL_CALL`` with BALANCED braces (a
    non-greedy regex truncates nested ``arguments`` objects at the first
    closing brace)."""
    m = re.search(r"TOOL_CALL\s*:?\s*\{", text or "", re.IGNORECASE)
    if not m:
        return None
    start = m.end() - 1
    depth = 0
    in_str = False
    esc = False
    for i in range(start, len(text)):
        ch = text[i]
        if in_str:
            if esc:
                esc = False
            elif ch == "\\":
                esc = True
            elif ch == '"':
                in_str = False
        elif ch == '"':
            in_str = True
        elif ch == "{":
            depth += 1
        elif ch == "}":
            depth -= 1
            if depth == 0:
                return text[start:i + 1]
    return None


class ToolCallPolicy:
    """Parse ``TOOL_CALL {json}`` syntax from model output; finish otherwise."""

    def __call__(self, text: str, iteration: int, ctx: dict) -> Any:
        raw = extract_tool_call_json(text or "")
        if raw is not None:
            try:
                call = json.loads(raw)
                return ToolCall(call.get("name", ""),
                                call.get("arguments", {}) or {})
            except json.JSONDecodeError:
                return Continue("unparseable tool call")
        return Finish(text)


@dataclass
class EpisodeResult:
    status: str            # SUCCESS | FAILED
    response: str
    iterations: int = 0
    tool_calls: int = 0
    failures: int = 0
    trace: list[dict] = field(default_factory=list)
    latency_s: float = 0.0   # submit -> finish (set by the scheduler)


def episode(agent: AgentSpec, user_prompt: str,
            policy: Callable[[str, int, dict], Any] | None = None,
            max_new_tokens: int = 128, debug: bool = False,
            tool_schemas: dict[str, dict] | None = None,
            ) -> Generator[tuple, Any, EpisodeResult]:
    """The AI_RUN_AGENT loop as a resumable generator.

    Yields ("llm", prompt, max_new_tokens[, grammar]) and
    ("tool", name, arguments); caller .send()s the string result back.
    Tool errors are sent as exceptions via .throw() or as
    ("__error__", msg) sentinel strings.  Returns EpisodeResult via
    StopIteration.value.

    tool_schemas ({tool -> MCP inputSchema}, from tools/list): when given,
    each LLM turn carries a models.grammar.TurnGrammar so the serving
    engine's constrained decode lets the MODEL choose the turn's action
    (tool call vs finish) — the reference's model-driven AI_RUN_AGENT
    control flow (LAB1-Walkthrough.md:155-181) under random-init weights.
    """
    policy = policy or ToolCallPolicy()
    transcript: list[str] = []
    ctx: dict = {"agent": agent, "observations": [], "calls": [],
                 "user_prompt": user_prompt}
    consecutive_failures = 0
    tool_calls = 0
    trace: list[dict] = []
    last_text = ""

    for it in range(agent.max_iterations):
        prompt = _build_prompt(agent, user_prompt, transcript)
        if tool_schemas is not None:
            from ..models.grammar import build_turn_grammar
            allowed = agent.tools.allowed_tools if agent.tools else None
            grammar = build_turn_grammar(tool_schemas, allowed, prompt,
                                         history=ctx["calls"])
            text = yield ("llm", prompt, max_new_tokens, grammar)
        else:
            text = yield ("llm", prompt, max_new_tokens)
        last_text = text or ""
        if debug:
            trace.append({"iteration": it, "model_output": last_text})
        action = policy(last_text, it, ctx)
        if isinstance(action, Finish):
            return EpisodeResult("SUCCESS", action.response, it + 1, tool_calls,
                                 0, trace)
        if isinstance(action, Continue):
            transcript.append(f"[thought] {action.note or last_text[:200]}")
            continue
        if isinstance(action, ToolCall):
            if agent.tools is not None and not agent.tools.allows(action.name):
                result = f"__error__ tool {action.name} not allowed"
            else:
                result = yield ("tool", action.name, action.arguments)
            tool_calls += 1
            if isinstance(result, str) and result.startswith("__error__"):
                consecutive_failures += 1
                if debug:
                    trace.append({"iteration": it, "tool": action.name,
                                  "error": result})
                if consecutive_failures >= agent.max_consecutive_failures:
                    return EpisodeResult(
                        "FAILED",
                        f"aborted after {consecutive_failures} consecutive "
                        f"tool failures: {result}",
                        it + 1, tool_calls, consecutive_failures, trace)
                transcript.append(f"[tool {action.name} failed] {result}")
            else:
                consecutive_failures = 0
                # only SUCCESSFUL calls leave the grammar's candidate set
                # (an identical retry of a FAILED call is legitimate and
                # bounded by max_consecutive_failures)
                ctx["calls"].append((action.name, action.arguments))
                ctx["observations"].append((action.name, result))
                if debug:
                    trace.append({"iteration": it, "tool": action.name,
                                  "result": str(result)[:500]})
                transcript.append(f"[observation from {action.name}] {result}")
    return EpisodeResult("FAILED", last_text, agent.max_iterations, tool_calls,
                         consecutive_failures, trace)


def _build_prompt(agent: AgentSpec, user_prompt: str,
                  transcript: list[str]) -> str:
    parts = [agent.prompt, "", user_prompt]
    if transcript:
        parts.append("")
        parts.extend(transcript)
    return "\n".join(parts)


# ---------------------------------------------------------------------------
# Synchronous drivers (CPU tests / single-record paths)
# ---------------------------------------------------------------------------


def drive_episode(ep: Generator, llm: Callable[[str, int], str],
                  tool: Callable[[str, dict], str]) -> EpisodeResult:
    """Run one episode to completion with direct (unbatched) calls."""
    try:
        req = ep.send(None)
        while True:
            if req[0] == "llm":
                if len(req) > 3 and req[3] is not None:
                    # grammar-capable backends take the TurnGrammar;
                    # plain 2-arg callables (stub LLMs) decode free
                    try:
                        result = llm(req[1], req[2], req[3])
                    except TypeError:
                        result = llm(req[1], req[2])
                else:
                    result = llm(req[1], req[2])
            elif req[0] == "tool":
                try:
                    result = tool(req[1], req[2])
                except Exception as e:
                    result = f"__error__ {e}"
            else:
                raise ValueError(f"unknown request {req[0]!r}")
            req = ep.send(result)
    except StopIteration as stop:
        return stop.value


def ai_tool_invoke(model_llm: Callable[[str, int], str],
                   tool: Callable[[str, dict], str], prompt: str,
                   tool_descriptions: dict[str, str],
                   debug: bool = False) -> dict:
    """AI_TOOL_INVOKE(model, prompt, MAP[], MAP[tool->desc], MAP[debug]):
    one LLM call + at most one tool round trip (LAB1-Walkthrough.md:80-92).
    Returns {tool -> result} plus the model response."""
    tool_block = "\n".join(f"- {name}: {desc}"
                           for name, desc in tool_descriptions.items())
    text = model_llm(f"{prompt}\n\nAvailable tools:\n{tool_block}", 128)
    out: dict[str, str] = {"response": text}
    raw = extract_tool_call_json(text or "")
    if raw is not None:
        try:
            call = json.loads(raw)
            name = call.get("name", "")
            if name in tool_descriptions:
                try:
                    out[name] = tool(name, call.get("arguments", {}) or {})
                except Exception as e:
                    out[name] = f"__error__ {e}"
        except json.JSONDecodeError:
            pass
    return out
