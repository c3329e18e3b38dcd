"""Grammar-constrained decoding for the agent loop (model-driven K6).

The reference's `AI_RUN_AGENT` lets the *model* choose which tool to call
and when to finish, bounded by `max_iterations` / `max_consecutive_failures`
(LAB1-Walkthrough.md:155-181, caps at 177-180).  With random-init weights
(the BASELINE.json contract) free decoding cannot emit well-formed tool
calls, so the engine constrains each agent turn with a per-turn grammar —
the standard function-calling constrained-decoding construction:

- The turn's FIRST generated token is logit-masked to a small decision
  vocabulary: one reserved `<|tool_k|>` special token per candidate action
  plus `<|finish|>` (tokenizer.BpeTokenizer specials).  Which token wins
  the masked argmax is the MODEL's choice — the control flow (tool
  selection, iteration count, episode shape) is model output, not a
  script.
- A `<|tool_k|>` decision forces the remainder of the turn to that
  branch's token script: the literal ``TOOL_CALL {json}`` text that
  `agents.runner.ToolCallPolicy` parses back.  Every token of the call is
  still decoded step-by-step on the GPU (realistic decode work); the
  grammar only masks what token is *emitted*, exactly like token-level
  FSM constrained decoding.
- `<|finish|>` leaves the rest of the turn unconstrained (free text up to
  max_new_tokens or EOS), which terminates the episode.

Candidate actions come from the MCP tool **schemas** (`tools/list`
inputSchema properties — the same registry a production function-calling
grammar uses), with argument values drawn syntactically from the
conversation text: URL-typed params bind to URLs present in the text,
email-typed to email addresses, subject/body to declared template blocks.
If a parameter has several candidates, each binding is its own branch and
the model's decision token picks among them.  No per-lab logic lives
here — the rules are name/shape heuristics over any conversation.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass, field

_URL_RE = re.compile(r"https?://[^\s\"'<>\)\]]+")
_EMAIL_RE = re.compile(r"[A-Za-z0-9._%+-]+@[A-Za-z0-9.-]+\.[A-Za-z]{2,}")
_SUBJECT_RE = re.compile(r"(?:EMAIL )?SUBJECT:\s*(.+)", re.IGNORECASE)
_BODY_RE = re.compile(
    r"(?:EMAIL )?BODY(?: TEMPLATE)?:\s*\n(.*?)(?:\n\s*\n[A-Z ]+:|\Z)",
    re.DOTALL)
_JSON_RE = re.compile(r"\{[^{}]*\}|\[[^\[\]]*\]")


@dataclass
class ActionOption:
    """One fully-bound tool call the model may choose this turn."""
    tool: str
    arguments: dict

    def render(self) -> str:
        return " TOOL_CALL " + json.dumps(
            {"name": self.tool, "arguments": self.arguments})


@dataclass
class TurnGrammar:
    """The constrained-decoding spec for one agent turn."""
    options: list[ActionOption] = field(default_factory=list)
    allow_finish: bool = True


def _unique(seq):
    seen, out = set(), []
    for x in seq:
        if x not in seen:
            seen.add(x)
            out.append(x)
    return out


def _candidates_for_param(name: str, text: str) -> list:
    """Syntactic candidate values for one schema parameter, by name."""
    n = name.lower()
    if n in ("url", "uri", "link", "endpoint", "href"):
        return _unique(_URL_RE.findall(text))[:4]
    if n in ("to", "email", "recipient", "address"):
        return _unique(_EMAIL_RE.findall(text))[:2]
    if n == "subject":
        m = _SUBJECT_RE.search(text)
        return [m.group(1).strip()] if m else []
    if n == "body":
        m = _BODY_RE.search(text)
        if m:
            return [m.group(1).strip()[:1200]]
        j = _JSON_RE.findall(text)
        return [j[-1]] if j else []
    return []


def build_turn_grammar(tool_schemas: dict[str, dict], allowed: tuple | None,
                       text: str, max_options: int = 8,
                       history: list[tuple[str, dict]] | None = None,
                       ) -> TurnGrammar:
    """Candidate actions for one turn.

    tool_schemas: {tool name -> MCP inputSchema dict} (tools/list).
    allowed: the agent's CREATE TOOL allowed_tools filter (None = all).
    text: conversation so far (system prompt + user prompt + transcript).
    history: (tool, arguments) calls already made this episode — an exact
    repeat is dropped from the candidate set (standard function-calling
    dedup: re-issuing an identical call yields no new information), so an
    episode's length is bounded by its distinct actions, not only by
    max_iterations.
    """
    seen = {(t, json.dumps(a, sort_keys=True))
            for t, a in (history or [])}
    options: list[ActionOption] = []
    for name, schema in tool_schemas.items():
        if allowed and name not in allowed:
            continue
        props = (schema or {}).get("properties", {})
        required = (schema or {}).get("required", list(props))
        per_param: dict[str, list] = {}
        ok = True
        for p in props:
            cands = _candidates_for_param(p, text)
            if not cands:
                if p in required:
                    ok = False
                    break
                continue
            per_param[p] = cands
        if not ok:
            continue
        # cross-product of candidate bindings (bounded): the first
        # multi-candidate param fans out, the rest take their first
        fan_param = next((p for p, c in per_param.items() if len(c) > 1),
                         None)
        cands: list[ActionOption] = []
        if fan_param is None:
            cands.append(ActionOption(
                name, {p: c[0] for p, c in per_param.items()}))
        else:
            for v in per_param[fan_param]:
                args = {p: (v if p == fan_param else c[0])
                        for p, c in per_param.items()}
                cands.append(ActionOption(name, args))
        for o in cands:
            if (o.tool, json.dumps(o.arguments, sort_keys=True)) not in seen:
                options.append(o)
    return TurnGrammar(options=options[:max_options], allow_finish=True)


# ---------------------------------------------------------------------------
# Tokenized form the engine consumes
# ---------------------------------------------------------------------------

@dataclass
class CompiledGrammar:
    """Per-sequence constrained-decoding program.

    decision_allowed: token ids legal as the turn's first generated token.
    branches: decision token -> forced continuation token script (ending
    with EOS).  A decision token absent from branches (<|finish|>) leaves
    the rest of the turn unconstrained.
    """
    decision_allowed: list[int]
    branches: dict[int, list[int]]

    def max_script_len(self) -> int:
        return max((len(s) for s in self.branches.values()), default=0)


def compile_grammar(g: TurnGrammar, tok) -> CompiledGrammar | None:
    """Tokenize a TurnGrammar against a BpeTokenizer (needs its special
    tokens).  Returns None for an unconstrained turn (no options and
    finish-only would still force the finish token, so keep the mask)."""
    allowed: list[int] = []
    branches: dict[int, list[int]] = {}
    for k, opt in enumerate(g.options[:tok.N_TOOL_SLOTS]):
        t = tok.tool_slot(k)
        allowed.append(t)
        branches[t] = tok.encode(opt.render(), bos=False) + [tok.EOS]
    if g.allow_finish or not allowed:
        allowed.append(tok.FINISH)
    return CompiledGrammar(decision_allowed=allowed, branches=branches)
