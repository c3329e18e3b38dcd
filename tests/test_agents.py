"""AI_RUN_AGENT / AI_TOOL_INVOKE semantics contracts
(LAB1-Walkthrough.md:155-181: iteration cap, consecutive-failure abort,
(status, response) result, debug traces)."""

from quickstart_streaming_agents_amd.agents.runner import (
    AgentSpec, Continue, Finish, ToolCall, ToolSet, ai_tool_invoke,
    drive_episode, episode)
from quickstart_streaming_agents_amd.agents.schedule import run_episodes


def _llm(prompts, toks, conv=None):
    return [f"resp:{len(p)}" for p in prompts]


def llm1(prompt, toks):
    return f"resp:{len(prompt)}"


def test_consecutive_failures_abort():
    agent = AgentSpec("a", "m", "sys", ToolSet("t", allowed_tools=("x",)),
                      max_iterations=10, max_consecutive_failures=2)
    calls = []

    def policy(text, it, ctx):
        calls.append(it)
        return ToolCall("x", {})

    def tool(name, args):
        raise RuntimeError("down")

    res = drive_episode(episode(agent, "go", policy=policy), llm1, tool)
    assert res.status == "FAILED"
    assert "consecutive" in res.response
    assert res.failures == 2
    assert len(calls) == 2          # aborted on the 2nd straight failure


def test_failure_counter_resets_on_success():
    agent = AgentSpec("a", "m", "sys", ToolSet("t", allowed_tools=("x",)),
                      max_iterations=10, max_consecutive_failures=2)
    state = {"n": 0}

    def tool(name, args):
        state["n"] += 1
        if state["n"] % 2 == 1:     # fail, succeed, fail, succeed...
            raise RuntimeError("flaky")
        return "ok"

    def policy(text, it, ctx):
        if len(ctx["observations"]) >= 2:
            return Finish("done")
        return ToolCall("x", {})

    res = drive_episode(episode(agent, "go", policy=policy), llm1, tool)
    assert res.status == "SUCCESS"
    assert res.response == "done"
    assert state["n"] == 4          # two failures interleaved, never 2 straight


def test_max_iterations_exhaustion():
    agent = AgentSpec("a", "m", "sys", None, max_iterations=3)
    res = drive_episode(
        episode(agent, "go", policy=lambda t, i, c: Continue("thinking")),
        llm1, lambda n, a: "")
    assert res.status == "FAILED"
    assert res.iterations == 3


def test_disallowed_tool_counts_as_failure():
    agent = AgentSpec("a", "m", "sys", ToolSet("t", allowed_tools=("ok",)),
                      max_iterations=5, max_consecutive_failures=2)
    res = drive_episode(
        episode(agent, "go", policy=lambda t, i, c: ToolCall("evil", {})),
        llm1, lambda n, a: "never called")
    assert res.status == "FAILED"
    assert "not allowed" in res.response


def test_debug_trace_collection():
    agent = AgentSpec("a", "m", "sys", ToolSet("t", allowed_tools=("x",)))

    def policy(text, it, ctx):
        if ctx["observations"]:
            return Finish("done")
        return ToolCall("x", {"k": 1})

    res = drive_episode(episode(agent, "go", policy=policy, debug=True),
                        llm1, lambda n, a: "obs")
    kinds = [set(t) for t in res.trace]
    assert any("model_output" in k for k in kinds)
    assert any("tool" in k for k in kinds)


def test_ai_tool_invoke_single_round():
    tools = {"http_get": "fetch a url"}

    def model(prompt, toks):
        assert "http_get" in prompt          # tool schema injected
        return 'TOOL_CALL {"name": "http_get", "arguments": {"url": "u"}}'

    out = ai_tool_invoke(model, lambda n, a: f"GET {a['url']}", "do it",
                         tools)
    assert out["http_get"] == "GET u"
    assert "response" in out

    # unknown tool in the call -> no invocation, response still present
    out2 = ai_tool_invoke(
        lambda p, t: 'TOOL_CALL {"name": "nope", "arguments": {}}',
        lambda n, a: "x", "do it", tools)
    assert "nope" not in out2


def test_scheduler_mixed_tool_and_llm_rounds():
    agent = AgentSpec("a", "m", "sys", ToolSet("t", allowed_tools=("x",)))

    def mk_policy(want_tool):
        def policy(text, it, ctx):
            if want_tool and not ctx["observations"]:
                return ToolCall("x", {})
            return Finish(f"fin{it}")
        return policy

    eps = [episode(agent, f"p{i}", policy=mk_policy(i % 2 == 0))
           for i in range(6)]
    results = run_episodes(eps, _llm, lambda n, a: "obs")
    assert all(r.status == "SUCCESS" for r in results)
    assert [r.tool_calls for r in results] == [1, 0, 1, 0, 1, 0]
