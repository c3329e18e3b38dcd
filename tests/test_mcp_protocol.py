"""MCP protocol-level contracts (streamable-HTTP JSON-RPC surface the
reference's CREATE CONNECTION ... 'type'='MCP_SERVER' path speaks:
LAB1-Walkthrough.md MCP section): initialize handshake, tools/list schema,
tools/call success + isError mapping, unknown-method errors, and the stub
fixture endpoints (competitor site, vessel catalog, dispatch API)."""

import json
import urllib.request

import pytest

from quickstart_streaming_agents_amd.agents.mcp import (
    COMPETITOR_PRICES, McpClient, McpError, StubMcpServer)


@pytest.fixture(scope="module")
def server():
    with StubMcpServer() as srv:
        yield srv


def test_initialize_handshake(server):
    cl = McpClient(server.mcp_endpoint)
    info = cl.initialize()
    assert info["serverInfo"]["name"] == "qsa-stub-mcp"
    assert "tools" in info["capabilities"]


def test_tools_list_schemas(server):
    tools = McpClient(server.mcp_endpoint).tools_list()
    by_name = {t["name"]: t for t in tools}
    assert set(by_name) == {"http_get", "http_post", "send_email"}
    assert by_name["http_get"]["inputSchema"]["required"] == ["url"]
    assert set(by_name["send_email"]["inputSchema"]["required"]) == {
        "to", "subject", "body"}


def test_unknown_method_raises(server):
    with pytest.raises(McpError, match="unknown method"):
        McpClient(server.mcp_endpoint)._call("resources/list")


def test_unknown_tool_maps_iserror(server):
    with pytest.raises(McpError, match="unknown tool"):
        McpClient(server.mcp_endpoint).tools_call("rm_rf", {})


def test_http_get_competitor_prices(server):
    cl = McpClient(server.mcp_endpoint)
    html = cl.tools_call("http_get", {"url": server.base_url + "/competitor"})
    for name, price in COMPETITOR_PRICES.items():
        assert name in html and f"${price:.2f}" in html


def test_send_email_records(server):
    cl = McpClient(server.mcp_endpoint)
    before = len(server.emails)
    out = cl.tools_call("send_email", {
        "to": "buyer@example.com", "subject": "Price match",
        "body": "Approved"})
    assert "buyer@example.com" in out
    assert server.emails[before]["subject"] == "Price match"


def test_http_post_dispatch_roundtrip(server):
    cl = McpClient(server.mcp_endpoint)
    before = len(server.dispatches)
    out = cl.tools_call("http_post", {
        "url": server.base_url + "/api/dispatch",
        "body": {"boats": ["BOAT-01", "BOAT-02"], "zone": "French Quarter"}})
    resp = json.loads(out)
    assert resp["status"] == "dispatched" and resp["count"] == 2
    assert server.dispatches[before]["zone"] == "French Quarter"


def test_vessel_catalog_endpoint(server):
    with urllib.request.urlopen(server.base_url + "/api/vessel_catalog",
                                timeout=10) as r:
        vessels = json.loads(r.read())["vessels"]
    assert len(vessels) == 10
    assert all(v["status"] == "available" for v in vessels)


def test_request_ids_increment_and_404(server):
    cl = McpClient(server.mcp_endpoint)
    cl.tools_list()
    cl.tools_list()
    assert cl._id >= 2
    with pytest.raises(urllib.error.HTTPError):
        urllib.request.urlopen(server.base_url + "/nope", timeout=10)


def test_concurrent_tool_call_burst(server):
    """Hundreds of concurrent episodes hit the stub MCP server at once
    (the bench's burst shape): every call must succeed — backlog 256 +
    client retry absorb connection churn (no dropped connections)."""
    import threading

    from quickstart_streaming_agents_amd.agents.mcp import McpClient
    errors = []
    results = []
    lock = threading.Lock()

    def worker(i):
        try:
            cl = McpClient(server.mcp_endpoint)
            out = cl.tools_call("send_email", {
                "to": f"u{i}@example.com", "subject": "s", "body": "b"})
            with lock:
                results.append(out)
        except Exception as e:        # noqa: BLE001
            with lock:
                errors.append(repr(e))

    threads = [threading.Thread(target=worker, args=(i,))
               for i in range(120)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    assert not errors, errors[:3]
    assert len(results) == 120
    assert len(server.emails) >= 120
