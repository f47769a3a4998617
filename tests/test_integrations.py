"""LLM + MCP serving integrations (reference analog:
sdk/src/beta9/abstractions/integrations/{vllm,fastmcp}.py)."""

import asyncio

import httpx
import pytest

from beta9_amd.integrations import LLM, LLMArgs, MCPServer
from beta9_amd.integrations.llm import LLMEngine


def tiny_args(**kw):
    base = dict(
        model="synthetic-llama", hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, max_model_len=256,
        max_num_seqs=4,
    )
    base.update(kw)
    return LLMArgs(**base)


@pytest.mark.asyncio
async def test_llm_engine_batched_decode():
    eng = LLMEngine(tiny_args())
    await eng.start()
    try:
        outs = await asyncio.gather(*[
            eng.generate(f"hello {i}", max_tokens=4) for i in range(6)
        ])
        assert len(outs) == 6
        for o in outs:
            assert isinstance(o, str)
    finally:
        await eng.stop()


@pytest.mark.asyncio
async def test_llm_openai_api_surface():
    llm = LLM(args=tiny_args())
    app = llm.app()
    async with httpx.AsyncClient(
        transport=httpx.ASGITransport(app=app), base_url="http://t"
    ) as client:
        r = await client.get("/v1/models")
        assert r.status_code == 200 and r.json()["data"][0]["id"] == "synthetic-llama"
        r = await client.post("/v1/completions", json={"prompt": "hi", "max_tokens": 3})
        assert r.status_code == 200
        body = r.json()
        assert body["object"] == "text_completion" and len(body["choices"]) == 1
        r = await client.post(
            "/v1/chat/completions",
            json={"messages": [{"role": "user", "content": "hi"}], "max_tokens": 3},
        )
        assert r.json()["choices"][0]["message"]["role"] == "assistant"


@pytest.mark.asyncio
async def test_mcp_jsonrpc_flow():
    srv = MCPServer(name="t")

    @srv.tool
    def add(a: int, b: int) -> int:
        """Add two numbers."""
        return a + b

    @srv.tool
    def boom():
        """Always fails."""
        raise ValueError("nope")

    app = srv.app()
    async with httpx.AsyncClient(
        transport=httpx.ASGITransport(app=app), base_url="http://t"
    ) as client:
        r = await client.post("/mcp", json={"jsonrpc": "2.0", "id": 1, "method": "initialize"})
        init = r.json()["result"]
        assert init["serverInfo"]["name"] == "t" and "tools" in init["capabilities"]
        r = await client.post("/mcp", json={"jsonrpc": "2.0", "id": 2, "method": "tools/list"})
        tools = {t["name"]: t for t in r.json()["result"]["tools"]}
        assert tools["add"]["inputSchema"]["required"] == ["a", "b"]
        assert tools["add"]["inputSchema"]["properties"]["a"]["type"] == "integer"
        r = await client.post("/mcp", json={
            "jsonrpc": "2.0", "id": 3, "method": "tools/call",
            "params": {"name": "add", "arguments": {"a": 2, "b": 3}},
        })
        res = r.json()["result"]
        assert res["isError"] is False and res["content"][0]["text"] == "5"
        # tool error -> isError result, not protocol error
        r = await client.post("/mcp", json={
            "jsonrpc": "2.0", "id": 4, "method": "tools/call", "params": {"name": "boom"},
        })
        assert r.json()["result"]["isError"] is True
        # unknown method -> -32601
        r = await client.post("/mcp", json={"jsonrpc": "2.0", "id": 5, "method": "bogus"})
        assert r.json()["error"]["code"] == -32601


@pytest.mark.gpu
@pytest.mark.asyncio
async def test_llm_engine_gpu_bf16_decode():
    import torch

    assert torch.cuda.is_available()
    eng = LLMEngine(tiny_args(hidden_size=256, intermediate_size=512, num_hidden_layers=4))
    assert eng.device == "cuda" and eng.dtype == torch.bfloat16
    await eng.start()
    try:
        outs = await asyncio.gather(*[
            eng.generate(f"prompt {i}", max_tokens=8) for i in range(8)
        ])
        assert len(outs) == 8
    finally:
        await eng.stop()


@pytest.mark.asyncio
async def test_llm_served_through_platform():
    """The LLM integration deployed as an asgi stub: OpenAI-surface requests
    route gateway HTTP -> endpoint buffer -> uvicorn in the container."""
    import json as _json

    from tests.helpers import cluster

    async with cluster(n_workers=1) as c:
        import llm_app  # fixture module: builds a tiny synthetic LLM app

        import beta9_amd

        ep = beta9_amd.Endpoint(llm_app.make_app, cpu=1000, memory=2048)
        ep.stub_kind = "asgi"
        stub_id = await asyncio.to_thread(ep.prepare_runtime)
        # wait for the uvicorn container to come up and serve
        base = f"http://127.0.0.1:{c.gateway.http_port}"
        headers = {"authorization": f"Bearer {c.token}"}
        deadline = 60.0
        async with httpx.AsyncClient() as hc:
            r = None
            import time as _time

            t0 = _time.monotonic()
            while _time.monotonic() - t0 < deadline:
                r = await hc.post(
                    f"{base}/asgi/id/{stub_id}/v1/completions",
                    headers=headers,
                    json={"prompt": "hi", "max_tokens": 3},
                    timeout=30.0,
                )
                if r.status_code == 200:
                    break
                await asyncio.sleep(1.0)
            assert r is not None and r.status_code == 200, r.text if r else "no response"
            body = r.json()
            payload = body.get("result", body)
            if isinstance(payload, str):
                payload = _json.loads(payload)
            assert payload["object"] == "text_completion", payload
