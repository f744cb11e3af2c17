"""Gateway endpoint-picker service (EPP-equivalent) tests."""

import asyncio

import httpx

from production_stack_amd.gateway.picker_service import build_picker_app


def run(app, fn):
    async def go():
        transport = httpx.ASGITransport(app=app)
        async with httpx.AsyncClient(
            transport=transport, base_url="http://p"
        ) as client:
            await fn(client)

    asyncio.run(go())


def test_roundrobin_picker():
    app = build_picker_app("roundrobin")
    picks = []

    async def go(client):
        for _ in range(4):
            r = await client.post(
                "/pick",
                json={"endpoints": ["http://a", "http://b"], "body": {}},
            )
            picks.append(r.json()["endpoint"])

    run(app, go)
    assert picks == ["http://a", "http://b", "http://a", "http://b"]


def test_prefixaware_picker_affinity():
    app = build_picker_app("prefixaware", min_match=8)
    body = {"prompt": "shared long prefix " * 20}
    picks = []

    async def go(client):
        for _ in range(3):
            r = await client.post(
                "/pick",
                json={
                    "endpoints": ["http://a", "http://b"],
                    "body": body,
                },
            )
            picks.append(r.json()["endpoint"])

    run(app, go)
    assert len(set(picks[1:])) == 1  # sticky after first insert
