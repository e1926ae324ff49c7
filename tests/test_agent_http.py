"""Signalling API tests against a stub pipeline (no model load).

Covers the reference's route surface (agent.py:466-472): /offer, /whip,
/whep (401 ordering), /config, health, plus our /stats. SURVEY.md §4 (d).
"""
import asyncio
import json

import pytest
import torch
from aiohttp.test_utils import TestClient, TestServer

from ai_rtc_agent_amd.agent import create_app
from ai_rtc_agent_amd.parallel.dispatch import PipelinePool

OFFER_SDP = "\r\n".join([
    "v=0", "o=- 1 2 IN IP4 127.0.0.1", "s=-", "t=0 0",
    "m=video 51000 UDP/TLS/RTP/SAVPF 97",
    "a=ice-ufrag:abcd", "a=ice-pwd:efghijklmnop", "a=mid:0", "a=sendrecv",
    "a=rtpmap:97 H264/90000",
    "a=candidate:1 1 udp 2130706431 127.0.0.1 51000 typ host",
]) + "\r\n"


class StubPipeline:
    def __init__(self):
        self.prompt = None
        self.t_index = None
        self.calls = 0

    def __call__(self, frame):
        self.calls += 1
        return frame

    def update_prompt(self, p):
        self.prompt = p

    def update_t_index_list(self, t):
        self.t_index = list(t)

    def stats(self):
        return {"frames": self.calls}


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


async def make_client(stub):
    app = create_app(pool=PipelinePool.single(stub), use_turn=False)
    client = TestClient(TestServer(app))
    await client.start_server()
    return client


def test_health_and_stats():
    async def body():
        stub = StubPipeline()
        client = await make_client(stub)
        r = await client.get("/")
        assert r.status == 200 and await r.text() == "OK"
        r = await client.get("/stats")
        data = await r.json()
        assert data["replicas"] == 1
        await client.close()

    run(body())


def test_whep_requires_publisher():
    async def body():
        client = await make_client(StubPipeline())
        r = await client.post("/whep", data=OFFER_SDP, headers={"Content-Type": "application/sdp"})
        assert r.status == 401  # reference agent.py:218-220
        await client.close()

    run(body())


def test_whip_then_whep():
    async def body():
        client = await make_client(StubPipeline())
        r = await client.post("/whip", data=OFFER_SDP, headers={"Content-Type": "application/sdp"})
        assert r.status == 201
        # per-session WHIP resource URL (the spec's DELETE target)
        assert r.headers["Location"].startswith("/whip/")
        whip_resource = r.headers["Location"]
        assert r.content_type == "application/sdp"
        answer = await r.text()
        assert "H264" in answer and "a=candidate" in answer

        # publisher stored a source track only after media arrives; whep is
        # still 401 until the first track (matching the reference's
        # source_track gate)
        r2 = await client.post("/whep", data=OFFER_SDP, headers={"Content-Type": "application/sdp"})
        assert r2.status == 401

        # DELETE on the per-session resource URL (WHIP spec behaviour)
        r3 = await client.delete(whip_resource)
        assert r3.status == 200
        r3b = await client.delete(whip_resource)
        assert r3b.status == 404  # already gone
        # bare DELETE /whip stays for reference-parity clients
        r3c = await client.delete("/whip")
        assert r3c.status == 200
        await client.close()

    run(body())


def test_offer_and_config():
    async def body():
        stub = StubPipeline()
        client = await make_client(stub)
        r = await client.post(
            "/offer",
            json={"room_id": "r1", "offer": {"sdp": OFFER_SDP, "type": "offer"}},
        )
        assert r.status == 200
        data = await r.json()
        assert data["type"] == "answer" and "m=video" in data["sdp"]

        r = await client.post("/config", json={"prompt": "new style", "t_index_list": [1, 2, 3]})
        assert r.status == 200
        assert stub.prompt == "new style"
        assert stub.t_index == [1, 2, 3]
        await client.close()

    run(body())


def test_pool_assignment():
    a, b = StubPipeline(), StubPipeline()
    pool = PipelinePool([a, b])
    p1 = pool.assign("s1")
    p2 = pool.assign("s2")
    assert {id(p1), id(p2)} == {id(a), id(b)}, "least-loaded spreads sessions"
    assert pool.assign("s1") is p1, "sticky affinity"
    pool.release("s1")
    p3 = pool.assign("s3")
    assert p3 is p1, "freed replica is reused"


def test_metrics_endpoint():
    async def body():
        client = await make_client(StubPipeline())
        r = await client.get("/metrics")
        assert r.status == 200
        text = await r.text()
        assert "airtc_replicas 1" in text
        assert "airtc_frames_total" in text
        assert "airtc_fps" in text
        await client.close()

    run(body())
