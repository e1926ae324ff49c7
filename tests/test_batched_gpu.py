"""GPU end-to-end of the multi-stream batched serving tier: two concurrent
/offer sessions share ONE engine replica (streams_per_replica=2) on a real
MI355X — frames of both streams travel one batched engine call
(profiles/batching_ab.md is the measured motivation)."""
import asyncio

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(300)
def test_two_offer_sessions_one_batched_replica(monkeypatch):
    monkeypatch.setenv("WARMUP_FRAMES", "0")
    monkeypatch.setenv("DROP_FRAMES", "0")

    async def body():
        import json

        from aiohttp.test_utils import TestClient, TestServer

        from ai_rtc_agent_amd.agent import create_app
        from ai_rtc_agent_amd.config import sd_turbo_config
        from ai_rtc_agent_amd.media.codec import select_codec
        from ai_rtc_agent_amd.media.rtp import RtpPacketizer
        from ai_rtc_agent_amd.media.sdp import SessionDescription
        from ai_rtc_agent_amd.media import stun
        from ai_rtc_agent_amd.parallel.dispatch import PipelinePool
        from ai_rtc_agent_amd.parallel.batching import BatchedPipeline
        from tests.test_tracks_loopback import (
            _ClientProto,
            _offer_sdp,
            _send_frame,
        )

        cfg = sd_turbo_config(device="cuda")
        pool = PipelinePool.create("stabilityai/sd-turbo", n_gpus=1, cfg=cfg,
                                   streams_per_replica=2)
        assert isinstance(pool.active()[0], BatchedPipeline)
        app = create_app(pool=pool, use_turn=False)
        http = TestClient(TestServer(app))
        await http.start_server()
        loop = asyncio.get_event_loop()

        base = torch.arange(512, dtype=torch.uint8).view(1, 512, 1)
        frames = [(base.expand(512, 512, 3).int() + 17 * i).clamp(0, 255)
                  .to(torch.uint8).contiguous() for i in range(2)]

        sessions = []
        for si in range(2):
            t, p = await loop.create_datagram_endpoint(
                _ClientProto, local_addr=("127.0.0.1", 0))
            port = t.get_extra_info("sockname")[1]
            r = await http.post("/offer", json={
                "room_id": f"r{si}",
                "offer": {"sdp": _offer_sdp(port), "type": "offer"}})
            assert r.status == 200
            ans = SessionDescription.parse((await r.json())["sdp"])
            srv_port = ans.media[0].port
            t.sendto(stun.make_binding_request("u:p", b"k"),
                     ("127.0.0.1", srv_port))
            sessions.append({"t": t, "p": p, "srv": ("127.0.0.1", srv_port),
                             "codec": select_codec(),
                             "pkz": RtpPacketizer(ssrc=100 + si)})
        await asyncio.sleep(0.3)

        # both sessions occupy slots of the SAME batched replica
        st = pool.stats()
        assert st["per_replica"][0]["batched_slots"] == 2
        assert st["per_replica"][0]["active_streams"] == 2

        got = [None, None]
        for i in range(120):
            for si, s in enumerate(sessions):
                _send_frame(s["codec"], s["pkz"], s["t"], s["srv"],
                            frames[si], (i + 1) * 3000)
            for si, s in enumerate(sessions):
                if got[si] is None:
                    try:
                        got[si] = await asyncio.wait_for(
                            s["p"].frames.get(), timeout=0.3)
                    except asyncio.TimeoutError:
                        pass
            if all(g is not None for g in got):
                break
        assert all(g is not None for g in got), \
            "both batched sessions must produce stylised frames"
        for g in got:
            assert g.shape == (512, 512, 3)
        # per-stream latency accounting is live
        st = pool.stats()
        lats = [x for x in st["per_replica"][0]["per_stream"] if x["frames"]]
        assert lats and all(x["p50_ms"] is not None for x in lats)

        for s in sessions:
            s["t"].close()
        await http.close()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(asyncio.wait_for(body(), 280))
    finally:
        loop.close()
