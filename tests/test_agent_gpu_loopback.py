"""Full-stack GPU loopback: HTTP signalling + RTP/UDP media + the real
SD-Turbo HIP pipeline on an MI355X — the closest offline stand-in for the
reference's manual OBS/browser verification."""
import asyncio

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(300)
def test_agent_gpu_loopback(monkeypatch):
    monkeypatch.setenv("WARMUP_FRAMES", "2")
    monkeypatch.setenv("DROP_FRAMES", "0")

    async def body():
        from aiohttp.test_utils import TestClient, TestServer

        from ai_rtc_agent_amd.agent import create_app
        from ai_rtc_agent_amd.config import sd_turbo_config
        from ai_rtc_agent_amd.media.codec import select_codec
        from ai_rtc_agent_amd.media.rtp import RtpPacketizer
        from ai_rtc_agent_amd.media.sdp import SessionDescription
        from ai_rtc_agent_amd.media import stun
        from ai_rtc_agent_amd.parallel.dispatch import PipelinePool
        from tests.test_tracks_loopback import (
            _ClientProto,
            _offer_sdp,
            _send_frame,
        )

        cfg = sd_turbo_config(device="cuda")
        pool = PipelinePool.create("stabilityai/sd-turbo", n_gpus=1, cfg=cfg)
        app = create_app(pool=pool, use_turn=False)
        http = TestClient(TestServer(app))
        await http.start_server()
        loop = asyncio.get_event_loop()

        pub_t, pub_p = await loop.create_datagram_endpoint(
            _ClientProto, local_addr=("127.0.0.1", 0))
        pub_port = pub_t.get_extra_info("sockname")[1]
        r = await http.post("/whip", data=_offer_sdp(pub_port),
                            headers={"Content-Type": "application/sdp"})
        assert r.status == 201
        srv_port = SessionDescription.parse(await r.text()).media[0].port
        pub_t.sendto(stun.make_binding_request("u:p", b"k"), ("127.0.0.1", srv_port))
        await asyncio.sleep(0.2)

        codec = select_codec()  # the default wire format (H.264 when built)
        pkz = RtpPacketizer(ssrc=42)
        # compressible frames (gradients): random 512² noise would make a
        # huge UDP burst that overflows loopback socket buffers; real
        # camera frames compress, synthetic ones must too
        base = torch.arange(512, dtype=torch.uint8).view(1, 512, 1).expand(512, 512, 3)
        frames = [(base.int() + 13 * i).clamp(0, 255).to(torch.uint8).contiguous()
                  for i in range(3)]

        async def send_frame(i):
            _send_frame(codec, pkz, pub_t, ("127.0.0.1", srv_port),
                        frames[i % 3], (i + 1) * 3000)
            await asyncio.sleep(0.005)

        sub_t, sub_p = await loop.create_datagram_endpoint(
            _ClientProto, local_addr=("127.0.0.1", 0))
        sub_port = sub_t.get_extra_info("sockname")[1]

        # publish a few frames first so the source track registers
        for i in range(3):
            await send_frame(i)
            await asyncio.sleep(0.05)

        r2 = await http.post("/whep", data=_offer_sdp(sub_port),
                             headers={"Content-Type": "application/sdp"})
        assert r2.status == 201, await r2.text()

        got = None
        for i in range(3, 120):
            await send_frame(i)
            try:
                got = await asyncio.wait_for(sub_p.frames.get(), timeout=0.5)
                break
            except asyncio.TimeoutError:
                continue
        assert got is not None, "no stylised frame reached the WHEP subscriber"
        assert got.shape == (512, 512, 3) and got.dtype == torch.uint8

        stats = (await (await http.get("/stats")).json())
        assert stats["per_replica"][0]["frames"] >= 1

        pub_t.close()
        sub_t.close()
        await http.close()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(asyncio.wait_for(body(), 280))
    finally:
        loop.close()
