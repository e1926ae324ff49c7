import asyncio, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("WARMUP_FRAMES", "2")
import torch

async def main():
    from aiohttp.test_utils import TestClient, TestServer
    from ai_rtc_agent_amd.agent import create_app
    from ai_rtc_agent_amd.config import sd_turbo_config
    from ai_rtc_agent_amd.media.codec import SoftwareCodec
    from ai_rtc_agent_amd.media.rtp import RtpPacketizer
    from ai_rtc_agent_amd.media.sdp import SessionDescription
    from ai_rtc_agent_amd.media import stun
    from ai_rtc_agent_amd.parallel.dispatch import PipelinePool
    from tests.test_tracks_loopback import _ClientProto, _offer_sdp

    if torch.cuda.is_available():
        cfg = sd_turbo_config(device="cuda")
        pool = PipelinePool.create("stabilityai/sd-turbo", n_gpus=1, cfg=cfg)
    else:
        pool = PipelinePool.create("x", 1)
    print("pool ready", flush=True)
    app = create_app(pool=pool, use_turn=False)
    http = TestClient(TestServer(app)); await http.start_server()
    loop = asyncio.get_event_loop()
    pub_t, pub_p = await loop.create_datagram_endpoint(_ClientProto, local_addr=("127.0.0.1", 0))
    pub_port = pub_t.get_extra_info("sockname")[1]
    r = await http.post("/whip", data=_offer_sdp(pub_port), headers={"Content-Type": "application/sdp"})
    srv_port = SessionDescription.parse(await r.text()).media[0].port
    pub_t.sendto(stun.make_binding_request("u:p", b"k"), ("127.0.0.1", srv_port))
    await asyncio.sleep(0.2)
    codec = SoftwareCodec(); pkz = RtpPacketizer(ssrc=42)
    base = torch.arange(512, dtype=torch.uint8).view(1,512,1).expand(512,512,3)
    frames = [(base.int() + 13*i).clamp(0,255).to(torch.uint8).contiguous() for i in range(3)]
    async def send_frame(i):
        pkts = pkz.packetize(codec.encode(frames[i%3]), timestamp=i*3000)
        for j,pkt in enumerate(pkts):
            pub_t.sendto(pkt.serialize(), ("127.0.0.1", srv_port))
            if j % 40 == 39: await asyncio.sleep(0.002)
    sub_t, sub_p = await loop.create_datagram_endpoint(_ClientProto, local_addr=("127.0.0.1", 0))
    sub_port = sub_t.get_extra_info("sockname")[1]
    for i in range(3):
        await send_frame(i); await asyncio.sleep(0.05)
    st = app["state"]
    print("source_track:", st["source_track"], flush=True)
    r2 = await http.post("/whep", data=_offer_sdp(sub_port), headers={"Content-Type": "application/sdp"})
    print("whep:", r2.status, flush=True)
    whep_pcs = [pc for pc in st["pcs"] if pc is not st.get("whip_pc")]
    got = None
    for i in range(3, 60):
        await send_frame(i)
        try:
            got = await asyncio.wait_for(sub_p.frames.get(), timeout=0.5)
            print("GOT frame at iter", i, got.shape, flush=True)
            break
        except asyncio.TimeoutError:
            if i % 10 == 0:
                stats = (await (await http.get("/stats")).json())
                pc = whep_pcs[0] if whep_pcs else None
                print(f"iter {i}: stats={stats['per_replica'][0].get('frames')} "
                      f"whep_state={getattr(pc,'connection_state',None)} "
                      f"whep_remote={getattr(pc,'_remote_addr',None)} "
                      f"sender_task={getattr(pc,'_sender_task',None) is not None}", flush=True)
    print("RESULT:", None if got is None else "OK", flush=True)
    pub_t.close(); sub_t.close(); await http.close()

asyncio.new_event_loop().run_until_complete(main())
