"""Track adapter behaviour + full loopback media test over localhost UDP.

Loopback (SURVEY.md §4 item d): a synthetic publisher WHIPs into the agent,
frames flow RTP -> codec -> pipeline -> RTP back to a WHEP subscriber, all
over real sockets on 127.0.0.1 with a stub pipeline.
"""
import asyncio
import os

import pytest
import torch

from ai_rtc_agent_amd.media.codec import SoftwareCodec
from ai_rtc_agent_amd.media.rtp import RtpDefragmenter, RtpPacket, RtpPacketizer
from ai_rtc_agent_amd.media.sdp import SessionDescription
from ai_rtc_agent_amd.media import stun
from ai_rtc_agent_amd.media.tracks import QueueTrack, VideoFrame, VideoStreamTrack


def run(coro, timeout=30):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(asyncio.wait_for(coro, timeout))
    finally:
        loop.close()


# ---------------------------------------------------------------------------
# VideoStreamTrack contract (reference lib/tracks.py:20-38)
# ---------------------------------------------------------------------------

def test_warmup_frames_discarded(monkeypatch):
    monkeypatch.setenv("WARMUP_FRAMES", "3")
    monkeypatch.setenv("DROP_FRAMES", "0")

    async def body():
        src = QueueTrack(maxsize=64)
        for i in range(10):
            src.push(VideoFrame(tensor=torch.full((2, 2, 3), i, dtype=torch.uint8), pts=i))
        calls = []

        def pipe(t):
            calls.append(int(t[0, 0, 0]))
            return t + 100

        track = VideoStreamTrack(src, pipe)
        outs = [await track.recv() for _ in range(5)]
        # first 3 pulls ran the pipeline but returned SOURCE frames
        assert [int(o.tensor[0, 0, 0]) for o in outs[:3]] == [0, 1, 2]
        assert track.warmed_up
        # subsequent pulls return processed frames
        assert [int(o.tensor[0, 0, 0]) for o in outs[3:]] == [103, 104]
        assert calls == [0, 1, 2, 3, 4]

    run(body())


def test_drop_frames(monkeypatch):
    monkeypatch.setenv("WARMUP_FRAMES", "0")
    monkeypatch.setenv("DROP_FRAMES", "2")

    async def body():
        src = QueueTrack(maxsize=64)
        for i in range(12):
            src.push(VideoFrame(tensor=torch.full((2, 2, 3), i, dtype=torch.uint8), pts=i))
        track = VideoStreamTrack(src, lambda t: t)
        o1 = await track.recv()
        o2 = await track.recv()
        # 2 source frames dropped per output (reference lib/tracks.py:27-31)
        assert int(o1.tensor[0, 0, 0]) == 2
        assert int(o2.tensor[0, 0, 0]) == 5

    run(body())


def test_pts_preserved(monkeypatch):
    monkeypatch.setenv("WARMUP_FRAMES", "0")
    monkeypatch.setenv("DROP_FRAMES", "0")

    async def body():
        src = QueueTrack()
        src.push(VideoFrame(tensor=torch.zeros(2, 2, 3, dtype=torch.uint8), pts=777))
        track = VideoStreamTrack(src, lambda t: t)
        out = await track.recv()
        assert out.pts == 777  # reference lib/pipeline.py:90-93

    run(body())


# ---------------------------------------------------------------------------
# loopback over real UDP sockets
# ---------------------------------------------------------------------------

class _ClientProto(asyncio.DatagramProtocol):
    """Test-side WebRTC client speaking the agent's default wire format
    (select_codec: standard H.264 over RFC 6184 when the native extension
    is built, RAWZ generic fragmentation otherwise)."""

    def __init__(self):
        from ai_rtc_agent_amd.media.codec import select_codec

        self.transport = None
        self.frames = asyncio.Queue()
        self.defrag = RtpDefragmenter()
        self.codec = select_codec(role="decode")
        self._au = []

    def connection_made(self, transport):
        from ai_rtc_agent_amd.media.rtc import tune_socket_buffers

        self.transport = transport
        tune_socket_buffers(transport)

    def datagram_received(self, data, addr):
        if stun.is_stun(data):
            return
        try:
            pkt = RtpPacket.parse(data)
        except ValueError:
            return
        if getattr(self.codec, "rtp_mode", "raw") == "rfc6184":
            from ai_rtc_agent_amd.media.h264 import H264Depacketizer, join_annexb

            self._au.append(pkt.payload)
            if not pkt.marker:
                return
            dp = H264Depacketizer()
            nals = [n for n in (dp.push(p) for p in self._au) if n is not None]
            self._au = []
            if not nals:
                return
            buf = join_annexb(nals)
        else:
            buf = self.defrag.push(pkt)
            if buf is None:
                return
        t = self.codec.decode(buf)
        if t is not None:
            self.frames.put_nowait(t)


def _send_frame(codec, pkz, transport, addr, frame, timestamp):
    """Packetize one encoded frame the way PeerConnection's sender does
    (rfc6184 single-NAL/FU-A for H.264 codecs, generic otherwise)."""
    data = codec.encode(frame)
    if getattr(codec, "rtp_mode", "raw") == "rfc6184":
        from ai_rtc_agent_amd.media.h264 import packetize_h264, split_annexb

        payloads = packetize_h264(split_annexb(data))
        for i, pl in enumerate(payloads):
            pkt = RtpPacket(
                payload_type=pkz.payload_type, sequence_number=pkz._seq,
                timestamp=timestamp, ssrc=pkz.ssrc,
                marker=1 if i == len(payloads) - 1 else 0, payload=pl)
            pkz._seq = (pkz._seq + 1) & 0xFFFF
            transport.sendto(pkt.serialize(), addr)
    else:
        for pkt in pkz.packetize(data, timestamp=timestamp):
            transport.sendto(pkt.serialize(), addr)


def _offer_sdp(port: int) -> str:
    return "\r\n".join([
        "v=0", "o=- 1 2 IN IP4 127.0.0.1", "s=-", "t=0 0",
        f"m=video {port} UDP/TLS/RTP/SAVPF 97",
        "a=ice-ufrag:testu", "a=ice-pwd:testpw0123456789", "a=mid:0", "a=sendrecv",
        "a=rtpmap:97 H264/90000",
        f"a=candidate:1 1 udp 2130706431 127.0.0.1 {port} typ host",
    ]) + "\r\n"


@pytest.mark.timeout(60)
def test_loopback_whip_whep(monkeypatch):
    monkeypatch.setenv("WARMUP_FRAMES", "0")
    monkeypatch.setenv("DROP_FRAMES", "0")

    async def body():
        from aiohttp.test_utils import TestClient, TestServer

        from ai_rtc_agent_amd.agent import create_app
        from ai_rtc_agent_amd.parallel.dispatch import PipelinePool

        stylize = lambda t: (t.int() + 10).clamp(0, 255).to(torch.uint8)
        app = create_app(pool=PipelinePool.single(stylize), use_turn=False)
        http = TestClient(TestServer(app))
        await http.start_server()
        loop = asyncio.get_event_loop()

        # publisher socket
        pub_t, pub_p = await loop.create_datagram_endpoint(
            _ClientProto, local_addr=("127.0.0.1", 0))
        pub_port = pub_t.get_extra_info("sockname")[1]

        r = await http.post("/whip", data=_offer_sdp(pub_port),
                            headers={"Content-Type": "application/sdp"})
        assert r.status == 201
        ans = SessionDescription.parse(await r.text())
        srv_port = ans.media[0].port

        # ICE-lite handshake: binding request -> agent learns our address
        req = stun.make_binding_request("u:p", b"k")
        pub_t.sendto(req, ("127.0.0.1", srv_port))
        await asyncio.sleep(0.1)

        # stream frames to the agent
        from ai_rtc_agent_amd.media.codec import select_codec
        codec = select_codec()
        pkz = RtpPacketizer(ssrc=99)
        g = torch.Generator().manual_seed(0)
        frames = [torch.randint(0, 200, (16, 16, 3), generator=g, dtype=torch.uint8)
                  for _ in range(6)]
        for i, f in enumerate(frames):
            _send_frame(codec, pkz, pub_t, ("127.0.0.1", srv_port), f, i * 3000)
            await asyncio.sleep(0.02)
        await asyncio.sleep(0.2)

        # subscriber: WHEP with our own socket
        sub_t, sub_p = await loop.create_datagram_endpoint(
            _ClientProto, local_addr=("127.0.0.1", 0))
        sub_port = sub_t.get_extra_info("sockname")[1]
        r2 = await http.post("/whep", data=_offer_sdp(sub_port),
                             headers={"Content-Type": "application/sdp"})
        assert r2.status == 201, await r2.text()

        # keep publishing so the subscriber's sender loop has frames to pull
        got = None
        for i in range(6, 40):
            _send_frame(codec, pkz, pub_t, ("127.0.0.1", srv_port),
                        frames[i % len(frames)], i * 3000)
            try:
                got = await asyncio.wait_for(sub_p.frames.get(), timeout=0.25)
                break
            except asyncio.TimeoutError:
                continue
        assert got is not None, "no stylised frame reached the WHEP subscriber"
        # stylize = +10: every received frame must show the pipeline's mark
        src_mean = torch.stack(frames).float().mean()
        assert abs(got.float().mean() - (src_mean + 10)) < 8.0

        pub_t.close()
        sub_t.close()
        await http.close()

    run(body(), timeout=50)


@pytest.mark.timeout(60)
def test_multi_viewer_whep_fanout(monkeypatch):
    """Two WHEP subscribers share one pipeline pull via the relay."""
    monkeypatch.setenv("WARMUP_FRAMES", "0")
    monkeypatch.setenv("DROP_FRAMES", "0")

    async def body():
        from aiohttp.test_utils import TestClient, TestServer

        from ai_rtc_agent_amd.agent import create_app
        from ai_rtc_agent_amd.parallel.dispatch import PipelinePool

        calls = {"n": 0}

        def stylize(t):
            calls["n"] += 1
            return (t.int() + 10).clamp(0, 255).to(torch.uint8)

        app = create_app(pool=PipelinePool.single(stylize), use_turn=False)
        http = TestClient(TestServer(app))
        await http.start_server()
        loop = asyncio.get_event_loop()

        pub_t, _ = await loop.create_datagram_endpoint(
            _ClientProto, local_addr=("127.0.0.1", 0))
        pub_port = pub_t.get_extra_info("sockname")[1]
        r = await http.post("/whip", data=_offer_sdp(pub_port),
                            headers={"Content-Type": "application/sdp"})
        srv_port = SessionDescription.parse(await r.text()).media[0].port
        pub_t.sendto(stun.make_binding_request("u:p", b"k"), ("127.0.0.1", srv_port))
        await asyncio.sleep(0.05)

        from ai_rtc_agent_amd.media.codec import select_codec
        codec = select_codec()
        pkz = RtpPacketizer(ssrc=7)
        g = torch.Generator().manual_seed(1)
        frames = [torch.randint(0, 200, (16, 16, 3), generator=g, dtype=torch.uint8)
                  for _ in range(4)]
        for i in range(4):
            _send_frame(codec, pkz, pub_t, ("127.0.0.1", srv_port), frames[i], i * 3000)
            await asyncio.sleep(0.02)

        subs = []
        for _ in range(2):
            t, p = await loop.create_datagram_endpoint(
                _ClientProto, local_addr=("127.0.0.1", 0))
            port = t.get_extra_info("sockname")[1]
            r2 = await http.post("/whep", data=_offer_sdp(port),
                                 headers={"Content-Type": "application/sdp"})
            assert r2.status == 201
            subs.append((t, p))

        got = [None, None]
        for i in range(4, 120):
            _send_frame(codec, pkz, pub_t, ("127.0.0.1", srv_port), frames[i % 4], i * 3000)
            for si, (_, p) in enumerate(subs):
                if got[si] is None:
                    try:
                        got[si] = await asyncio.wait_for(p.frames.get(), timeout=0.15)
                    except asyncio.TimeoutError:
                        pass
            if all(g is not None for g in got):
                break
        assert all(g is not None for g in got), "both viewers must receive frames"
        # fan-out means ONE pipeline invocation per source frame, not one per viewer
        assert calls["n"] <= 70

        pub_t.close()
        for t, _ in subs:
            t.close()
        await http.close()

    run(body(), timeout=50)


@pytest.mark.timeout(60)
def test_datachannel_config_over_media_socket(monkeypatch):
    """In-band config updates: AIRC-magic JSON datagrams on the media port
    (parity with the reference's datachannel config, agent.py:154-168)."""
    monkeypatch.setenv("WARMUP_FRAMES", "0")

    async def body():
        import json

        from aiohttp.test_utils import TestClient, TestServer

        from ai_rtc_agent_amd.agent import create_app
        from ai_rtc_agent_amd.media.rtc import CONFIG_MAGIC
        from ai_rtc_agent_amd.parallel.dispatch import PipelinePool
        from tests.test_agent_http import StubPipeline

        stub = StubPipeline()
        app = create_app(pool=PipelinePool.single(stub), use_turn=False)
        http = TestClient(TestServer(app))
        await http.start_server()
        loop = asyncio.get_event_loop()

        pub_t, _ = await loop.create_datagram_endpoint(
            _ClientProto, local_addr=("127.0.0.1", 0))
        pub_port = pub_t.get_extra_info("sockname")[1]
        r = await http.post("/whip", data=_offer_sdp(pub_port),
                            headers={"Content-Type": "application/sdp"})
        srv_port = SessionDescription.parse(await r.text()).media[0].port

        msg = CONFIG_MAGIC + json.dumps(
            {"prompt": "via datachannel", "t_index_list": [2, 4]}
        ).encode()
        pub_t.sendto(msg, ("127.0.0.1", srv_port))
        for _ in range(50):
            await asyncio.sleep(0.02)
            if stub.prompt == "via datachannel":
                break
        assert stub.prompt == "via datachannel"
        assert stub.t_index == [2, 4]
        pub_t.close()
        await http.close()

    run(body(), timeout=50)
