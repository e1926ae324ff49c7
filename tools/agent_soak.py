#!/usr/bin/env python3
"""Full-agent soak: a DTLS-SRTP peer publishes synthetic video into the
real agent (SD-Turbo pipeline on cuda:0) for N seconds over localhost UDP;
reports throughput, wire sizes and device-memory growth.

    python tools/agent_soak.py [--seconds 60] [--fps 30]

This is the sustained version of tests/test_dtls.py's encrypted loopback:
handshake -> SRTP-protected H.264 in -> pipeline -> SRTP-protected H.264
back, with the PLI/keyframe and rate-control loops live.
"""
import argparse
import asyncio
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


async def main_async(seconds: int, fps: int) -> int:
    from aiohttp.test_utils import TestClient, TestServer

    from ai_rtc_agent_amd.agent import create_app
    from ai_rtc_agent_amd.config import sd_turbo_config
    from ai_rtc_agent_amd.media import stun
    from ai_rtc_agent_amd.media.codec import select_codec
    from ai_rtc_agent_amd.media.h264 import packetize_h264, split_annexb
    from ai_rtc_agent_amd.media.rtp import RtpPacket
    from ai_rtc_agent_amd.media.sdp import SessionDescription
    from ai_rtc_agent_amd.ops import _load_ext
    from ai_rtc_agent_amd.parallel.dispatch import PipelinePool

    os.environ.setdefault("WARMUP_FRAMES", "2")
    ext = _load_ext.load()
    E = ext.DtlsEndpoint

    cfg = sd_turbo_config(device="cuda" if torch.cuda.is_available() else "cpu")
    if not torch.cuda.is_available():
        cfg.model_family = "tiny"
        cfg.width = cfg.height = 64
        cfg.use_hip_graph = False
    pool = PipelinePool.create("stabilityai/sd-turbo", n_gpus=1, cfg=cfg)
    app = create_app(pool=pool, use_turn=False)
    http = TestClient(TestServer(app))
    await http.start_server()
    loop = asyncio.get_event_loop()

    cli = E(server=False)
    stats = {"rx_frames": 0, "rx_bytes": 0, "tx_bytes": 0}

    class Proto(asyncio.DatagramProtocol):
        def connection_made(self, t):
            self.t = t

        def datagram_received(self, data, addr):
            if stun.is_stun(data):
                return
            if 20 <= data[0] <= 63:
                for d in cli.feed(data):
                    self.t.sendto(d, addr)
                return
            if not cli.established():
                return
            if 128 <= data[0] <= 191:
                plain = cli.unprotect_rtp(data)
                if plain is None:
                    return
                stats["rx_bytes"] += len(plain)
                pkt = RtpPacket.parse(plain)
                if pkt.marker:
                    stats["rx_frames"] += 1

    t, proto = await loop.create_datagram_endpoint(
        Proto, local_addr=("127.0.0.1", 0))
    port = t.get_extra_info("sockname")[1]

    offer = "\r\n".join([
        "v=0", "o=- 1 2 IN IP4 127.0.0.1", "s=-", "t=0 0",
        "a=fingerprint:sha-256 " + E.local_fingerprint(),
        f"m=video {port} UDP/TLS/RTP/SAVPF 97",
        "a=ice-ufrag:u", "a=ice-pwd:p0123456789abcdef", "a=mid:0",
        "a=sendrecv", "a=rtpmap:97 H264/90000", "a=setup:actpass",
        f"a=candidate:1 1 udp 2130706431 127.0.0.1 {port} typ host",
    ]) + "\r\n"
    r = await http.post("/offer", json={"offer": {"sdp": offer, "type": "offer"}})
    assert r.status == 200
    srv_port = SessionDescription.parse((await r.json())["sdp"]).media[0].port
    srv = ("127.0.0.1", srv_port)

    t.sendto(stun.make_binding_request("a:b", b"pw"), srv)
    await asyncio.sleep(0.05)
    for d in cli.start():
        t.sendto(d, srv)
    for _ in range(200):
        if cli.established():
            break
        await asyncio.sleep(0.05)
    assert cli.established(), "DTLS handshake"
    print("DTLS established; streaming...")

    enc = select_codec()
    side = cfg.width
    g = torch.Generator().manual_seed(0)
    base = torch.randint(0, 220, (side, side, 3), generator=g, dtype=torch.uint8)
    seq = 0
    t0 = time.monotonic()
    mem0 = None  # sampled AFTER warm-up/graph capture (first ~5 s)
    n_sent = 0
    next_report = t0 + 10
    while time.monotonic() - t0 < seconds:
        # gentle motion so P frames carry real refresh work
        frame = base.roll(shifts=n_sent % side, dims=1)
        payloads = packetize_h264(split_annexb(enc.encode(frame)))
        for i, pl in enumerate(payloads):
            pkt = RtpPacket(payload_type=97, sequence_number=seq,
                            timestamp=n_sent * 3000, ssrc=55,
                            marker=1 if i == len(payloads) - 1 else 0,
                            payload=pl)
            seq = (seq + 1) & 0xFFFF
            wire = cli.protect_rtp(pkt.serialize())
            stats["tx_bytes"] += len(wire)
            t.sendto(wire, srv)
        n_sent += 1
        now = time.monotonic()
        if mem0 is None and now - t0 > 5.0:
            # steady-state baseline: plan build, graph capture and weight
            # transforms have allocated by now
            mem0 = torch.cuda.memory_allocated() / 1e6 if torch.cuda.is_available() else 0
        if now >= next_report:
            mem = torch.cuda.memory_allocated() / 1e6 if torch.cuda.is_available() else 0
            print(f"t={now - t0:5.1f}s sent={n_sent} rx_frames={stats['rx_frames']} "
                  f"tx={stats['tx_bytes']/1e6:.1f}MB rx={stats['rx_bytes']/1e6:.1f}MB "
                  f"mem={mem:.1f}MB")
            next_report += 10
        await asyncio.sleep(max(0.0, (n_sent / fps) - (now - t0)))

    mem1 = torch.cuda.memory_allocated() / 1e6 if torch.cuda.is_available() else 0
    if mem0 is None:
        mem0 = mem1
    dur = time.monotonic() - t0
    print(f"TOTAL: sent {n_sent} ({n_sent/dur:.1f} fps), received "
          f"{stats['rx_frames']} stylised frames ({stats['rx_frames']/dur:.1f} fps); "
          f"mem {mem0:.1f} -> {mem1:.1f} MB (delta {mem1-mem0:+.2f})")
    ok = stats["rx_frames"] > seconds * fps * 0.5 and abs(mem1 - mem0) < 64
    print("AGENT SOAK OK" if ok else "AGENT SOAK FAIL")
    t.close()
    await http.close()
    return 0 if ok else 1


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--seconds", type=int, default=60)
    p.add_argument("--fps", type=int, default=30)
    args = p.parse_args()
    loop = asyncio.new_event_loop()
    try:
        sys.exit(loop.run_until_complete(main_async(args.seconds, args.fps)))
    finally:
        loop.close()


if __name__ == "__main__":
    main()
