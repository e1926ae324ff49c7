"""Process-per-GPU serving integration test (CPU: N worker processes).

SURVEY.md §5.8 / round-1 verdict Missing #6: worker agents each own their
media sockets and pipeline; the front-end only dispatches signalling.
Runs fully on CPU with the tiny model family (one worker == one process,
GPU pinning is a no-op here).
"""
import asyncio

import pytest
import torch

from ai_rtc_agent_amd.media import stun
from ai_rtc_agent_amd.media.sdp import SessionDescription


def _offer_sdp(port: int) -> str:
    return "\r\n".join([
        "v=0", "o=- 1 2 IN IP4 127.0.0.1", "s=-", "t=0 0",
        f"m=video {port} UDP/TLS/RTP/SAVPF 97",
        "a=ice-ufrag:testu", "a=ice-pwd:testpw0123456789", "a=mid:0",
        "a=sendrecv", "a=rtpmap:97 H264/90000",
        f"a=candidate:1 1 udp 2130706431 127.0.0.1 {port} typ host",
    ]) + "\r\n"


@pytest.mark.timeout(300)
def test_two_worker_processes_serve_independent_sessions(monkeypatch):
    monkeypatch.setenv("WARMUP_FRAMES", "0")
    monkeypatch.setenv("DROP_FRAMES", "0")

    from ai_rtc_agent_amd.parallel.frontend import WorkerFrontend

    async def body():
        import aiohttp
        from aiohttp.test_utils import TestClient, TestServer

        fe = WorkerFrontend(2, family="tiny", resolution=64, pin_gpu=False)
        fe.spawn()
        try:
            await fe.wait_ready(timeout=240)
            http = TestClient(TestServer(fe.create_app()))
            await http.start_server()

            r = await http.get("/")
            assert r.status == 200 and await r.text() == "OK"

            # two publishers -> two different worker processes
            loop = asyncio.get_event_loop()
            answers = []
            from tests.test_tracks_loopback import _ClientProto, _send_frame

            clients = []
            for _ in range(2):
                t, p = await loop.create_datagram_endpoint(
                    _ClientProto, local_addr=("127.0.0.1", 0))
                port = t.get_extra_info("sockname")[1]
                r = await http.post("/whip", data=_offer_sdp(port),
                                    headers={"Content-Type": "application/sdp"})
                assert r.status == 201, await r.text()
                ans = SessionDescription.parse(await r.text())
                answers.append(ans)
                clients.append((t, p, port))

            srv_ports = [a.media[0].port for a in answers]
            # sessions landed on different processes -> different UDP ports
            assert srv_ports[0] != srv_ports[1]
            r = await http.get("/stats")
            st = await r.json()
            assert len(st["workers"]) == 2
            assert st["assignments"]["load"] == [1, 1]

            # loopback media through EACH worker concurrently:
            # publish to both, subscribe (whep routes to the most recent
            # publisher's worker) and verify processed frames flow
            from ai_rtc_agent_amd.media.codec import select_codec
            from ai_rtc_agent_amd.media.rtp import RtpPacketizer

            codecs = [select_codec() for _ in range(2)]
            pkzs = [RtpPacketizer(ssrc=11), RtpPacketizer(ssrc=22)]
            g = torch.Generator().manual_seed(0)
            frames = [torch.randint(0, 200, (64, 64, 3), generator=g,
                                    dtype=torch.uint8) for _ in range(4)]
            for ci, (t, p, port) in enumerate(clients):
                t.sendto(stun.make_binding_request("u:p", b"k"),
                         ("127.0.0.1", srv_ports[ci]))
            await asyncio.sleep(0.2)
            # a few frames first so each worker's publisher track fires
            # (whep 401s until the worker has a source track)
            for i in range(3):
                for ci, (t, p, port) in enumerate(clients):
                    _send_frame(codecs[ci], pkzs[ci], t,
                                ("127.0.0.1", srv_ports[ci]),
                                frames[i % 4], (i + 1) * 3000)
                await asyncio.sleep(0.1)

            # subscriber against the publisher_worker (client 1's worker)
            sub_t, sub_p = await loop.create_datagram_endpoint(
                _ClientProto, local_addr=("127.0.0.1", 0))
            sub_port = sub_t.get_extra_info("sockname")[1]
            r2 = await http.post("/whep", data=_offer_sdp(sub_port),
                                 headers={"Content-Type": "application/sdp"})
            assert r2.status == 201, await r2.text()

            got = None
            for i in range(60):
                for ci, (t, p, port) in enumerate(clients):
                    _send_frame(codecs[ci], pkzs[ci], t,
                                ("127.0.0.1", srv_ports[ci]),
                                frames[i % 4], (i + 1) * 3000)
                try:
                    got = await asyncio.wait_for(sub_p.frames.get(), timeout=0.3)
                    break
                except asyncio.TimeoutError:
                    continue
            assert got is not None, "no frame through the worker process"

            # config broadcast reaches every worker
            r3 = await http.post("/config", json={"prompt": "hello"})
            assert (await r3.json())["workers"] == 2

            for t, _, _ in clients:
                t.close()
            sub_t.close()
            await http.close()
        finally:
            fe.shutdown()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(asyncio.wait_for(body(), 280))
    finally:
        loop.close()


@pytest.mark.timeout(300)
def test_worker_crash_respawns(monkeypatch):
    """Elastic recovery: a killed worker process comes
    back and the front-end keeps serving."""
    from ai_rtc_agent_amd.parallel.frontend import WorkerFrontend

    async def body():
        from aiohttp.test_utils import TestClient, TestServer

        fe = WorkerFrontend(2, family="tiny", resolution=64, pin_gpu=False)
        fe.spawn()
        try:
            await fe.wait_ready(timeout=240)
            http = TestClient(TestServer(fe.create_app()))
            await http.start_server()  # starts the monitor
            victim = fe.procs[1]
            victim.terminate()
            victim.join(timeout=10)
            assert not victim.is_alive()
            # the monitor notices within ~2s and respawns
            for _ in range(120):
                await asyncio.sleep(0.5)
                if fe.procs[1] is not victim and fe.procs[1].is_alive():
                    break
            assert fe.procs[1] is not victim and fe.procs[1].is_alive()
            # the respawned worker serves again
            await fe.wait_ready(timeout=240)
            r = await http.get("/")
            assert r.status == 200
            await http.close()
        finally:
            fe.shutdown()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(asyncio.wait_for(body(), 280))
    finally:
        loop.close()
