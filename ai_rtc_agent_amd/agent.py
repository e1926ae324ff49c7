"""HTTP signalling server + CLI — API parity with the reference agent.

Route table (reference agent.py:466-472):
    POST/DELETE /whip    publish via WHIP (SDP in/out, 201 + Location)
    POST/DELETE /whep    subscribe via WHEP (401 until a publisher exists)
    POST /offer          browser round-trip mode (JSON sdp exchange)
    POST /config         runtime {prompt, t_index_list} updates
    GET  /               health "OK" (polled by the runpod handler)
    GET  /stats          FPS + per-stage latency (ours; SURVEY.md §5.5 gap)

CLI flags mirror reference agent.py:441-455: --model-id --port --udp-ports
--log-level (+ MI355X extras: --gpus, --family, --resolution).

Session model: connection handling is identical to the reference at the
signalling level; the media transport underneath is our ICE-lite RTP stack
(media/rtc.py — see its DTLS note). Multi-GPU serving assigns each new
session to a pipeline replica (parallel/dispatch.py), replacing the
reference's single shared pipeline (agent.py:423).
"""
from __future__ import annotations

import argparse
import logging
import uuid
from typing import Optional

from aiohttp import web

from .media.ice import get_ice_servers, get_link_headers
from .media.rtc import MediaRelay, PeerConnection, set_udp_port_pool
from .media.tracks import VideoStreamTrack
from .utils.events import StreamEventHandler

logger = logging.getLogger(__name__)


def _state(app: web.Application) -> dict:
    return app["state"]


async def offer(request: web.Request) -> web.Response:
    """Browser round-trip: same PC receives source video and sends the
    stylised stream back (reference agent.py:123-208)."""
    params = await request.json()
    room_id = params.get("room_id")
    stream_id = str(uuid.uuid4())
    st = _state(request.app)

    pc = PeerConnection(ice_servers=st["ice_servers"])
    st["pcs"].add(pc)
    pipeline = st["pool"].assign(stream_id)
    events: StreamEventHandler = st["events"]

    @pc.on("datachannel_message")
    def on_config(msg: dict) -> None:
        _apply_config(pipeline, msg)

    @pc.on("track")
    def on_track(track) -> None:
        logger.info("track received for stream %s", stream_id)
        video = VideoStreamTrack(track, pipeline)
        pc.add_track(video)

    @pc.on("connectionstatechange")
    async def on_state() -> None:
        logger.info("pc state %s (stream %s)", pc.connection_state, stream_id)
        if pc.connection_state == "connected":
            await events.stream_started(stream_id, room_id)
        elif pc.connection_state in ("failed", "closed"):
            st["pcs"].discard(pc)
            st["pool"].release(stream_id)
            if pc.connection_state == "closed":
                await events.stream_ended(stream_id, room_id)

    await pc.set_remote_description(params["offer"]["sdp"])
    sdp = await pc.create_answer(host=request.app["host"])
    return web.json_response({"sdp": sdp, "type": "answer"})


async def whip(request: web.Request) -> web.Response:
    """WHIP publish: ingests the source; output is consumed via /whep
    (reference agent.py:285-395 — note whip does NOT send video back)."""
    offer_sdp = await request.text()
    st = _state(request.app)
    stream_id = str(uuid.uuid4())

    # No TURN on WHIP: OBS does not trickle ICE (reference comment
    # agent.py:299-315), so the answer carries only host candidates.
    pc = PeerConnection()
    st["pcs"].add(pc)
    pipeline = st["pool"].assign(stream_id)
    events: StreamEventHandler = st["events"]

    @pc.on("datachannel_message")
    def on_config(msg: dict) -> None:
        _apply_config(pipeline, msg)

    @pc.on("track")
    def on_track(track) -> None:
        st["source_track"] = VideoStreamTrack(track, pipeline)
        logger.info("whip publisher track stored (stream %s)", stream_id)

    @pc.on("connectionstatechange")
    async def on_state() -> None:
        if pc.connection_state == "connected":
            await events.stream_started(stream_id)
        elif pc.connection_state in ("failed", "closed"):
            st["pcs"].discard(pc)
            st["pool"].release(stream_id)
            if st.get("whip_pc") is pc:
                st["source_track"] = None
                st["whip_pc"] = None
            if pc.connection_state == "closed":
                await events.stream_ended(stream_id)

    st["whip_pc"] = pc
    st.setdefault("whip_sessions", {})[stream_id] = pc
    await pc.set_remote_description(offer_sdp)
    sdp = await pc.create_answer(host=request.app["host"], direction="recvonly")
    # per-session resource URL (the WHIP spec's DELETE target; OBS follows
    # the Location header). Bare DELETE /whip still closes the current
    # publisher for reference-parity clients.
    headers = {"Location": f"/whip/{stream_id}"}
    for link in get_link_headers(st["ice_servers"]):
        headers.setdefault("Link", link)
    return web.Response(status=201, content_type="application/sdp", text=sdp, headers=headers)


async def whip_delete(request: web.Request) -> web.Response:
    st = _state(request.app)
    sid = request.match_info.get("sid")
    if sid is not None:
        pc = st.get("whip_sessions", {}).pop(sid, None)
        if pc is None:
            return web.Response(status=404)
        await pc.close()
        if st.get("whip_pc") is pc:
            st["whip_pc"] = None
            st["source_track"] = None
        st["pool"].release(sid)
        return web.Response(status=200)
    pc = st.get("whip_pc")
    if pc is not None:
        await pc.close()
        st["whip_pc"] = None
        st["source_track"] = None
    return web.Response(status=200)


async def whep(request: web.Request) -> web.Response:
    """WHEP subscribe to the current publisher's stylised stream
    (reference agent.py:211-282; 401 without a publisher, :218-220)."""
    st = _state(request.app)
    if st.get("source_track") is None:
        return web.Response(status=401, text="no active publisher")
    offer_sdp = await request.text()

    pc = PeerConnection()
    st["pcs"].add(pc)
    st.setdefault("whep_pcs", set()).add(pc)
    # relay fan-out so N viewers share one pipeline pull (the reference
    # attaches the track directly and leaves its MediaRelay unused,
    # agent.py:248-252 — with >1 viewer they would steal frames from each
    # other; the relay fixes that)
    pc.add_track(st["relay"].subscribe(st["source_track"]))

    @pc.on("connectionstatechange")
    def on_state() -> None:
        if pc.connection_state in ("failed", "closed"):
            st["pcs"].discard(pc)
            st.get("whep_pcs", set()).discard(pc)

    await pc.set_remote_description(offer_sdp)
    sdp = await pc.create_answer(host=request.app["host"], direction="sendonly")
    return web.Response(
        status=201, content_type="application/sdp", text=sdp,
        headers={"Location": "/whep"},
    )


async def whep_delete(request: web.Request) -> web.Response:
    """Close every WHEP subscriber PC (the WHEP resource URL carries no
    per-session id in this minimal server, so DELETE tears down all)."""
    st = _state(request.app)
    for pc in list(st.get("whep_pcs", ())):
        await pc.close()
        st["pcs"].discard(pc)
        st["whep_pcs"].discard(pc)
    return web.Response(status=200)


def _apply_config(pipeline, params: dict) -> None:
    """Shared by POST /config and the datachannel messages
    (reference agent.py:154-168, 324-337, 398-412)."""
    if "t_index_list" in params and params["t_index_list"] is not None:
        pipeline.update_t_index_list(params["t_index_list"])
    if "prompt" in params and params["prompt"] is not None:
        pipeline.update_prompt(params["prompt"])


async def update_config(request: web.Request) -> web.Response:
    params = await request.json()
    st = _state(request.app)
    for pipeline in st["pool"].active():
        _apply_config(pipeline, params)
    return web.json_response({"status": "ok"})


async def health(request: web.Request) -> web.Response:
    return web.Response(text="OK")


async def stats(request: web.Request) -> web.Response:
    st = _state(request.app)
    return web.json_response(st["pool"].stats())


async def metrics(request: web.Request) -> web.Response:
    """Prometheus exposition of the /stats content (the
    reference has no metrics endpoint at all; /stats is the JSON form,
    this is the scrapeable one)."""
    st = _state(request.app)
    pool = st["pool"].stats()
    lines = [
        "# TYPE airtc_replicas gauge",
        f"airtc_replicas {pool.get('replicas', 0)}",
        "# TYPE airtc_sessions gauge",
        f"airtc_sessions {len(pool.get('sessions', {}))}",
        "# TYPE airtc_peer_connections gauge",
        f"airtc_peer_connections {len(st.get('pcs', ()))}",
        "# TYPE airtc_frames_total counter",
        "# TYPE airtc_fps gauge",
        "# TYPE airtc_stage_ms gauge",
    ]
    for i, rep in enumerate(pool.get("per_replica", [])):
        lines.append(f'airtc_frames_total{{replica="{i}"}} {rep.get("frames", 0)}')
        lines.append(f'airtc_fps{{replica="{i}"}} {rep.get("fps", 0.0)}')
        for stage, v in rep.get("stages_ms", {}).items():
            for q in ("p50", "p90"):
                lines.append(
                    f'airtc_stage_ms{{replica="{i}",stage="{stage}",q="{q}"}} '
                    f'{v.get(q, 0.0)}')
        for si, slot in enumerate(rep.get("per_stream", [])):
            if slot.get("p50_ms") is not None:
                lines.append(
                    f'airtc_stream_latency_ms{{replica="{i}",slot="{si}"}} '
                    f'{slot["p50_ms"]}')
    return web.Response(text="\n".join(lines) + "\n",
                        content_type="text/plain")


async def on_startup(app: web.Application) -> None:
    st = _state(app)
    if app["udp_ports"]:
        set_udp_port_pool(app["udp_ports"])
    from .parallel.dispatch import PipelinePool

    if st.get("pool") is None:
        from .config import EngineConfig

        cfg = EngineConfig(
            model_id=app["model_id"],
            model_family=app.get("family", "sd15"),
            width=app.get("resolution", 512),
            height=app.get("resolution", 512),
        )
        st["pool"] = PipelinePool.create(
            model_id=app["model_id"], n_gpus=app["n_gpus"], cfg=cfg,
            streams_per_replica=app.get("streams_per_gpu", 1),
        )
    st["ice_servers"] = get_ice_servers() if app["use_turn"] else []


async def on_shutdown(app: web.Application) -> None:
    st = _state(app)
    for pc in list(st["pcs"]):
        await pc.close()
    st["pcs"].clear()


@web.middleware
async def cors_middleware(request: web.Request, handler):
    """CORS for browser clients (the reference pulls in aiohttp_middlewares'
    cors_middleware, agent.py:459; ours is self-contained)."""
    if request.method == "OPTIONS":
        resp = web.Response(status=204)
    else:
        resp = await handler(request)
    resp.headers["Access-Control-Allow-Origin"] = "*"
    resp.headers["Access-Control-Allow-Methods"] = "GET, POST, DELETE, OPTIONS"
    resp.headers["Access-Control-Allow-Headers"] = "Content-Type, Authorization"
    return resp


def create_app(
    model_id: str = "lykon/dreamshaper-8",
    udp_ports: Optional[list] = None,
    pool=None,
    host: str = "127.0.0.1",
    n_gpus: int = 1,
    use_turn: bool = True,
    family: str = "sd15",
    resolution: int = 512,
    streams_per_gpu: int = 1,
) -> web.Application:
    app = web.Application(middlewares=[cors_middleware])
    app["model_id"] = model_id
    app["udp_ports"] = udp_ports
    app["host"] = host
    app["n_gpus"] = n_gpus
    app["use_turn"] = use_turn
    app["family"] = family
    app["resolution"] = resolution
    app["streams_per_gpu"] = streams_per_gpu
    app["state"] = {
        "pcs": set(),
        "source_track": None,
        "whip_pc": None,
        "events": StreamEventHandler(),
        "relay": MediaRelay(),
        "pool": pool,
        "ice_servers": [],
    }
    app.router.add_post("/offer", offer)
    app.router.add_post("/whip", whip)
    app.router.add_delete("/whip", whip_delete)
    app.router.add_delete("/whip/{sid}", whip_delete)
    app.router.add_post("/whep", whep)
    app.router.add_delete("/whep", whep_delete)
    app.router.add_post("/config", update_config)
    app.router.add_get("/", health)
    app.router.add_get("/stats", stats)
    app.router.add_get("/metrics", metrics)
    app.on_startup.append(on_startup)
    app.on_shutdown.append(on_shutdown)
    return app


def main() -> None:
    parser = argparse.ArgumentParser(description="MI355X real-time video-diffusion agent")
    parser.add_argument("--model-id", default="lykon/dreamshaper-8")
    parser.add_argument("--port", type=int, default=8888)
    parser.add_argument("--udp-ports", default=None, help="e.g. 40000-40100")
    parser.add_argument("--log-level", default="INFO")
    parser.add_argument("--gpus", type=int, default=1, help="pipeline replicas (one per GPU)")
    parser.add_argument("--host", default="0.0.0.0")
    parser.add_argument("--family", default="sd15", choices=["sd15", "sd21", "sdxl"],
                        help="UNet family served by the pipeline")
    parser.add_argument("--resolution", type=int, default=512)
    parser.add_argument("--streams-per-gpu", type=int, default=1,
                        help="multi-stream batched serving: sessions per "
                             "GPU batched through one engine (fbs=K; "
                             "profiles/batching_ab.md measured +124% "
                             "aggregate at K=8)")
    parser.add_argument("--workers", type=int, default=0,
                        help="process-per-GPU serving: spawn N worker agents "
                             "(one per GPU, own media sockets) behind a "
                             "signalling front-end (SURVEY.md §5.8); 0 = "
                             "single-process mode")
    args = parser.parse_args()

    logging.basicConfig(level=getattr(logging, args.log_level.upper(), logging.INFO))
    ports = None
    if args.udp_ports:
        lo, _, hi = args.udp_ports.partition("-")
        ports = list(range(int(lo), int(hi or lo) + 1))
    if args.workers > 0:
        import asyncio

        import torch

        from .parallel.frontend import WorkerFrontend

        fe = WorkerFrontend(args.workers, model_id=args.model_id,
                            family=args.family, resolution=args.resolution,
                            pin_gpu=torch.cuda.is_available(),
                            udp_ports=ports)
        fe.spawn()
        asyncio.get_event_loop().run_until_complete(fe.wait_ready())
        try:
            web.run_app(fe.create_app(), host=args.host, port=args.port)
        finally:
            fe.shutdown()
        return
    app = create_app(model_id=args.model_id, udp_ports=ports, n_gpus=args.gpus,
                     family=args.family, resolution=args.resolution,
                     streams_per_gpu=args.streams_per_gpu)
    web.run_app(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
