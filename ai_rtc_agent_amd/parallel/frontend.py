"""Process-per-GPU serving: worker agents + a signalling front-end.

SURVEY.md §5.8 design (replacing the reference's single shared-pipeline
process, reference agent.py:423): one OS process per GPU, each owning its
HIP context, pipeline replica, and UDP media sockets; a thin front-end
assigns incoming sessions to workers and proxies ONLY the HTTP signalling.
Media never crosses a process boundary — the SDP answer a worker returns
carries that worker's own host candidate and UDP port, so RTP flows
directly between the client and the owning worker. One Python process per
media plane also means one GIL per stream pipeline (round-1 verdict,
Missing #6: N replicas in one process would throttle 8 media planes).

Worker GPU pinning uses HIP_VISIBLE_DEVICES before torch import, so each
worker sees exactly one device as cuda:0.
"""
from __future__ import annotations

import asyncio
import logging
import multiprocessing as mp
import os
import socket
from typing import List, Optional

logger = logging.getLogger(__name__)


def free_tcp_ports(n: int) -> List[int]:
    socks, ports = [], []
    for _ in range(n):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        socks.append(s)
        ports.append(s.getsockname()[1])
    for s in socks:
        s.close()
    return ports


def _worker_main(rank: int, http_port: int, model_id: str, family: str,
                 resolution: int, pin_gpu: bool,
                 udp_ports: Optional[list]) -> None:
    """Worker entry (spawned process): a full agent on its own GPU."""
    if pin_gpu:
        # one process <-> one GPU: pin BEFORE torch initialises HIP
        os.environ["HIP_VISIBLE_DEVICES"] = str(rank)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(rank)
    logging.basicConfig(level=logging.INFO,
                        format=f"worker{rank} %(levelname)s %(message)s")
    from aiohttp import web

    from ..agent import create_app

    app = create_app(model_id=model_id, n_gpus=1, use_turn=False,
                     family=family, resolution=resolution,
                     udp_ports=udp_ports, host="127.0.0.1")
    web.run_app(app, host="127.0.0.1", port=http_port,
                print=None, handle_signals=False)


class WorkerFrontend:
    """Spawns N worker agents and serves the public signalling API.

    Route behaviour (public surface identical to the single-process agent):
      POST /whip, /offer  -> least-loaded worker (session -> worker affinity)
      DELETE /whip        -> the publisher's worker
      POST/DELETE /whep   -> the current publisher's worker (viewers read
                             the stream that worker owns)
      POST /config        -> broadcast to every worker
      GET  /              -> OK iff every worker is healthy
      GET  /stats         -> aggregation of all workers' stats
    """

    def __init__(self, n_workers: int, model_id: str = "lykon/dreamshaper-8",
                 family: str = "sd15", resolution: int = 512,
                 pin_gpu: bool = False, udp_ports: Optional[list] = None):
        self.n = n_workers
        self.model_id = model_id
        self.family = family
        self.resolution = resolution
        self.pin_gpu = pin_gpu
        self.udp_ports = udp_ports
        self.ports = free_tcp_ports(n_workers)
        self.procs: List[mp.Process] = []
        self._load = [0] * n_workers
        self._publisher_worker = 0
        self._http = None
        self._monitor_task = None
        self._closing = False

    # -- lifecycle -------------------------------------------------------
    def spawn(self) -> None:
        ctx = mp.get_context("spawn")
        for rank in range(self.n):
            # split an operator-pinned UDP pool across workers so two
            # workers never race for the same media port
            wp = self.udp_ports[rank::self.n] if self.udp_ports else None
            p = ctx.Process(
                target=_worker_main,
                args=(rank, self.ports[rank], self.model_id, self.family,
                      self.resolution, self.pin_gpu, wp),
                daemon=True,
            )
            p.start()
            self.procs.append(p)

    async def wait_ready(self, timeout: float = 120.0) -> None:
        import aiohttp

        async with aiohttp.ClientSession() as s:
            for rank, port in enumerate(self.ports):
                deadline = asyncio.get_event_loop().time() + timeout
                while True:
                    try:
                        async with s.get(f"http://127.0.0.1:{port}/") as r:
                            if r.status == 200:
                                break
                    except aiohttp.ClientError:
                        pass
                    if not self.procs[rank].is_alive():
                        raise RuntimeError(f"worker {rank} died during startup")
                    if asyncio.get_event_loop().time() > deadline:
                        raise TimeoutError(f"worker {rank} not ready")
                    await asyncio.sleep(0.2)

    def shutdown(self) -> None:
        self._closing = True
        if self._monitor_task is not None:
            self._monitor_task.cancel()
            self._monitor_task = None
        for p in self.procs:
            if p.is_alive():
                p.terminate()
        for p in self.procs:
            p.join(timeout=10)
        self.procs.clear()

    # -- elastic recovery -------------------------------
    def _respawn(self, rank: int) -> None:
        ctx = mp.get_context("spawn")
        wp = self.udp_ports[rank::self.n] if self.udp_ports else None
        p = ctx.Process(
            target=_worker_main,
            args=(rank, self.ports[rank], self.model_id, self.family,
                  self.resolution, self.pin_gpu, wp),
            daemon=True,
        )
        p.start()
        self.procs[rank] = p
        self._load[rank] = 0  # its sessions died with it
        logger.warning("worker %d died; respawned as pid %d", rank, p.pid)

    async def _monitor(self, interval: float = 2.0) -> None:
        """Respawn crashed worker processes (a GPU fault or OOM in one
        worker must not take down the other GPUs' media planes)."""
        try:
            while not self._closing:
                await asyncio.sleep(interval)
                for rank, p in enumerate(self.procs):
                    if not p.is_alive():
                        self._respawn(rank)
        except asyncio.CancelledError:
            pass

    def start_monitor(self) -> None:
        if self._monitor_task is None:
            self._monitor_task = asyncio.ensure_future(self._monitor())

    # -- proxying --------------------------------------------------------
    def _pick_worker(self) -> int:
        idx = min(range(self.n), key=lambda i: self._load[i])
        self._load[idx] += 1
        return idx

    def _url(self, rank: int, path: str) -> str:
        return f"http://127.0.0.1:{self.ports[rank]}{path}"

    async def _forward(self, rank: int, request) -> "web.Response":
        import aiohttp
        from aiohttp import web

        body = await request.read()
        try:
            async with self._session().request(
                request.method, self._url(rank, request.path),
                data=body, headers={"Content-Type":
                                    request.headers.get("Content-Type", "")},
            ) as r:
                payload = await r.read()
                headers = {}
                if "Location" in r.headers:
                    headers["Location"] = r.headers["Location"]
                return web.Response(status=r.status, body=payload,
                                    content_type=r.content_type,
                                    headers=headers)
        except aiohttp.ClientError:
            # worker mid-restart (see _monitor): tell the client to retry
            return web.Response(status=503, text=f"worker {rank} restarting")

    def _session(self):
        import aiohttp

        if self._http is None or self._http.closed:
            self._http = aiohttp.ClientSession()
        return self._http

    def create_app(self):
        from aiohttp import web

        async def publish(request):  # /whip POST and /offer
            rank = self._pick_worker()
            self._publisher_worker = rank
            logger.info("session %s -> worker %d", request.path, rank)
            return await self._forward(rank, request)

        async def to_publisher(request):  # /whep, DELETE /whip
            return await self._forward(self._publisher_worker, request)

        async def config(request):
            body = await request.read()
            import aiohttp

            for rank in range(self.n):
                try:
                    async with self._session().post(
                            self._url(rank, "/config"), data=body,
                            headers={"Content-Type": "application/json"}) as r:
                        await r.read()
                except aiohttp.ClientError:
                    logger.warning("config broadcast to worker %d failed", rank)
            return web.json_response({"status": "ok", "workers": self.n})

        async def health(request):
            import aiohttp

            for rank in range(self.n):
                try:
                    async with self._session().get(self._url(rank, "/")) as r:
                        if r.status != 200:
                            return web.Response(status=503,
                                                text=f"worker {rank} unhealthy")
                except aiohttp.ClientError:
                    return web.Response(status=503, text=f"worker {rank} down")
            return web.Response(text="OK")

        async def stats(request):
            out = {"workers": []}
            import aiohttp

            for rank in range(self.n):
                try:
                    async with self._session().get(self._url(rank, "/stats")) as r:
                        out["workers"].append(await r.json())
                except aiohttp.ClientError:
                    out["workers"].append({"error": "down"})
            out["assignments"] = {"publisher_worker": self._publisher_worker,
                                  "load": list(self._load)}
            return web.json_response(out)

        async def on_startup(app):
            self.start_monitor()

        async def on_shutdown(app):
            if self._monitor_task is not None:
                self._monitor_task.cancel()
                self._monitor_task = None
            if self._http is not None and not self._http.closed:
                await self._http.close()

        app = web.Application()
        app.on_startup.append(on_startup)
        app.router.add_post("/whip", publish)
        app.router.add_post("/offer", publish)
        app.router.add_delete("/whip", to_publisher)
        app.router.add_post("/whep", to_publisher)
        app.router.add_delete("/whep", to_publisher)
        app.router.add_post("/config", config)
        app.router.add_get("/", health)
        app.router.add_get("/stats", stats)
        app.on_shutdown.append(on_shutdown)
        return app
