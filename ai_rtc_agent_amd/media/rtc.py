"""Peer connection + ICE-lite UDP media transport — from scratch.

Stands in for the aiortc RTCPeerConnection the reference uses
(agent.py:136,299). Scope: SDP offer/answer signalling, ICE-lite (STUN
binding answerer, peer-reflexive address learning), DTLS-SRTP (native
OpenSSL endpoint, media/dtls.py — armed whenever the remote SDP carries a
fingerprint, which is what every browser/OBS offer does), a TURN relay
client (media/turn.py — allocated when a turn: ice server with credentials
is configured, advertised as an extra relay candidate), RTP media with the
codec HAL (standard H.264 by default), and a lightweight JSON config
channel over the same socket (magic-prefixed datagrams) mirroring the
reference's datachannel config updates (agent.py:154-168). Offers WITHOUT
a fingerprint (this repo's own test peers) run plain RTP.

UDP port pinning: the reference monkey-patches asyncio's datagram endpoint
factory to force media onto operator ports (agent.py:32-69, for firewalls /
OBS). We own the transport, so the pool is first-class: pass allowed ports
to PeerConnection and the bind loop walks that pool.
"""
from __future__ import annotations

import asyncio
import json
import logging
import random
import socket
import time
from typing import Callable, Dict, List, Optional, Tuple

from .codec import select_codec
from .h264 import H264Depacketizer, join_annexb, packetize_h264, split_annexb
from .rtcp import is_rtcp, make_pli, make_rr, parse_pli, parse_rr
from .rtp import RtpDefragmenter, RtpPacket, RtpPacketizer
from .sdp import SessionDescription, build_answer
from .stun import StunMessage, BINDING_REQUEST, is_stun, make_binding_response
from .tracks import MediaStreamTrack, QueueTrack, VideoFrame

logger = logging.getLogger(__name__)

CONFIG_MAGIC = b"AIRC"  # JSON config datagrams (datachannel-lite)

MEDIA_SOCKET_BUF = 8 << 20  # 8 MiB: a 512p frame fragments to ~700 KB of RTP


def tune_socket_buffers(transport) -> None:
    """Large SO_RCVBUF/SO_SNDBUF on a datagram transport: one video frame
    bursts hundreds of datagrams, far beyond the 212 KB Linux default (the
    kernel silently clamps plain setsockopt to rmem_max, so try the FORCE
    variants first — we run as root in the container)."""
    sock = transport.get_extra_info("socket")
    if sock is None:
        return
    for opt_force, opt in ((33, socket.SO_RCVBUF), (32, socket.SO_SNDBUF)):
        # SO_RCVBUFFORCE=33, SO_SNDBUFFORCE=32 (linux)
        try:
            sock.setsockopt(socket.SOL_SOCKET, opt_force, MEDIA_SOCKET_BUF)
        except (OSError, PermissionError):
            try:
                sock.setsockopt(socket.SOL_SOCKET, opt, MEDIA_SOCKET_BUF)
            except OSError:
                pass


_port_pool: Optional[List[int]] = None


def set_udp_port_pool(ports: Optional[List[int]]) -> None:
    """Operator port pinning (parity: reference --udp-ports, agent.py:460)."""
    global _port_pool
    _port_pool = ports


class _Proto(asyncio.DatagramProtocol):
    def __init__(self, pc: "PeerConnection"):
        self.pc = pc

    def datagram_received(self, data: bytes, addr) -> None:
        self.pc._on_datagram(data, addr)


class PeerConnection:
    """One media session (the reference's RTCPeerConnection role)."""

    def __init__(self, ice_servers: Optional[list] = None):
        self.ice_servers = ice_servers or []
        self.connection_state = "new"
        self.remote_description: Optional[SessionDescription] = None
        self.local_description: Optional[SessionDescription] = None
        self._handlers: Dict[str, List[Callable]] = {}
        self._transport = None
        self._remote_addr: Optional[Tuple[str, int]] = None
        self._send_track: Optional[MediaStreamTrack] = None
        self._sender_task: Optional[asyncio.Task] = None
        self._recv_track: Optional[QueueTrack] = None
        self._track_fired = False
        self._encoder = select_codec(role="encode")
        self._decoder = select_codec(role="decode")
        self._packetizer = RtpPacketizer(ssrc=random.randint(1, 2**31))
        self._defrag = RtpDefragmenter()
        self._h264_depack = H264Depacketizer()
        self._rx_au: Dict[int, dict] = {}
        self._force_keyframe = False
        self._decode_misses = 0
        self._last_pli = 0.0
        # loss accounting (receiver) + adaptive bitrate (sender)
        self._rx_lock_ssrc: Optional[int] = None
        self._rx_count = 0
        self._rx_base_seq: Optional[int] = None
        self._rx_high_seq = 0
        self._rr_last_sent = (0, 0)  # (expected, received) at last RR
        self.port: Optional[int] = None
        self._ice_pwd = ""
        # DTLS-SRTP (created when the remote SDP carries a fingerprint)
        self._dtls = None
        self._dtls_expected_fp: Optional[str] = None
        self._dtls_verified = False
        self._dtls_task: Optional[asyncio.Task] = None
        # TURN relay (allocated when a turn: ice server is configured)
        self._relay = None
        self._relay_peers: set = set()

    # -- event API (aiortc-style) --------------------------------------
    def on(self, event: str, handler: Optional[Callable] = None):
        def register(h):
            self._handlers.setdefault(event, []).append(h)
            return h

        return register(handler) if handler else register

    def _emit(self, event: str, *args) -> None:
        for h in self._handlers.get(event, []):
            res = h(*args)
            if asyncio.iscoroutine(res):
                asyncio.ensure_future(res)

    def _set_state(self, state: str) -> None:
        if state != self.connection_state:
            self.connection_state = state
            self._emit("connectionstatechange")

    # -- signalling ------------------------------------------------------
    async def set_remote_description(self, sdp: str) -> None:
        self.remote_description = SessionDescription.parse(sdp)
        # learn the peer's host candidate for outbound-first flows
        for m in self.remote_description.media:
            for cand in m.candidates:
                parts = cand.split()
                if len(parts) >= 6 and parts[2].lower() == "udp":
                    self._remote_addr = (parts[4], int(parts[5]))
                    break
        # a remote fingerprint means the peer (browser/OBS) requires
        # DTLS-SRTP; we always answer setup:passive, so we are the server
        fp = self.remote_description.fingerprint
        for m in self.remote_description.media:
            fp = m.fingerprint or fp
        if fp is not None:
            from . import dtls as dtls_mod

            if dtls_mod.dtls_available():
                self._dtls = dtls_mod.create_endpoint(server=True)
                self._dtls_expected_fp = fp
            else:
                logger.error(
                    "peer requires DTLS-SRTP but the native endpoint is "
                    "unavailable; answering plain RTP (will not connect)"
                )

    async def create_answer(self, host: str = "127.0.0.1", direction: str = "sendrecv") -> str:
        await self._bind(host)
        assert self.remote_description is not None, "set_remote_description first"
        from . import dtls as dtls_mod

        await self._maybe_allocate_relay()
        ans = build_answer(
            self.remote_description,
            host,
            self.port,
            codec_name="H264",
            ssrc=self._packetizer.ssrc,
            direction=direction,
            fingerprint=dtls_mod.local_fingerprint(),
        )
        if self._relay is not None and self._relay.client.relayed_addr:
            rip, rport = self._relay.client.relayed_addr
            for m in ans.media:
                m.candidates.append(
                    f"candidate:2 1 udp 16777215 {rip} {rport} typ relay "
                    f"raddr 0.0.0.0 rport 0")
        self.local_description = ans
        self._ice_pwd = ans.media[0].ice_pwd if ans.media else ""
        if self._dtls is not None and self._dtls_task is None:
            self._dtls_task = asyncio.ensure_future(self._dtls_timer())
        self._set_state("connecting")
        return ans.serialize()

    async def _maybe_allocate_relay(self) -> None:
        """Allocate a TURN relay when a turn: ice server with credentials
        is configured (reference gets this from aioice + Twilio,
        agent.py:80-109). Host candidates remain primary; the relayed
        address is advertised as an extra candidate and any peer that
        reaches us through it is answered through it."""
        if self._relay is not None or not self.ice_servers:
            return
        from .turn import TurnClient, TurnTransport, parse_turn_url

        for srv in self.ice_servers:
            urls = getattr(srv, "urls", None) if not isinstance(srv, dict) \
                else srv.get("urls")
            username = getattr(srv, "username", None) if not isinstance(srv, dict) \
                else srv.get("username")
            credential = getattr(srv, "credential", None) if not isinstance(srv, dict) \
                else srv.get("credential")
            if not urls or not username:
                continue
            for url in urls if isinstance(urls, list) else [urls]:
                addr = parse_turn_url(url)
                if addr is None:
                    continue
                tt = TurnTransport(TurnClient(username, credential or ""),
                                   addr, self._on_relay_data)
                try:
                    if await tt.allocate(timeout=3.0):
                        self._relay = tt
                        logger.info("TURN relay allocated at %s:%d",
                                    *tt.client.relayed_addr)
                        return
                except OSError:
                    pass
                tt.close()

    def _on_relay_data(self, data: bytes, peer) -> None:
        self._relay_peers.add(peer)
        self._on_datagram(data, peer)

    def _sendto(self, data: bytes, addr) -> None:
        """Route to the peer: via the TURN relay for peers that reached us
        through it, else directly over our UDP socket."""
        if self._relay is not None and addr in self._relay_peers:
            self._relay.sendto(data, addr)
        elif self._transport is not None:
            self._transport.sendto(data, addr)

    async def _dtls_timer(self) -> None:
        """Drive DTLS retransmissions until the handshake completes (the
        transport is datagram-lossy; OpenSSL's timer needs a pump)."""
        t0 = time.monotonic()
        try:
            while (self._dtls is not None and not self._dtls.established()
                   and self.connection_state not in ("closed", "failed")
                   and time.monotonic() - t0 < 30.0):
                await asyncio.sleep(0.4)
                if self._transport is not None and self._remote_addr:
                    for out in self._dtls.handle_timeout():
                        self._sendto(out, self._remote_addr)
        except asyncio.CancelledError:
            pass

    async def _bind(self, host: str) -> None:
        if self._transport is not None:
            return
        loop = asyncio.get_event_loop()
        last_err: Optional[Exception] = None
        ports = _port_pool or [0]
        for p in ports:
            try:
                self._transport, _ = await loop.create_datagram_endpoint(
                    lambda: _Proto(self), local_addr=("0.0.0.0", p)
                )
                tune_socket_buffers(self._transport)
                self.port = self._transport.get_extra_info("sockname")[1]
                return
            except OSError as e:  # port in use: walk the pool
                last_err = e
        raise OSError(f"no usable UDP port in pool {ports}: {last_err}")

    # -- media ----------------------------------------------------------
    def add_track(self, track: MediaStreamTrack) -> None:
        """Outbound processed video (the reference's pc.addTrack,
        agent.py:178)."""
        self._send_track = track
        if self._sender_task is None:
            self._sender_task = asyncio.ensure_future(self._sender_loop())

    async def _sender_loop(self) -> None:
        counter = 0
        try:
            while self.connection_state not in ("closed", "failed"):
                frame = await self._send_track.recv()
                if self._remote_addr is None or self._transport is None:
                    continue
                want_key = counter == 0 or self._force_keyframe
                self._force_keyframe = False
                data = self._encoder.encode(frame.tensor, keyframe=want_key)
                ts = frame.pts if frame.pts else counter * 3000
                if getattr(self._encoder, "rtp_mode", "raw") == "rfc6184":
                    # H.264: standard single-NAL / FU-A payloads per NAL
                    pkts = []
                    nals = split_annexb(data)
                    payloads = packetize_h264(nals)
                    for pi, pl in enumerate(payloads):
                        pkts.append(RtpPacket(
                            payload_type=self._packetizer.payload_type,
                            sequence_number=self._packetizer._seq,
                            timestamp=ts,
                            ssrc=self._packetizer.ssrc,
                            marker=1 if pi == len(payloads) - 1 else 0,
                            payload=pl,
                        ))
                        self._packetizer._seq = (self._packetizer._seq + 1) & 0xFFFF
                else:
                    pkts = self._packetizer.packetize(data, ts)
                # paced send: large frames fragment into hundreds of
                # datagrams; an unpaced burst overflows receiver socket
                # buffers (and starves the event loop's read side)
                for j, pkt in enumerate(pkts):
                    self._send_media(pkt.serialize())
                    if j % 32 == 31:
                        await asyncio.sleep(0)
                    if j % 256 == 255:
                        await asyncio.sleep(0.002)
                counter += 1
        except asyncio.CancelledError:
            pass
        except Exception:
            logger.exception("sender loop failed")
            self._set_state("failed")

    # -- outbound media (SRTP protection when DTLS is active) ------------
    def _send_media(self, data: bytes) -> None:
        """Send one RTP/RTCP datagram, SRTP/SRTCP-protected when the
        session negotiated DTLS; plain otherwise. Media is gated until the
        handshake completes (RFC 5764: no SRTP keys before then)."""
        if self._transport is None or self._remote_addr is None:
            return
        if self._dtls is not None:
            if not self._dtls.established():
                return
            data = (self._dtls.protect_rtcp(data) if is_rtcp(data)
                    else self._dtls.protect_rtp(data))
        self._sendto(data, self._remote_addr)

    def _dtls_on_established(self) -> None:
        from . import dtls as dtls_mod

        self._dtls_verified = True
        actual = self._dtls.peer_fingerprint()
        if not dtls_mod.fingerprints_match(self._dtls_expected_fp or "", actual):
            logger.error("DTLS peer fingerprint mismatch (%s != sdp %s)",
                         actual, self._dtls_expected_fp)
            self._set_state("failed")
            return
        logger.info("DTLS-SRTP established (peer fingerprint verified)")
        self._set_state("connected")

    # -- inbound --------------------------------------------------------
    def _on_datagram(self, data: bytes, addr) -> None:
        if is_stun(data):
            try:
                msg = StunMessage.parse(data)
            except ValueError:
                return
            if msg.msg_type == BINDING_REQUEST and self._transport is not None:
                resp = make_binding_response(msg, addr, self._ice_pwd.encode())
                self._sendto(resp, addr)
                self._remote_addr = addr  # peer-reflexive
                if self._dtls is None:
                    self._set_state("connected")
            return
        # RFC 5764 5.1.2 demux: DTLS record types occupy [20, 63]
        if self._dtls is not None and data and 20 <= data[0] <= 63:
            self._remote_addr = addr
            for out in self._dtls.feed(data):
                self._sendto(out, addr)
            if self._dtls.established() and not self._dtls_verified:
                self._dtls_on_established()
            return
        if self._dtls is not None and data and 128 <= data[0] <= 191:
            if not self._dtls.established():
                return
            data = (self._dtls.unprotect_rtcp(data) if is_rtcp(data)
                    else self._dtls.unprotect_rtp(data))
            if data is None:
                return  # auth failed / out of window: drop
        if is_rtcp(data):
            if parse_pli(data) is not None:
                # peer lost decodability: force a keyframe on the next frame
                self._force_keyframe = True
            rr = parse_rr(data)
            if rr is not None:
                self._adapt_bitrate(rr[1])
            return
        if data[:4] == CONFIG_MAGIC:
            try:
                payload = json.loads(data[4:].decode())
            except (UnicodeDecodeError, json.JSONDecodeError):
                return
            self._emit("datachannel_message", payload)
            return
        try:
            pkt = RtpPacket.parse(data)
        except ValueError:
            return
        self._remote_addr = addr
        if self.connection_state == "connecting":
            self._set_state("connected")
        # lock onto the first media SSRC: browsers may carry rtx or
        # simulcast layers on other SSRCs, which must not interleave into
        # this stream's access-unit reassembly
        if self._rx_lock_ssrc is None:
            self._rx_lock_ssrc = pkt.ssrc
        elif pkt.ssrc != self._rx_lock_ssrc:
            return
        self._account_rx(pkt)
        if getattr(self._decoder, "rtp_mode", "raw") == "rfc6184":
            # buffer the access unit's packets per timestamp and reassemble
            # in SEQUENCE order on the marker (packets may arrive reordered)
            au = self._rx_au.setdefault(pkt.timestamp, {})
            au[pkt.sequence_number] = (pkt.payload, pkt.marker)
            if len(self._rx_au) > 8:  # stale AUs (lost markers): drop oldest
                # order by SIGNED wrapped distance from the first-seen AU
                # timestamp so the drop survives the 32-bit timestamp wrap
                ts_base = next(iter(self._rx_au))
                for old in sorted(
                    self._rx_au,
                    key=lambda t: ((t - ts_base + 0x80000000) & 0xFFFFFFFF)
                    - 0x80000000,
                )[:-4]:
                    self._rx_au.pop(old, None)
            # reassemble only once the AU is COMPLETE: its (unwrapped)
            # sequence numbers form a contiguous run whose last packet
            # carries the marker. A marker that arrives before reordered
            # fragments keeps the AU buffered instead of dropping them
            # (the stale-AU GC above bounds how long we wait).
            if not any(m for (_, m) in au.values()):
                return
            # SIGNED wrapped distance from the first-seen packet: correct
            # ordering across the 0xFFFF->0 wrap and for late low-seq arrivals
            seq_base = next(iter(au))
            ordered = sorted(
                au, key=lambda s: ((s - seq_base + 0x8000) & 0xFFFF) - 0x8000
            )
            offs = [((s - seq_base + 0x8000) & 0xFFFF) - 0x8000 for s in ordered]
            if offs != list(range(offs[0], offs[0] + len(offs))) \
                    or not au[ordered[-1]][1]:
                return  # gap mid-AU or marker not last yet: wait
            self._h264_depack = H264Depacketizer()
            nals = []
            for seq in ordered:
                nal = self._h264_depack.push(au[seq][0])
                if nal is not None:
                    nals.append(nal)
            self._rx_au.pop(pkt.timestamp, None)
            if not nals:
                return
            frame_bytes = join_annexb(nals)
        else:
            frame_bytes = self._defrag.push(pkt)
        if frame_bytes is None:
            return
        try:
            tensor = self._decoder.decode(frame_bytes)
        except Exception:
            logger.exception("decode failed")
            return
        if tensor is None:
            # undecodable (lost reference frame): ask for a keyframe
            self._decode_misses += 1
            now = time.monotonic()
            if self._decode_misses >= 2 and now - self._last_pli > 0.5 \
                    and self._transport is not None and self._remote_addr:
                self._send_media(make_pli(self._packetizer.ssrc, pkt.ssrc))
                self._last_pli = now
            return
        self._decode_misses = 0
        if self._recv_track is None:
            self._recv_track = QueueTrack()
        self._recv_track.push(VideoFrame(tensor=tensor, pts=pkt.timestamp))
        if not self._track_fired:
            self._track_fired = True
            self._emit("track", self._recv_track)

    def _account_rx(self, pkt: RtpPacket) -> None:
        """Track loss from RTP sequence numbers; report every ~128 packets
        so the sender's rate control sees fresh loss fractions."""
        seq = pkt.sequence_number
        if self._rx_base_seq is None:
            self._rx_base_seq = seq
            self._rx_high_seq = seq
        # unwrap 16-bit sequence space
        high = self._rx_high_seq
        diff = (seq - (high & 0xFFFF)) & 0xFFFF
        if 0 < diff < 0x8000:  # in-order advance (16-bit space unwrapped)
            self._rx_high_seq = high + diff
        self._rx_count += 1
        expected = self._rx_high_seq - self._rx_base_seq + 1
        exp0, rcv0 = self._rr_last_sent
        if expected - exp0 >= 128 and self._transport is not None and self._remote_addr:
            interval_exp = expected - exp0
            interval_rcv = self._rx_count - rcv0
            lost_frac = max(0.0, 1.0 - interval_rcv / max(1, interval_exp))
            self._send_media(
                make_rr(self._packetizer.ssrc, pkt.ssrc, lost_frac,
                        max(0, expected - self._rx_count), self._rx_high_seq))
            self._rr_last_sent = (expected, self._rx_count)

    def _adapt_bitrate(self, fraction_lost: float) -> None:
        """AIMD-style sender rate control on RR loss feedback, within the
        EncoderConfig min/max bitrate bounds (the knobs NVENC would own in
        the reference)."""
        cfg = getattr(self._encoder, "cfg", None)
        if cfg is None:
            return
        if fraction_lost > 0.05:
            cfg.default_bitrate = max(cfg.min_bitrate, int(cfg.default_bitrate * 0.7))
        elif fraction_lost < 0.01:
            cfg.default_bitrate = min(cfg.max_bitrate, int(cfg.default_bitrate * 1.08))

    # -- teardown --------------------------------------------------------
    async def close(self) -> None:
        if self.connection_state == "closed":
            return
        self._set_state("closed")
        if self._dtls_task is not None:
            self._dtls_task.cancel()
            self._dtls_task = None
        if self._sender_task is not None:
            self._sender_task.cancel()
            try:
                await self._sender_task
            except (asyncio.CancelledError, Exception):
                pass
        if self._relay is not None:
            self._relay.close()
            self._relay = None
        if self._transport is not None:
            self._transport.close()
            self._transport = None


class MediaRelay:
    """Fan-out of one source track to multiple subscribers (the reference
    imports aiortc's MediaRelay, agent.py:427)."""

    def __init__(self):
        self._subscribers: Dict[int, List[QueueTrack]] = {}
        self._pumps: Dict[int, asyncio.Task] = {}

    def subscribe(self, track: MediaStreamTrack) -> QueueTrack:
        key = id(track)
        sub = QueueTrack()
        self._subscribers.setdefault(key, []).append(sub)
        if key not in self._pumps:
            self._pumps[key] = asyncio.ensure_future(self._pump(key, track))
        return sub

    async def _pump(self, key: int, track: MediaStreamTrack) -> None:
        try:
            while True:
                frame = await track.recv()
                for sub in self._subscribers.get(key, []):
                    sub.push(frame)
        except asyncio.CancelledError:
            pass
        except Exception:
            logger.exception("media relay pump failed")
