"""Minimal SDP (RFC 4566 / JSEP subset) — from scratch.

The reference delegates SDP to aiortc (not available in this environment);
the signalling surface it exposes is SDP offer/answer over HTTP for three
modes (/offer, /whip, /whep — reference agent.py:123-395). This module
implements the subset those flows need: session/media sections, H264 codec
preference (reference force_codec, agent.py:72-77), ICE credentials and host
candidates, and answer generation.
"""
from __future__ import annotations

import random
import string
from dataclasses import dataclass, field
from typing import Dict, List, Optional


def _rand(n: int) -> str:
    return "".join(random.choice(string.ascii_letters + string.digits) for _ in range(n))


@dataclass
class RtpCodec:
    payload_type: int
    name: str
    clock_rate: int
    parameters: str = ""


@dataclass
class MediaSection:
    kind: str = "video"
    port: int = 9
    protocol: str = "UDP/TLS/RTP/SAVPF"
    codecs: List[RtpCodec] = field(default_factory=list)
    ice_ufrag: str = ""
    ice_pwd: str = ""
    candidates: List[str] = field(default_factory=list)
    direction: str = "sendrecv"
    mid: str = "0"
    ssrc: Optional[int] = None
    setup: str = "actpass"
    fingerprint: Optional[str] = None  # "sha-256 AA:BB:..." (RFC 8122)

    def codec_by_name(self, name: str) -> Optional[RtpCodec]:
        for c in self.codecs:
            if c.name.lower() == name.lower():
                return c
        return None


@dataclass
class SessionDescription:
    session_id: str = ""
    media: List[MediaSection] = field(default_factory=list)
    fingerprint: Optional[str] = None  # session-level a=fingerprint
    bundle: Optional[str] = None       # a=group:BUNDLE mids (echoed back)
    ice_lite: bool = False             # a=ice-lite (we answer as lite agent)

    @staticmethod
    def parse(sdp: str) -> "SessionDescription":
        sd = SessionDescription()
        cur: Optional[MediaSection] = None
        rtpmap: Dict[int, RtpCodec] = {}
        for raw in sdp.replace("\r\n", "\n").split("\n"):
            line = raw.strip()
            if not line:
                continue
            if line.startswith("o="):
                parts = line[2:].split()
                if len(parts) >= 2:
                    sd.session_id = parts[1]
            elif line.startswith("a=fingerprint:") and cur is None:
                sd.fingerprint = line.split(":", 1)[1].strip()
            elif line.startswith("a=group:BUNDLE") and cur is None:
                sd.bundle = line[len("a=group:"):].strip()
            elif line == "a=ice-lite" and cur is None:
                sd.ice_lite = True
            elif line.startswith("m="):
                parts = line[2:].split()
                cur = MediaSection(kind=parts[0], port=int(parts[1]), protocol=parts[2])
                rtpmap = {}
                for pt in parts[3:]:
                    try:
                        c = RtpCodec(int(pt), "", 90000)
                        cur.codecs.append(c)
                        rtpmap[c.payload_type] = c
                    except ValueError:
                        pass
                sd.media.append(cur)
            elif cur is not None:
                if line.startswith("a=rtpmap:"):
                    body = line[len("a=rtpmap:"):]
                    pt_s, rest = body.split(" ", 1)
                    name, _, clock = rest.partition("/")
                    pt = int(pt_s)
                    if pt in rtpmap:
                        rtpmap[pt].name = name
                        rtpmap[pt].clock_rate = int(clock.split("/")[0] or 90000)
                elif line.startswith("a=fmtp:"):
                    body = line[len("a=fmtp:"):]
                    pt_s, _, params = body.partition(" ")
                    pt = int(pt_s)
                    if pt in rtpmap:
                        rtpmap[pt].parameters = params
                elif line.startswith("a=ice-ufrag:"):
                    cur.ice_ufrag = line.split(":", 1)[1]
                elif line.startswith("a=ice-pwd:"):
                    cur.ice_pwd = line.split(":", 1)[1]
                elif line.startswith("a=candidate:"):
                    cur.candidates.append(line[2:])
                elif line.startswith("a=mid:"):
                    cur.mid = line.split(":", 1)[1]
                elif line.startswith("a=setup:"):
                    cur.setup = line.split(":", 1)[1]
                elif line.startswith("a=fingerprint:"):
                    cur.fingerprint = line.split(":", 1)[1].strip()
                elif line.startswith("a=ssrc:"):
                    try:
                        cur.ssrc = int(line[len("a=ssrc:"):].split()[0])
                    except (ValueError, IndexError):
                        pass
                elif line in ("a=sendrecv", "a=sendonly", "a=recvonly", "a=inactive"):
                    cur.direction = line[2:]
        return sd

    def serialize(self) -> str:
        lines = [
            "v=0",
            f"o=- {self.session_id or random.randint(10**8, 10**9)} 2 IN IP4 127.0.0.1",
            "s=-",
            "t=0 0",
        ]
        if self.ice_lite:
            # RFC 8445 ICE-lite: tells the full agent (the browser) it owns
            # the connectivity checks and the controlling role
            lines.append("a=ice-lite")
        if self.bundle:
            lines.append(f"a=group:{self.bundle}")
        for m in self.media:
            pts = " ".join(str(c.payload_type) for c in m.codecs)
            lines.append(f"m={m.kind} {m.port} {m.protocol} {pts}")
            lines.append("c=IN IP4 0.0.0.0")
            if m.ice_ufrag:
                lines.append(f"a=ice-ufrag:{m.ice_ufrag}")
            if m.ice_pwd:
                lines.append(f"a=ice-pwd:{m.ice_pwd}")
            lines.append(f"a=mid:{m.mid}")
            lines.append(f"a={m.direction}")
            lines.append(f"a=setup:{m.setup}")
            # RTP and RTCP share the one media socket (RFC 5761); browsers
            # offer rtcp-mux and expect the answer to accept it
            lines.append("a=rtcp-mux")
            if m.fingerprint:
                lines.append(f"a=fingerprint:{m.fingerprint}")
            for c in m.codecs:
                lines.append(f"a=rtpmap:{c.payload_type} {c.name}/{c.clock_rate}")
                if c.parameters:
                    lines.append(f"a=fmtp:{c.payload_type} {c.parameters}")
            if m.ssrc is not None:
                lines.append(f"a=ssrc:{m.ssrc} cname:airtc")
            for cand in m.candidates:
                lines.append(f"a={cand}")
        return "\r\n".join(lines) + "\r\n"


def prefer_codec(section: MediaSection, name: str) -> None:
    """Reorder so `name` is the (only) negotiated codec — the reference
    forces H264 on every video transceiver (agent.py:72-77,148-152)."""
    chosen = [c for c in section.codecs if c.name.lower() == name.lower()]
    if chosen:
        section.codecs = chosen


def build_answer(
    offer: "SessionDescription",
    host: str,
    port: int,
    codec_name: str,
    ssrc: int,
    direction: str = "sendrecv",
    fingerprint: Optional[str] = None,
) -> "SessionDescription":
    """Answer an offer: echo media sections, pick our codec, attach our ICE
    credentials + host candidate (+ DTLS fingerprint, setup:passive — the
    offerer is the DTLS client)."""
    ans = SessionDescription(session_id=str(random.randint(10**8, 10**9)),
                             bundle=offer.bundle, ice_lite=True)
    # one transport for the whole (bundled) answer -> ONE ICE credential
    # pair shared by every m-section (RFC 8843: same transport, same
    # ufrag/pwd; browsers reject per-section credentials within a bundle)
    ufrag, pwd = _rand(8), _rand(24)
    for i, m in enumerate(offer.media):
        sec = MediaSection(
            kind=m.kind,
            port=port,
            protocol=m.protocol,
            mid=m.mid or str(i),
            # browsers typically offer audio+video; this agent serves video
            # only, so every other kind is answered a=inactive (the section
            # must still be echoed for BUNDLE mid alignment)
            direction=direction if m.kind == "video" else "inactive",
            ice_ufrag=ufrag,
            ice_pwd=pwd,
            ssrc=ssrc,
            setup="passive",
            fingerprint=fingerprint,
        )
        codec = m.codec_by_name(codec_name)
        if codec is None and m.codecs:
            codec = m.codecs[0]
        if codec is None:
            codec = RtpCodec(96, codec_name, 90000)
        sec.codecs = [codec]
        sec.candidates = [
            f"candidate:1 1 udp 2130706431 {host} {port} typ host"
        ]
        ans.media.append(sec)
    return ans
