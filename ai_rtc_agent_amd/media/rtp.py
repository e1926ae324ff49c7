"""RTP packetization (RFC 3550 subset) + frame fragmentation — from scratch.

Replaces the aiortc fork's RTP layer for our media plane. Video frames are
fragmented FU-A-style: each fragment carries a 2-byte fragment header
(start/end bits + 14-bit fragment index) after the RTP header; the RTP
marker bit flags the last packet of a frame (standard for video RTP).
"""
from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import List, Optional

RTP_VERSION = 2
MAX_PAYLOAD = 1188  # 1200 MTU budget - 12 RTP header


@dataclass
class RtpPacket:
    payload_type: int = 96
    sequence_number: int = 0
    timestamp: int = 0
    ssrc: int = 0
    marker: int = 0
    payload: bytes = b""

    def serialize(self) -> bytes:
        b0 = RTP_VERSION << 6
        b1 = (self.marker << 7) | (self.payload_type & 0x7F)
        return (
            struct.pack(
                "!BBHII",
                b0,
                b1,
                self.sequence_number & 0xFFFF,
                self.timestamp & 0xFFFFFFFF,
                self.ssrc & 0xFFFFFFFF,
            )
            + self.payload
        )

    @staticmethod
    def parse(data: bytes) -> "RtpPacket":
        if len(data) < 12:
            raise ValueError("short RTP packet")
        b0, b1, seq, ts, ssrc = struct.unpack("!BBHII", data[:12])
        if b0 >> 6 != RTP_VERSION:
            raise ValueError("bad RTP version")
        cc = b0 & 0x0F
        offset = 12 + 4 * cc
        if b0 & 0x10:  # extension
            if len(data) < offset + 4:
                raise ValueError("short RTP extension")
            ext_len = struct.unpack("!H", data[offset + 2 : offset + 4])[0]
            offset += 4 + 4 * ext_len
        end = len(data)
        if b0 & 0x20 and end > offset:  # padding: last byte = pad count
            pad = data[-1]
            if 0 < pad <= end - offset:
                end -= pad
        return RtpPacket(
            payload_type=b1 & 0x7F,
            sequence_number=seq,
            timestamp=ts,
            ssrc=ssrc,
            marker=b1 >> 7,
            payload=data[offset:end],
        )


class RtpPacketizer:
    """Fragments encoded frames into RTP packets."""

    def __init__(self, payload_type: int = 96, ssrc: int = 1, clock_rate: int = 90000):
        self.payload_type = payload_type
        self.ssrc = ssrc
        self.clock_rate = clock_rate
        self._seq = 0

    def packetize(self, frame: bytes, timestamp: int) -> List[RtpPacket]:
        chunks = [frame[i : i + MAX_PAYLOAD] for i in range(0, len(frame), MAX_PAYLOAD)] or [b""]
        pkts = []
        n = len(chunks)
        for i, chunk in enumerate(chunks):
            hdr = struct.pack("!H", ((1 if i == 0 else 0) << 15) | (i & 0x3FFF))
            pkts.append(
                RtpPacket(
                    payload_type=self.payload_type,
                    sequence_number=self._seq,
                    timestamp=timestamp,
                    ssrc=self.ssrc,
                    marker=1 if i == n - 1 else 0,
                    payload=hdr + chunk,
                )
            )
            self._seq = (self._seq + 1) & 0xFFFF
        return pkts


class RtpDefragmenter:
    """Reassembles frames from (possibly reordered) RTP packets."""

    def __init__(self):
        self._frames: dict[int, dict[int, bytes]] = {}
        self._done: dict[int, int] = {}  # ts -> expected count (when marker seen)

    def push(self, pkt: RtpPacket) -> Optional[bytes]:
        ts = pkt.timestamp
        if len(pkt.payload) < 2:
            return None
        (h,) = struct.unpack("!H", pkt.payload[:2])
        idx = h & 0x3FFF
        frags = self._frames.setdefault(ts, {})
        frags[idx] = pkt.payload[2:]
        if pkt.marker:
            self._done[ts] = idx + 1
        want = self._done.get(ts)
        if want is not None and len(frags) == want:
            data = b"".join(frags[i] for i in range(want))
            del self._frames[ts]
            del self._done[ts]
            # GC stale partial frames
            if len(self._frames) > 32:
                for old in sorted(self._frames)[:-16]:
                    self._frames.pop(old, None)
                    self._done.pop(old, None)
            return data
        return None
