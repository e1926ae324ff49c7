"""Minimal RTCP (RFC 4585 subset): Picture Loss Indication.

Packet loss desyncs predictive codecs until the next keyframe; a receiver
that can't decode sends PLI and the sender responds with an immediate
keyframe. This is the feedback half of real-time rate/recovery control
(the reference inherits it from aiortc's RTCP machinery)."""
from __future__ import annotations

import struct

PT_PSFB = 206
FMT_PLI = 1


def is_rtcp(data: bytes) -> bool:
    return len(data) >= 8 and (data[0] >> 6) == 2 and 200 <= data[1] <= 206


def make_pli(sender_ssrc: int, media_ssrc: int) -> bytes:
    header = bytes([(2 << 6) | FMT_PLI, PT_PSFB]) + struct.pack("!H", 2)
    return header + struct.pack("!II", sender_ssrc & 0xFFFFFFFF, media_ssrc & 0xFFFFFFFF)


def parse_pli(data: bytes):
    """Returns (sender_ssrc, media_ssrc) or None if not a PLI."""
    if not is_rtcp(data) or data[1] != PT_PSFB or (data[0] & 0x1F) != FMT_PLI:
        return None
    if len(data) < 12:
        return None
    return struct.unpack("!II", data[4:12])
