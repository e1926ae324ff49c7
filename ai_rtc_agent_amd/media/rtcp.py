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


# --- Receiver Reports (RFC 3550 subset): loss feedback for rate control ---
PT_RR = 201


def make_rr(sender_ssrc: int, media_ssrc: int, fraction_lost: float,
            cumulative_lost: int, highest_seq: int) -> bytes:
    # length=7 -> 32 bytes total: header word + sender SSRC + a full RFC 3550
    # report block (SSRC, lost, highest seq, jitter, LSR, DLSR)
    header = bytes([(2 << 6) | 1, PT_RR]) + struct.pack("!H", 7)
    fl = min(255, max(0, int(fraction_lost * 256)))
    cl = min(0xFFFFFF, max(0, cumulative_lost))
    block = struct.pack("!IIBBHIIII", sender_ssrc & 0xFFFFFFFF,
                        media_ssrc & 0xFFFFFFFF, fl,
                        (cl >> 16) & 0xFF, cl & 0xFFFF,
                        highest_seq & 0xFFFFFFFF, 0, 0, 0)
    return header + block


def parse_rr(data: bytes):
    """Returns (media_ssrc, fraction_lost [0..1], cumulative_lost) or None."""
    if not is_rtcp(data) or data[1] != PT_RR or len(data) < 32:
        return None
    _, media_ssrc, fl, cl_hi, cl_lo, _, _, _, _ = struct.unpack("!IIBBHIIII", data[4:32])
    return media_ssrc, fl / 256.0, (cl_hi << 16) | cl_lo
