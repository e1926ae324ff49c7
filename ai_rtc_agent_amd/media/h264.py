"""H.264 stream utilities: Annex-B NAL handling + RFC 6184 RTP payloads.

The transport half of the VCN hardware codec path:
when the VCN encode session emits Annex-B (SPS/PPS built natively in
ops/csrc/vcn.cpp), these packetizers put it on the wire the way WebRTC
peers expect — single-NAL-unit packets and FU-A fragmentation — and the
depacketizer reassembles access units for the decode session. Fully
exercised by tests today (hardware-independent); the VCN slice encoder
plugs in on a box with the VA userspace.
"""
from __future__ import annotations

from typing import Iterator, List, Optional

FU_A = 28
START_CODE = b"\x00\x00\x00\x01"


def split_annexb(data: bytes) -> List[bytes]:
    """Split an Annex-B stream into raw NAL units (no start codes)."""
    nals = []
    i = 0
    n = len(data)
    # find first start code
    starts = []
    j = 0
    while j < n - 2:
        if data[j] == 0 and data[j + 1] == 0:
            if data[j + 2] == 1:
                starts.append((j, 3))
                j += 3
                continue
            if j < n - 3 and data[j + 2] == 0 and data[j + 3] == 1:
                starts.append((j, 4))
                j += 4
                continue
        j += 1
    for k, (pos, sc) in enumerate(starts):
        end = starts[k + 1][0] if k + 1 < len(starts) else n
        nal = data[pos + sc : end]
        if nal:
            nals.append(nal)
    return nals


def join_annexb(nals: List[bytes]) -> bytes:
    return b"".join(START_CODE + n for n in nals)


def nal_type(nal: bytes) -> int:
    return nal[0] & 0x1F


def packetize_h264(nals: List[bytes], mtu: int = 1188) -> List[bytes]:
    """RFC 6184: single-NAL packets when they fit, FU-A otherwise."""
    payloads: List[bytes] = []
    for nal in nals:
        if len(nal) <= mtu:
            payloads.append(nal)
            continue
        hdr = nal[0]
        nri = hdr & 0x60
        typ = hdr & 0x1F
        fu_indicator = bytes([nri | FU_A])
        body = nal[1:]
        first = True
        while body:
            chunk, body = body[: mtu - 2], body[mtu - 2 :]
            fu_header = typ
            if first:
                fu_header |= 0x80  # S
                first = False
            if not body:
                fu_header |= 0x40  # E
            payloads.append(fu_indicator + bytes([fu_header]) + chunk)
    return payloads


class H264Depacketizer:
    """Reassemble NAL units from RFC 6184 payloads (in-order within a
    timestamp; the RTP layer above handles sequencing)."""

    def __init__(self) -> None:
        self._fu: Optional[bytearray] = None

    def push(self, payload: bytes) -> Optional[bytes]:
        if not payload:
            return None
        typ = payload[0] & 0x1F
        if typ != FU_A:
            self._fu = None
            return payload  # single NAL unit packet
        if len(payload) < 2:
            return None
        fu_header = payload[1]
        if fu_header & 0x80:  # start
            nal_hdr = (payload[0] & 0x60) | (fu_header & 0x1F)
            self._fu = bytearray([nal_hdr])
        if self._fu is None:
            return None  # lost the start fragment
        self._fu += payload[2:]
        if fu_header & 0x40:  # end
            out = bytes(self._fu)
            self._fu = None
            return out
        return None
