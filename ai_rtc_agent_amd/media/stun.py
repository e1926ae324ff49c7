"""STUN (RFC 5389 subset) — binding requests/responses for ICE-lite.

stdlib-only (struct + hmac/hashlib): enough for connectivity checks with an
ICE-lite answerer (we always take the passive role, like a media server).
"""
from __future__ import annotations

import hashlib
import hmac
import os
import struct
import zlib
from dataclasses import dataclass, field
from typing import Dict, Optional, Tuple

MAGIC_COOKIE = 0x2112A442
BINDING_REQUEST = 0x0001
BINDING_RESPONSE = 0x0101

ATTR_USERNAME = 0x0006
ATTR_MESSAGE_INTEGRITY = 0x0008
ATTR_XOR_MAPPED_ADDRESS = 0x0020
ATTR_FINGERPRINT = 0x8028


def is_stun(data: bytes) -> bool:
    return (
        len(data) >= 20
        and data[0] < 4
        and struct.unpack("!I", data[4:8])[0] == MAGIC_COOKIE
    )


@dataclass
class StunMessage:
    msg_type: int
    transaction_id: bytes
    attributes: Dict[int, bytes] = field(default_factory=dict)

    @staticmethod
    def parse(data: bytes) -> "StunMessage":
        if len(data) < 20:
            raise ValueError("short STUN message")
        msg_type, length = struct.unpack("!HH", data[:4])
        tid = data[8:20]
        attrs: Dict[int, bytes] = {}
        off = 20
        end = 20 + length
        while off + 4 <= end:
            at, al = struct.unpack("!HH", data[off : off + 4])
            attrs[at] = data[off + 4 : off + 4 + al]
            off += 4 + al
            off += (4 - off % 4) % 4  # padding
        return StunMessage(msg_type, tid, attrs)

    def serialize(self, integrity_key: Optional[bytes] = None) -> bytes:
        body = b""
        for at, av in self.attributes.items():
            body += struct.pack("!HH", at, len(av)) + av
            body += b"\x00" * ((4 - len(av) % 4) % 4)
        if integrity_key is not None:
            # integrity over header with length including the MI attribute
            hdr = struct.pack("!HHI", self.msg_type, len(body) + 24, MAGIC_COOKIE) + self.transaction_id
            mac = hmac.new(integrity_key, hdr + body, hashlib.sha1).digest()
            body += struct.pack("!HH", ATTR_MESSAGE_INTEGRITY, 20) + mac
        hdr = struct.pack("!HHI", self.msg_type, len(body) + 8, MAGIC_COOKIE) + self.transaction_id
        crc = (zlib.crc32(hdr + body) ^ 0x5354554E) & 0xFFFFFFFF
        body += struct.pack("!HHI", ATTR_FINGERPRINT, 4, crc)
        hdr = struct.pack("!HHI", self.msg_type, len(body), MAGIC_COOKIE) + self.transaction_id
        return hdr + body


def make_binding_request(username: str, key: bytes) -> bytes:
    msg = StunMessage(BINDING_REQUEST, os.urandom(12))
    msg.attributes[ATTR_USERNAME] = username.encode()
    return msg.serialize(integrity_key=key)


def xor_mapped_address(addr: Tuple[str, int], tid: bytes) -> bytes:
    import socket

    ip = socket.inet_aton(addr[0])
    xport = addr[1] ^ (MAGIC_COOKIE >> 16)
    xip = bytes(a ^ b for a, b in zip(ip, struct.pack("!I", MAGIC_COOKIE)))
    return struct.pack("!BBH", 0, 0x01, xport) + xip


def make_binding_response(req: StunMessage, addr: Tuple[str, int], key: bytes) -> bytes:
    msg = StunMessage(BINDING_RESPONSE, req.transaction_id)
    msg.attributes[ATTR_XOR_MAPPED_ADDRESS] = xor_mapped_address(addr, req.transaction_id)
    return msg.serialize(integrity_key=key)
