"""Property-based tests (hypothesis) over the pure-CPU contracts:
codec round-trips at arbitrary dims/QP, RFC 6184 packetization, the fp8
quantization error law, and fused-scheduler parity. Example counts are
bounded to keep the suite fast."""
import math

import pytest
import torch
from hypothesis import given, settings, strategies as st

from ai_rtc_agent_amd import ops


def _ext():
    from ai_rtc_agent_amd.ops import _load_ext

    try:
        return _load_ext.load()
    except ImportError:
        pytest.skip("extension not built")


@settings(max_examples=20, deadline=None)
@given(
    mbw=st.integers(1, 6),
    mbh=st.integers(1, 6),
    qp=st.integers(10, 48),
    seed=st.integers(0, 2**31 - 1),
    slices=st.integers(1, 8),
)
def test_h264_roundtrip_any_dims(mbw, mbh, qp, seed, slices):
    ext = _ext()
    w, h = mbw * 16, mbh * 16
    g = torch.Generator().manual_seed(seed)
    f1 = torch.randint(0, 255, (h, w, 3), generator=g, dtype=torch.uint8)
    f2 = torch.randint(0, 255, (h, w, 3), generator=g, dtype=torch.uint8)
    enc = ext.H264SwEncoder(w, h, slices, 2)
    dec = ext.H264SwDecoder()
    r = dec.decode(enc.encode(f1.numpy().tobytes(), qp, keyframe=True))
    assert r is not None and (r[1], r[2]) == (w, h)
    rp = dec.decode(enc.encode(f2.numpy().tobytes(), qp, keyframe=False))
    assert rp is not None and (rp[1], rp[2]) == (w, h)


@settings(max_examples=30, deadline=None)
@given(
    sizes=st.lists(st.integers(1, 6000), min_size=1, max_size=5),
    mtu=st.integers(60, 1500),
    seed=st.integers(0, 2**31 - 1),
)
def test_rfc6184_roundtrip_property(sizes, mtu, seed):
    import random

    from ai_rtc_agent_amd.media.h264 import H264Depacketizer, packetize_h264

    rng = random.Random(seed)
    nals = [bytes([0x65]) + bytes(rng.randrange(256) for _ in range(n - 1))
            for n in sizes]
    payloads = packetize_h264(nals, mtu=mtu)
    assert all(len(p) <= mtu for p in payloads)
    d = H264Depacketizer()
    out = [n for n in (d.push(p) for p in payloads) if n is not None]
    assert out == nals


@settings(max_examples=40, deadline=None)
@given(
    scale_exp=st.floats(-8, 4),
    seed=st.integers(0, 2**31 - 1),
)
def test_fp8_roundtrip_error_law(scale_exp, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(512, generator=g) * (2.0 ** scale_exp)
    scale = max(x.abs().max().item(), 1e-12) / ops.FP8_MAX
    y = ops.fp8_roundtrip(x, scale)
    err = (y - x).abs()
    bound = x.abs() * 2 ** -4 + scale * 2 ** -9 + 1e-12
    assert (err <= bound + 1e-9).all()


@settings(max_examples=25, deadline=None)
@given(
    steps=st.lists(st.integers(0, 49), min_size=1, max_size=4, unique=True),
    fbs=st.integers(1, 3),
    seed=st.integers(0, 2**31 - 1),
)
def test_sched_fused_parity_property(steps, fbs, seed):
    from ai_rtc_agent_amd.engine.scheduler import StreamScheduler

    sch = StreamScheduler(num_inference_steps=50)
    co = sch.coefficients(sorted(steps), fbs, torch.device("cpu"), torch.float32)
    B = len(steps) * fbs
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(B, 4, 4, 4, generator=g)
    e = torch.randn(B, 4, 4, 4, generator=g)
    got = ops.sched_blend(x, e, co["alpha_f32"], co["beta_f32"],
                          co["c_out_f32"], co["c_skip_f32"])
    ref = sch.step_batch(e, x, co)
    assert torch.allclose(got, ref, atol=1e-5)
    got2 = ops.sched_add_noise(x, e, co["alpha_f32"], co["beta_f32"])
    ref2 = sch.add_noise(x, e, co["alpha_prod_t_sqrt"], co["beta_prod_t_sqrt"])
    assert torch.allclose(got2, ref2, atol=1e-5)


@settings(max_examples=40, deadline=None)
@given(
    pt=st.integers(0, 127),
    seq=st.integers(0, 0xFFFF),
    ts=st.integers(0, 0xFFFFFFFF),
    ssrc=st.integers(0, 0xFFFFFFFF),
    marker=st.integers(0, 1),
    n_csrc=st.integers(0, 15),
    ext_words=st.one_of(st.none(), st.integers(0, 8)),
    pad=st.integers(0, 16),
    payload=st.binary(min_size=0, max_size=512),
)
def test_rtp_parse_handles_csrc_extension_padding(pt, seq, ts, ssrc, marker,
                                                  n_csrc, ext_words, pad,
                                                  payload):
    """Wire-realistic RTP: CSRC list, header extension and padding around a
    serialized packet must parse back to the same logical fields."""
    import struct as _s

    from ai_rtc_agent_amd.media.rtp import RtpPacket

    b0 = (2 << 6) | n_csrc
    if ext_words is not None:
        b0 |= 0x10
    if pad:
        b0 |= 0x20
    b1 = (marker << 7) | pt
    data = _s.pack("!BBHII", b0, b1, seq, ts, ssrc)
    data += bytes(4 * n_csrc)
    if ext_words is not None:
        data += _s.pack("!HH", 0xBEDE, ext_words) + bytes(4 * ext_words)
    data += payload
    if pad:
        data += bytes(pad - 1) + bytes([pad])
    pkt = RtpPacket.parse(data)
    assert (pkt.payload_type, pkt.sequence_number, pkt.timestamp, pkt.ssrc,
            pkt.marker) == (pt, seq, ts, ssrc, marker)
    assert pkt.payload == payload


def test_srtp_property_roundtrip_sizes():
    """SRTP protect/unprotect round-trips across payload sizes and detects
    tampering (one handshake, many packets — RFC 3711 AES-CTR + HMAC)."""
    import struct as _s

    from tests.test_dtls import _endpoint_cls, _handshake

    E = _endpoint_cls()
    cli, srv = E(server=False), E(server=True)
    _handshake(cli, srv)
    import random

    rng = random.Random(5)
    for i, size in enumerate([0, 1, 2, 15, 16, 17, 159, 160, 161, 1200, 1471]):
        hdr = _s.pack("!BBHII", 0x80, 96, 100 + i, 90000 + i, 0xABCD0123)
        pkt = hdr + bytes(rng.randrange(256) for _ in range(size))
        prot = cli.protect_rtp(pkt)
        assert srv.unprotect_rtp(prot) == pkt
        if len(prot) > 14:
            bad = bytearray(prot)
            bad[rng.randrange(12, len(bad))] ^= 0xFF
            assert srv.unprotect_rtp(bytes(bad)) is None
