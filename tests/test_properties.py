"""Property-based invariants (hypothesis): codecs and protocol framing
hold for arbitrary inputs, not just the hand-picked cases."""
import hashlib
import struct

import numpy as np
from hypothesis import given, settings, strategies as st

from distributedllm_amd.cluster import protocol as P
from distributedllm_amd.formats import q4


@settings(max_examples=50, deadline=None)
@given(st.lists(st.floats(min_value=-1e4, max_value=1e4, width=32),
                min_size=32, max_size=96).filter(lambda v: len(v) % 32 == 0))
def test_q4_0_roundtrip_error_bound(vals):
    x = np.array(vals, dtype=np.float32).reshape(1, -1)
    deq = q4.dequantize_q4_0(q4.quantize_q4_0(x), x.shape[-1])
    # per-block error <= one quantization step (the asymmetric [-8, 7]
    # range clips one whole level at the positive extreme)
    for b in range(x.shape[-1] // 32):
        blk = x[0, b * 32:(b + 1) * 32]
        step = max(abs(float(np.float16(np.max(np.abs(blk)) / 8.0))), 1e-12)
        err = np.max(np.abs(deq[0, b * 32:(b + 1) * 32] - blk))
        # one quantization level (the asymmetric [-8, 7] range clips a
        # whole level at the positive extreme) + d's own f16 rounding
        # amplified by |q| <= 8 (relative 2^-11, subnormal-absolute for
        # tiny scales)
        assert err <= step * (1.0 + 8 * 2.0**-11) + 8 * 3e-8 +             1e-6 * max(1.0, step)


@settings(max_examples=50, deadline=None)
@given(st.binary(min_size=0, max_size=2048), st.integers(0, 3))
def test_q4_1_roundtrip_monotone_range(raw, pad):
    n = (len(raw) // 4 // 32) * 32
    if n == 0:
        return
    x = np.frombuffer(raw[:n * 4], dtype=np.float32).copy()
    x[~np.isfinite(x)] = 0.0
    x = np.clip(x, -1e4, 1e4).reshape(1, n)
    deq = q4.dequantize_q4_1(q4.quantize_q4_1(x), n)
    # q4_1 reconstruction stays within the block's [min, max] envelope
    for b in range(n // 32):
        blk = x[0, b * 32:(b + 1) * 32]
        lo, hi = float(blk.min()), float(blk.max())
        span = max(hi - lo, 1e-12)
        got = deq[0, b * 32:(b + 1) * 32]
        assert got.min() >= lo - 0.1 * span - 1e-3
        assert got.max() <= hi + 0.1 * span + 1e-3


class _Sock:
    def __init__(self):
        self.buf = bytearray()
        self.pos = 0

    def sendall(self, b):
        self.buf += b

    def recv(self, n):
        out = bytes(self.buf[self.pos:self.pos + min(n, 37)])
        self.pos += len(out)
        return out


@settings(max_examples=50, deadline=None)
@given(st.text(max_size=64), st.binary(max_size=256),
       st.integers(-2**40, 2**40),
       st.lists(st.floats(allow_nan=False, allow_infinity=False,
                          width=32), max_size=64))
def test_protocol_roundtrip_arbitrary_values(name, blob, num, arr):
    msg = P.RequestUploadEnd(upload_id=num, total_size=abs(num),
                             checksum=name[:64])
    sock = _Sock()
    P.send_message(sock, msg)
    got = P.receive_message(sock)
    assert got == msg

    m2 = P.RequestPropagateForward(
        values=np.array(arr, dtype=np.float32), axis0=len(arr), axis1=1,
        start_pos=abs(num) % 1000)
    sock = _Sock()
    P.send_message(sock, m2)
    got = P.receive_message(sock)
    assert got.axis0 == m2.axis0 and got.start_pos == m2.start_pos
    np.testing.assert_array_equal(got.values, m2.values)


@settings(max_examples=50, deadline=None)
@given(st.lists(st.floats(min_value=-1e4, max_value=1e4, width=32),
                min_size=32, max_size=96).filter(lambda v: len(v) % 32 == 0))
def test_q8_0_roundtrip_error_bound(vals):
    x = np.array(vals, dtype=np.float32).reshape(1, -1)
    deq = q4.dequantize_q8_0(q4.quantize_q8_0(x), x.shape[-1])
    for b in range(x.shape[-1] // 32):
        blk = x[0, b * 32:(b + 1) * 32]
        step = max(abs(float(np.float16(np.max(np.abs(blk)) / 127.0))),
                   1e-12)
        err = np.max(np.abs(deq[0, b * 32:(b + 1) * 32] - blk))
        # 0.5*step from rounding q, + d's own f16 rounding amplified by
        # |q| <= 127: relative 2^-11 for normal d, up to half the min
        # f16 subnormal (2.98e-8) absolute for tiny scales
        assert err <= step * 0.57 + 127 * 3.0e-8 + 1e-6 * step


@settings(max_examples=50, deadline=None)
@given(st.lists(st.floats(min_value=-1e4, max_value=1e4, width=32),
                min_size=32, max_size=96).filter(lambda v: len(v) % 32 == 0),
       st.sampled_from(["q5_0", "q5_1"]))
def test_q5_roundtrip_fixed_point_and_bits(vals, which):
    """Quantization is a fixed point (requantizing the dequantized values
    reproduces the bytes) and every decoded 5-bit value is in [0, 31]."""
    x = np.array(vals, dtype=np.float32).reshape(1, -1)
    n = x.shape[-1]
    quant = q4.quantize_q5_0 if which == "q5_0" else q4.quantize_q5_1
    deq = q4.dequantize_q5_0 if which == "q5_0" else q4.dequantize_q5_1
    raw = quant(x)
    y = deq(raw, n)
    # byte-level idempotence can flip the sign of a ZERO scale in an
    # all-zero block (d = -0.0 vs +0.0 — llama.cpp float math does the
    # same); the VALUE fixed point is the real invariant
    assert np.array_equal(deq(quant(y), n), y)
    bs = q4.Q5_0_BLOCK_BYTES if which == "q5_0" else q4.Q5_1_BLOCK_BYTES
    hoff = 2 if which == "q5_0" else 4
    b = raw.reshape(1, -1, bs)
    q5 = q4._unpack_q5(b[..., hoff + 4:],
                       np.ascontiguousarray(b[..., hoff:hoff + 4]))
    assert q5.max() <= 31
