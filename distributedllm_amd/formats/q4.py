"""q4_0 / q4_1 block quantization codecs (GGJT v3 block layout).

Bit-exact with the GGJT v3 layouts the reference engine consumes
(/root/reference/distllm/tensor_processor.cpp:846-855 lists the ftypes; the
vendored ``quantize`` binary produced the blocks — SURVEY.md §2.2 N4).

Block layout, 32 weights per block:

* q4_0: ``d`` (f16 scale) ‖ 16 bytes of nibbles.  Byte ``j`` holds weight
  ``j`` in its low nibble and weight ``j+16`` in its high nibble.
  Dequant: ``x[j] = d * (q[j] - 8)``.
  Quant: ``amax``-signed scaling — let ``m`` be the element with the largest
  absolute value; ``d = m / -8``; ``q = clamp(round(x/d) + 8, 0, 15)``.
* q4_1: ``d`` (f16) ‖ ``m`` (f16) ‖ 16 nibble bytes.
  Dequant: ``x = d*q + m``; quant: ``d=(max-min)/15``, ``m=min``.

All functions are vectorized numpy; used by provisioning (offline tooling)
and by tests as the ground truth the HIP dequant kernels must match.
"""
from __future__ import annotations

import numpy as np

from typing import Tuple

QK4 = 32  # weights per block
Q4_0_BLOCK_BYTES = 2 + 16
Q4_1_BLOCK_BYTES = 4 + 16


def _check_shape(n: int) -> int:
    if n % QK4 != 0:
        raise ValueError(f"row length {n} is not a multiple of {QK4}")
    return n // QK4


def quantize_q4_0(x: np.ndarray) -> np.ndarray:
    """Quantize a float array (last dim multiple of 32) to q4_0 bytes.

    Returns a uint8 array of shape ``(*x.shape[:-1], nblocks*18)``.
    """
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check_shape(x.shape[-1])
    b = x.reshape(-1, nb, QK4)

    # signed amax: the element with the largest |value|, keeping its sign
    idx = np.argmax(np.abs(b), axis=-1)
    m = np.take_along_axis(b, idx[..., None], axis=-1)[..., 0]
    # llama.cpp semantics: the STORED scale is f16 but the quantization
    # divides by the unrounded f32 scale (quantize_row_q4_0_reference)
    d32 = (m / -8.0).astype(np.float32)
    d = d32.astype(np.float16)
    # sub-tiny |d32| would overflow 1/d32 to inf (int cast of the
    # products is then undefined); such blocks quantize to zeros
    with np.errstate(over="ignore"):
        inv = np.divide(1.0, d32, out=np.zeros_like(d32),
                        where=np.abs(d32) >= np.finfo(np.float32).tiny)
    q = np.clip(np.rint(b * inv[..., None]) + 8, 0, 15).astype(np.uint8)
    lo, hi = q[..., :16], q[..., 16:]
    packed = (lo | (hi << 4)).astype(np.uint8)

    out = np.empty(b.shape[:2] + (Q4_0_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:2] = d[..., None].view(np.uint8).reshape(d.shape + (2,))
    out[..., 2:] = packed
    return out.reshape(lead + (nb * Q4_0_BLOCK_BYTES,))


def dequantize_q4_0(raw: np.ndarray, n: int) -> np.ndarray:
    """Dequantize q4_0 bytes back to f32. ``n`` = row length in weights."""
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check_shape(n)
    lead = raw.shape[:-1]
    if raw.shape[-1] != nb * Q4_0_BLOCK_BYTES:
        raise ValueError(
            f"raw length {raw.shape[-1]} != {nb * Q4_0_BLOCK_BYTES} for n={n}")
    b = raw.reshape(-1, nb, Q4_0_BLOCK_BYTES)
    d = b[..., 0:2].copy().view(np.float16)[..., 0].astype(np.float32)
    qs = b[..., 2:]
    lo = (qs & 0x0F).astype(np.int8) - 8
    hi = (qs >> 4).astype(np.int8) - 8
    q = np.concatenate([lo, hi], axis=-1).astype(np.float32)
    return (q * d[..., None]).reshape(lead + (n,))


def quantize_q4_1(x: np.ndarray) -> np.ndarray:
    """Quantize to q4_1 (min/max affine)."""
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check_shape(x.shape[-1])
    b = x.reshape(-1, nb, QK4)

    mn = b.min(axis=-1)
    mx = b.max(axis=-1)
    d32 = ((mx - mn) / 15.0).astype(np.float32)
    d = d32.astype(np.float16)
    m = mn.astype(np.float16)
    # sub-tiny |d32| would overflow 1/d32 to inf (int cast of the
    # products is then undefined); such blocks quantize to zeros
    with np.errstate(over="ignore"):
        inv = np.divide(1.0, d32, out=np.zeros_like(d32),
                        where=np.abs(d32) >= np.finfo(np.float32).tiny)
    q = np.clip(np.rint((b - mn[..., None]) * inv[..., None]),
                0, 15).astype(np.uint8)
    lo, hi = q[..., :16], q[..., 16:]
    packed = (lo | (hi << 4)).astype(np.uint8)

    out = np.empty(b.shape[:2] + (Q4_1_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:2] = d[..., None].view(np.uint8).reshape(d.shape + (2,))
    out[..., 2:4] = m[..., None].view(np.uint8).reshape(m.shape + (2,))
    out[..., 4:] = packed
    return out.reshape(lead + (nb * Q4_1_BLOCK_BYTES,))


def dequantize_q4_1(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check_shape(n)
    lead = raw.shape[:-1]
    if raw.shape[-1] != nb * Q4_1_BLOCK_BYTES:
        raise ValueError(
            f"raw length {raw.shape[-1]} != {nb * Q4_1_BLOCK_BYTES} for n={n}")
    b = raw.reshape(-1, nb, Q4_1_BLOCK_BYTES)
    d = b[..., 0:2].copy().view(np.float16)[..., 0].astype(np.float32)
    m = b[..., 2:4].copy().view(np.float16)[..., 0].astype(np.float32)
    qs = b[..., 4:]
    lo = (qs & 0x0F).astype(np.float32)
    hi = (qs >> 4).astype(np.float32)
    q = np.concatenate([lo, hi], axis=-1)
    return (q * d[..., None] + m[..., None]).reshape(lead + (n,))


# ---------------------------------------------------------------- q5 / q8
# The remaining GGJT-v3 "classic" block formats the reference engine loads
# (tensor_processor.cpp:210-212 accepts GGML_TYPE_Q5_0/Q5_1/Q8_0):
#
# * q5_0: ``d`` (f16) ‖ ``qh`` (u32, the 5th bits) ‖ 16 nibble bytes.
#   Weight j's low 4 bits sit in nibble j (same j/j+16 byte split as
#   q4_0); bit j of ``qh`` is weight j's 5th bit (j+16 for the high
#   nibbles). Dequant: ``x = d * (q5 - 16)``; quant: signed-amax
#   ``d = m/-16``, ``q5 = clamp(round(x/d) + 16, 0, 31)``.
# * q5_1: ``d`` (f16) ‖ ``m`` (f16) ‖ ``qh`` ‖ 16 nibble bytes.
#   ``x = d*q5 + m``; ``d = (max-min)/31``, ``m = min``.
# * q8_0: ``d`` (f16) ‖ 32 int8. ``x = d*q``; ``d = amax/127``.

Q5_0_BLOCK_BYTES = 2 + 4 + 16
Q5_1_BLOCK_BYTES = 4 + 4 + 16
Q8_0_BLOCK_BYTES = 2 + 32


def _pack_q5(q: np.ndarray) -> Tuple[np.ndarray, np.ndarray]:
    """5-bit values [nb, 32] -> (16 nibble bytes, qh u32 as 4 bytes)."""
    lo5, hi5 = q[..., :16], q[..., 16:]
    packed = ((lo5 & 0xF) | ((hi5 & 0xF) << 4)).astype(np.uint8)
    bits = np.concatenate([(lo5 >> 4) & 1, (hi5 >> 4) & 1],
                          axis=-1).astype(np.uint32)
    qh = (bits << np.arange(32, dtype=np.uint32)).sum(
        axis=-1, dtype=np.uint32)
    return packed, qh.astype("<u4")[..., None].view(np.uint8)


def _unpack_q5(qs: np.ndarray, qh_bytes: np.ndarray) -> np.ndarray:
    """(16 nibble bytes, 4 qh bytes) -> 5-bit values [..., 32]."""
    qh = qh_bytes.copy().view("<u4")[..., 0]
    bits = (qh[..., None] >> np.arange(32, dtype=np.uint32)) & 1
    lo = (qs & 0x0F) | ((bits[..., :16] << 4).astype(np.uint8))
    hi = (qs >> 4) | ((bits[..., 16:] << 4).astype(np.uint8))
    return np.concatenate([lo, hi], axis=-1)


def quantize_q5_0(x: np.ndarray) -> np.ndarray:
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check_shape(x.shape[-1])
    b = x.reshape(-1, nb, QK4)
    idx = np.argmax(np.abs(b), axis=-1)
    m = np.take_along_axis(b, idx[..., None], axis=-1)[..., 0]
    d32 = (m / -16.0).astype(np.float32)
    d = d32.astype(np.float16)
    # sub-tiny |d32| would overflow 1/d32 to inf (int cast of the
    # products is then undefined); such blocks quantize to zeros
    with np.errstate(over="ignore"):
        inv = np.divide(1.0, d32, out=np.zeros_like(d32),
                        where=np.abs(d32) >= np.finfo(np.float32).tiny)
    q = np.clip(np.rint(b * inv[..., None]) + 16, 0, 31).astype(np.uint8)
    packed, qh = _pack_q5(q)
    out = np.empty(b.shape[:2] + (Q5_0_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:2] = d[..., None].view(np.uint8).reshape(d.shape + (2,))
    out[..., 2:6] = qh
    out[..., 6:] = packed
    return out.reshape(lead + (nb * Q5_0_BLOCK_BYTES,))


def dequantize_q5_0(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check_shape(n)
    lead = raw.shape[:-1]
    b = raw.reshape(-1, nb, Q5_0_BLOCK_BYTES)
    d = b[..., 0:2].copy().view(np.float16)[..., 0].astype(np.float32)
    q = _unpack_q5(b[..., 6:], b[..., 2:6]).astype(np.float32) - 16.0
    return (q * d[..., None]).reshape(lead + (n,))


def quantize_q5_1(x: np.ndarray) -> np.ndarray:
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check_shape(x.shape[-1])
    b = x.reshape(-1, nb, QK4)
    mn = b.min(axis=-1)
    mx = b.max(axis=-1)
    d32 = ((mx - mn) / 31.0).astype(np.float32)
    d = d32.astype(np.float16)
    m = mn.astype(np.float16)
    # sub-tiny |d32| would overflow 1/d32 to inf (int cast of the
    # products is then undefined); such blocks quantize to zeros
    with np.errstate(over="ignore"):
        inv = np.divide(1.0, d32, out=np.zeros_like(d32),
                        where=np.abs(d32) >= np.finfo(np.float32).tiny)
    q = np.clip(np.rint((b - mn[..., None]) * inv[..., None]),
                0, 31).astype(np.uint8)
    packed, qh = _pack_q5(q)
    out = np.empty(b.shape[:2] + (Q5_1_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:2] = d[..., None].view(np.uint8).reshape(d.shape + (2,))
    out[..., 2:4] = m[..., None].view(np.uint8).reshape(m.shape + (2,))
    out[..., 4:8] = qh
    out[..., 8:] = packed
    return out.reshape(lead + (nb * Q5_1_BLOCK_BYTES,))


def dequantize_q5_1(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check_shape(n)
    lead = raw.shape[:-1]
    b = raw.reshape(-1, nb, Q5_1_BLOCK_BYTES)
    d = b[..., 0:2].copy().view(np.float16)[..., 0].astype(np.float32)
    m = b[..., 2:4].copy().view(np.float16)[..., 0].astype(np.float32)
    q = _unpack_q5(b[..., 8:], b[..., 4:8]).astype(np.float32)
    return (q * d[..., None] + m[..., None]).reshape(lead + (n,))


def quantize_q8_0(x: np.ndarray) -> np.ndarray:
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check_shape(x.shape[-1])
    b = x.reshape(-1, nb, QK4)
    amax = np.abs(b).max(axis=-1)
    d32 = (amax / 127.0).astype(np.float32)
    d = d32.astype(np.float16)
    # subnormal d32 would overflow 1/d32 to inf and make the int8 cast
    # undefined — treat sub-tiny-scale blocks as all-zero (their values
    # are below f16 resolution anyway) and clamp the quantized range
    tiny = np.finfo(np.float32).tiny
    with np.errstate(over="ignore", invalid="ignore"):
        inv = np.divide(1.0, d32, out=np.zeros_like(d32),
                        where=d32 >= tiny)
        q = np.clip(np.rint(b * inv[..., None]), -127.0, 127.0)
    q = np.nan_to_num(q, nan=0.0, posinf=127.0, neginf=-127.0)
    q = q.astype(np.int8)
    out = np.empty(b.shape[:2] + (Q8_0_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:2] = d[..., None].view(np.uint8).reshape(d.shape + (2,))
    out[..., 2:] = q.view(np.uint8)
    return out.reshape(lead + (nb * Q8_0_BLOCK_BYTES,))


def dequantize_q8_0(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check_shape(n)
    lead = raw.shape[:-1]
    b = raw.reshape(-1, nb, Q8_0_BLOCK_BYTES)
    d = b[..., 0:2].copy().view(np.float16)[..., 0].astype(np.float32)
    q = b[..., 2:].copy().view(np.int8).astype(np.float32)
    return (q * d[..., None]).reshape(lead + (n,))
