"""k-quant super-block codecs (q2_K .. q6_K), QK_K = 256.

The reference's loader era recognizes the k-quant ftypes
(/root/reference/distllm/tensor_processor.cpp:846-855) but its vendored
runtime decodes them; this module is the clean-room codec for the same
on-disk super-block layouts:

  q2_K:  84 B = scales[16] (4-bit sc | 4-bit min per 16 w) ‖ qs[64]
         (2-bit) ‖ d:f16 ‖ dmin:f16;      w = d*sc*q - dmin*m
  q3_K: 110 B = hmask[32] ‖ qs[64] (low 2 bits) ‖ scales[12] (6-bit,
         two-plane packing) ‖ d:f16;       w = d*(sc-32)*(q - (hm?0:4))
  q4_K: 144 B = d:f16 ‖ dmin:f16 ‖ scales[12] (6-bit sc/min per 32 w)
         ‖ qs[128] (4-bit);               w = d*sc*q - dmin*m
  q5_K: 176 B = q4_K fields + qh[32] (5th bit);  same affine form
  q6_K: 210 B = ql[128] ‖ qh[64] ‖ scales[16] (int8 per 16 w) ‖ d:f16;
         w = d*sc*(q - 32)

Dequantization follows the upstream bit layout exactly (positions,
shifts, the get_scale_min_k4 6-bit packing, the q3_K two-plane scale
packing). The QUANTIZERS here are simple per-group nearest-rounding
(upstream runs an error-minimizing search), so files we produce are
valid k-quant files but not byte-identical to upstream `quantize`
output — round-trip consistency is asserted by tests.
"""
from __future__ import annotations

import numpy as np

QK_K = 256

Q2_K_BLOCK_BYTES = 16 + 64 + 2 + 2            # 84
Q3_K_BLOCK_BYTES = 32 + 64 + 12 + 2           # 110
Q4_K_BLOCK_BYTES = 2 + 2 + 12 + 128           # 144
Q5_K_BLOCK_BYTES = 2 + 2 + 12 + 128 + 32      # 176
Q6_K_BLOCK_BYTES = 128 + 64 + 16 + 2          # 210


def _check(n: int) -> int:
    if n % QK_K:
        raise ValueError(f"row length {n} not a multiple of QK_K={QK_K}")
    return n // QK_K


def _f16(x: np.ndarray) -> np.ndarray:
    return x.astype(np.float16)


def _safe_inv(d: np.ndarray) -> np.ndarray:
    tiny = np.finfo(np.float32).tiny
    with np.errstate(over="ignore", invalid="ignore"):
        return np.divide(1.0, d, out=np.zeros_like(d), where=d >= tiny)


# ------------------------------------------------------------- q4_K/q5_K
# 6-bit (scale, min) pairs for the 8 32-weight groups, packed in 12
# bytes (upstream get_scale_min_k4): groups 0-3 in bytes 0-3/4-7 low 6
# bits; groups 4-7 split across the nibbles of bytes 8-11 plus the high
# 2 bits of bytes 0-7.

def _pack_scales_k4(sc: np.ndarray, mn: np.ndarray) -> np.ndarray:
    """sc, mn: u8 [..., 8] (6-bit) -> [..., 12] packed bytes."""
    out = np.zeros(sc.shape[:-1] + (12,), dtype=np.uint8)
    out[..., 0:4] = (sc[..., 0:4] & 63) | ((sc[..., 4:8] >> 4) << 6)
    out[..., 4:8] = (mn[..., 0:4] & 63) | ((mn[..., 4:8] >> 4) << 6)
    out[..., 8:12] = (sc[..., 4:8] & 0xF) | ((mn[..., 4:8] & 0xF) << 4)
    return out


def _unpack_scales_k4(p: np.ndarray):
    """[..., 12] -> (sc u8 [..., 8], mn u8 [..., 8])."""
    sc = np.zeros(p.shape[:-1] + (8,), dtype=np.uint8)
    mn = np.zeros_like(sc)
    sc[..., 0:4] = p[..., 0:4] & 63
    mn[..., 0:4] = p[..., 4:8] & 63
    sc[..., 4:8] = (p[..., 8:12] & 0xF) | ((p[..., 0:4] >> 6) << 4)
    mn[..., 4:8] = (p[..., 8:12] >> 4) | ((p[..., 4:8] >> 6) << 4)
    return sc, mn


def _affine_group_params(g: np.ndarray, maxq: int):
    """Per-group affine quantization params: g [..., n] floats ->
    (scale, neg_min) with w ≈ scale*q - neg_min*1, q in [0, maxq]."""
    gmax = g.max(axis=-1)
    gmin = g.min(axis=-1)
    gmin = np.minimum(gmin, 0.0)          # keep min <= 0 so -min >= 0
    gmax = np.maximum(gmax, gmin)
    scale = (gmax - gmin) / maxq
    return scale.astype(np.float32), (-gmin).astype(np.float32)


def quantize_q4_K(x: np.ndarray) -> np.ndarray:
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check(x.shape[-1])
    b = x.reshape(-1, nb, 8, 32)          # 8 groups of 32
    scale, negmin = _affine_group_params(b, 15)
    d = _f16(scale.max(axis=-1) / 63.0).astype(np.float32)   # super scales
    dmin = _f16(negmin.max(axis=-1) / 63.0).astype(np.float32)
    sc = np.clip(np.rint(scale * _safe_inv(d)[..., None]), 0, 63
                 ).astype(np.uint8)
    mn = np.clip(np.rint(negmin * _safe_inv(dmin)[..., None]), 0, 63
                 ).astype(np.uint8)
    # effective reconstruction params
    dg = d[..., None] * sc
    mg = dmin[..., None] * mn
    q = np.clip(np.rint((b + mg[..., None]) * _safe_inv(dg)[..., None]),
                0, 15).astype(np.uint8)
    out = np.zeros(b.shape[:2] + (Q4_K_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:2] = _f16(d)[..., None].view(np.uint8).reshape(d.shape + (2,))
    out[..., 2:4] = _f16(dmin)[..., None].view(np.uint8).reshape(
        dmin.shape + (2,))
    out[..., 4:16] = _pack_scales_k4(sc, mn)
    # qs: per 64-weight pair of groups (2g, 2g+1): byte l = q[2g][l] |
    # q[2g+1][l] << 4
    qs = (b.shape[0], nb, 4, 32)
    lo = q[..., 0::2, :].reshape(qs)
    hi = q[..., 1::2, :].reshape(qs)
    out[..., 16:144] = (lo | (hi << 4)).reshape(b.shape[:2] + (128,))
    return out.reshape(lead + (nb * Q4_K_BLOCK_BYTES,))


def dequantize_q4_K(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check(n)
    lead = raw.shape[:-1]
    blk = raw.reshape(-1, nb, Q4_K_BLOCK_BYTES)
    d = blk[..., 0:2].copy().view(np.float16)[..., 0].astype(np.float32)
    dmin = blk[..., 2:4].copy().view(np.float16)[..., 0].astype(np.float32)
    sc, mn = _unpack_scales_k4(blk[..., 4:16])
    qs = blk[..., 16:144].reshape(blk.shape[:2] + (4, 32))
    lo = (qs & 0xF).astype(np.float32)
    hi = (qs >> 4).astype(np.float32)
    q = np.empty(blk.shape[:2] + (8, 32), dtype=np.float32)
    q[..., 0::2, :] = lo
    q[..., 1::2, :] = hi
    dg = d[..., None] * sc.astype(np.float32)
    mg = dmin[..., None] * mn.astype(np.float32)
    w = dg[..., None] * q - mg[..., None]
    return w.reshape(lead + (n,))


def quantize_q5_K(x: np.ndarray) -> np.ndarray:
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check(x.shape[-1])
    b = x.reshape(-1, nb, 8, 32)
    scale, negmin = _affine_group_params(b, 31)
    d = _f16(scale.max(axis=-1) / 63.0).astype(np.float32)
    dmin = _f16(negmin.max(axis=-1) / 63.0).astype(np.float32)
    sc = np.clip(np.rint(scale * _safe_inv(d)[..., None]), 0, 63
                 ).astype(np.uint8)
    mn = np.clip(np.rint(negmin * _safe_inv(dmin)[..., None]), 0, 63
                 ).astype(np.uint8)
    dg = d[..., None] * sc
    mg = dmin[..., None] * mn
    q = np.clip(np.rint((b + mg[..., None]) * _safe_inv(dg)[..., None]),
                0, 31).astype(np.uint8)
    out = np.zeros(b.shape[:2] + (Q5_K_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:2] = _f16(d)[..., None].view(np.uint8).reshape(d.shape + (2,))
    out[..., 2:4] = _f16(dmin)[..., None].view(np.uint8).reshape(
        dmin.shape + (2,))
    out[..., 4:16] = _pack_scales_k4(sc, mn)
    # qh bit u1 = 1 << (2g) for group 2g (low nibbles), u2 = 1 << (2g+1)
    qs4 = (b.shape[0], nb, 4, 32)
    lo = q[..., 0::2, :].reshape(qs4)
    hi = q[..., 1::2, :].reshape(qs4)
    out[..., 16:48] = sum(
        (((lo[..., j, :] >> 4) & 1) << (2 * j)) |
        (((hi[..., j, :] >> 4) & 1) << (2 * j + 1))
        for j in range(4)).astype(np.uint8)
    out[..., 48:176] = ((lo & 0xF) | ((hi & 0xF) << 4)).reshape(
        b.shape[:2] + (128,))
    return out.reshape(lead + (nb * Q5_K_BLOCK_BYTES,))


def dequantize_q5_K(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check(n)
    lead = raw.shape[:-1]
    blk = raw.reshape(-1, nb, Q5_K_BLOCK_BYTES)
    d = blk[..., 0:2].copy().view(np.float16)[..., 0].astype(np.float32)
    dmin = blk[..., 2:4].copy().view(np.float16)[..., 0].astype(np.float32)
    sc, mn = _unpack_scales_k4(blk[..., 4:16])
    qh = blk[..., 16:48]
    ql = blk[..., 48:176].reshape(blk.shape[:2] + (4, 32))
    q = np.empty(blk.shape[:2] + (8, 32), dtype=np.float32)
    for j in range(4):
        h1 = ((qh >> (2 * j)) & 1).astype(np.float32) * 16.0
        h2 = ((qh >> (2 * j + 1)) & 1).astype(np.float32) * 16.0
        q[..., 2 * j, :] = (ql[..., j, :] & 0xF).astype(np.float32) + h1
        q[..., 2 * j + 1, :] = (ql[..., j, :] >> 4).astype(np.float32) + h2
    dg = d[..., None] * sc.astype(np.float32)
    mg = dmin[..., None] * mn.astype(np.float32)
    w = dg[..., None] * q - mg[..., None]
    return w.reshape(lead + (n,))


# ------------------------------------------------------------------ q6_K

def quantize_q6_K(x: np.ndarray) -> np.ndarray:
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check(x.shape[-1])
    b = x.reshape(-1, nb, 16, 16)          # 16 groups of 16
    # asymmetric range: q-32 in [-32, 31] — scale so neither side clips
    gscale = np.maximum(b.max(axis=-1) / 31.0, b.min(axis=-1) / -32.0)
    d = _f16(gscale.max(axis=-1) / 127.0).astype(np.float32)
    sc = np.clip(np.rint(gscale * _safe_inv(d)[..., None]), -128, 127
                 ).astype(np.int8)
    dg = d[..., None] * sc.astype(np.float32)
    q = np.clip(np.rint(b * _safe_inv(np.abs(dg))[..., None] *
                        np.sign(dg)[..., None]), -32, 31) + 32.0
    q = q.astype(np.uint8).reshape(b.shape[:2] + (2, 8, 16))  # halves of 128
    # per 128-weight half: rows r = 0..7 are 16-weight groups; weight
    # w128[l + 32*k] for l<32, k<4; q6 bit split (upstream):
    #   ql[l]      = w[l]    low | w[l+64]  low << 4
    #   ql[l + 32] = w[l+32] low | w[l+96]  low << 4
    #   qh[l] = hi2(w[l]) | hi2(w[l+32])<<2 | hi2(w[l+64])<<4 | hi2(w[l+96])<<6
    w128 = q.reshape(b.shape[:2] + (2, 128))
    wl = w128[..., 0:32], w128[..., 32:64], w128[..., 64:96], \
        w128[..., 96:128]
    ql = np.concatenate([
        (wl[0] & 0xF) | ((wl[2] & 0xF) << 4),
        (wl[1] & 0xF) | ((wl[3] & 0xF) << 4)], axis=-1)   # [..., 2, 64]
    qh = ((wl[0] >> 4) | ((wl[1] >> 4) << 2) | ((wl[2] >> 4) << 4) |
          ((wl[3] >> 4) << 6))                            # [..., 2, 32]
    out = np.zeros(b.shape[:2] + (Q6_K_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:128] = ql.reshape(b.shape[:2] + (128,))
    out[..., 128:192] = qh.reshape(b.shape[:2] + (64,))
    out[..., 192:208] = sc.view(np.uint8)
    out[..., 208:210] = _f16(d)[..., None].view(np.uint8).reshape(
        d.shape + (2,))
    return out.reshape(lead + (nb * Q6_K_BLOCK_BYTES,))


def dequantize_q6_K(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check(n)
    lead = raw.shape[:-1]
    blk = raw.reshape(-1, nb, Q6_K_BLOCK_BYTES)
    ql = blk[..., 0:128].reshape(blk.shape[:2] + (2, 2, 32))
    qh = blk[..., 128:192].reshape(blk.shape[:2] + (2, 32))
    sc = blk[..., 192:208].copy().view(np.int8).astype(np.float32)
    d = blk[..., 208:210].copy().view(np.float16)[..., 0].astype(np.float32)
    q = np.empty(blk.shape[:2] + (2, 4, 32), dtype=np.float32)
    q[..., 0, :] = ((ql[..., 0, :] & 0xF) |
                    (((qh >> 0) & 3) << 4)).astype(np.float32) - 32.0
    q[..., 1, :] = ((ql[..., 1, :] & 0xF) |
                    (((qh >> 2) & 3) << 4)).astype(np.float32) - 32.0
    q[..., 2, :] = ((ql[..., 0, :] >> 4) |
                    (((qh >> 4) & 3) << 4)).astype(np.float32) - 32.0
    q[..., 3, :] = ((ql[..., 1, :] >> 4) |
                    (((qh >> 6) & 3) << 4)).astype(np.float32) - 32.0
    # scales: group g16 = (half*8) + (k*2) + l//16
    qg = q.reshape(blk.shape[:2] + (2, 4, 2, 16))  # [.., half, k, l16, 16]
    scg = sc.reshape(blk.shape[:2] + (2, 8))       # [.., half, 8]
    dg = d[..., None, None] * scg                  # [.., half, 8]
    dg = dg.reshape(blk.shape[:2] + (2, 1, 8)).reshape(
        blk.shape[:2] + (2, 8))
    w = qg * dg.reshape(blk.shape[:2] + (2, 4, 2))[..., None]
    return w.reshape(lead + (n,))


# ------------------------------------------------------------------ q2_K

def quantize_q2_K(x: np.ndarray) -> np.ndarray:
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check(x.shape[-1])
    b = x.reshape(-1, nb, 16, 16)          # 16 groups of 16
    scale, negmin = _affine_group_params(b, 3)
    d = _f16(scale.max(axis=-1) / 15.0).astype(np.float32)
    dmin = _f16(negmin.max(axis=-1) / 15.0).astype(np.float32)
    sc = np.clip(np.rint(scale * _safe_inv(d)[..., None]), 0, 15
                 ).astype(np.uint8)
    mn = np.clip(np.rint(negmin * _safe_inv(dmin)[..., None]), 0, 15
                 ).astype(np.uint8)
    dg = d[..., None] * sc
    mg = dmin[..., None] * mn
    q = np.clip(np.rint((b + mg[..., None]) * _safe_inv(dg)[..., None]),
                0, 3).astype(np.uint8)
    out = np.zeros(b.shape[:2] + (Q2_K_BLOCK_BYTES,), dtype=np.uint8)
    out[..., 0:16] = sc | (mn << 4)
    # qs: per 128-weight half, 32 bytes; weight (j*32 + l) of the half at
    # byte l bits 2j (j = 0..3 -> shift 0,2,4,6)
    qh = q.reshape(b.shape[:2] + (2, 4, 32))       # [.., half, j, l]
    out[..., 16:80] = sum(
        (qh[..., j, :].astype(np.uint8) << (2 * j)) for j in range(4)
    ).reshape(b.shape[:2] + (2, 32)).reshape(b.shape[:2] + (64,))
    out[..., 80:82] = _f16(d)[..., None].view(np.uint8).reshape(
        d.shape + (2,))
    out[..., 82:84] = _f16(dmin)[..., None].view(np.uint8).reshape(
        dmin.shape + (2,))
    return out.reshape(lead + (nb * Q2_K_BLOCK_BYTES,))


def dequantize_q2_K(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check(n)
    lead = raw.shape[:-1]
    blk = raw.reshape(-1, nb, Q2_K_BLOCK_BYTES)
    sc = (blk[..., 0:16] & 0xF).astype(np.float32)
    mn = (blk[..., 0:16] >> 4).astype(np.float32)
    qs = blk[..., 16:80].reshape(blk.shape[:2] + (2, 32))
    d = blk[..., 80:82].copy().view(np.float16)[..., 0].astype(np.float32)
    dmin = blk[..., 82:84].copy().view(np.float16)[..., 0].astype(
        np.float32)
    q = np.empty(blk.shape[:2] + (2, 4, 32), dtype=np.float32)
    for j in range(4):
        q[..., j, :] = ((qs >> (2 * j)) & 3).astype(np.float32)
    dg = (d[..., None] * sc).reshape(blk.shape[:2] + (2, 4, 2))
    mg = (dmin[..., None] * mn).reshape(blk.shape[:2] + (2, 4, 2))
    qg = q.reshape(blk.shape[:2] + (2, 4, 2, 16))
    w = dg[..., None] * qg - mg[..., None]
    return w.reshape(lead + (n,))


# ------------------------------------------------------------------ q3_K

_KM1 = 0x03030303
_KM2 = 0x0F0F0F0F


def _pack_scales_q3(sc: np.ndarray) -> np.ndarray:
    """int8 6-bit scales [..., 16] (value range [-32, 31] stored +32)
    -> 12 packed bytes (upstream two-plane packing via aux words)."""
    u = (sc.astype(np.int16) + 32).astype(np.uint32)  # [0, 63]
    lead = sc.shape[:-1]
    aux = np.zeros(lead + (4,), dtype=np.uint32)
    for k in range(4):   # aux word k holds 6-bit values 4k..4k+3... via
        pass
    # Upstream unpack: scales int8[16] = aux[0..3] where
    #  aux[0] = (a0 & kmask2) | (((t >> 0) & kmask1) << 4)
    #  aux[1] = (a1 & kmask2) | (((t >> 2) & kmask1) << 4)
    #  aux[2] = ((a0 >> 4) & kmask2) | (((t >> 4) & kmask1) << 4)
    #  aux[3] = ((a1 >> 4) & kmask2) | (((t >> 6) & kmask1) << 4)
    # with a0, a1, t = the three LE u32 of scales[12]. So byte i of the
    # unpacked int8[16]: value v[i] = low nibble from a-planes + high
    # 2 bits from t.
    v = u.reshape(lead + (4, 4))  # [word w][byte i]
    a = np.zeros(lead + (2, 4), dtype=np.uint32)
    t = np.zeros(lead + (4,), dtype=np.uint32)
    for i in range(4):
        a[..., 0, i] = (v[..., 0, i] & 0xF) | ((v[..., 2, i] & 0xF) << 4)
        a[..., 1, i] = (v[..., 1, i] & 0xF) | ((v[..., 3, i] & 0xF) << 4)
        t[..., i] = ((v[..., 0, i] >> 4) << 0) | ((v[..., 1, i] >> 4) << 2) \
            | ((v[..., 2, i] >> 4) << 4) | ((v[..., 3, i] >> 4) << 6)
    out = np.zeros(lead + (12,), dtype=np.uint8)
    out[..., 0:4] = a[..., 0, :].astype(np.uint8)
    out[..., 4:8] = a[..., 1, :].astype(np.uint8)
    out[..., 8:12] = t.astype(np.uint8)
    return out


def _unpack_scales_q3(p: np.ndarray) -> np.ndarray:
    """12 packed bytes -> int8 scales [..., 16] (already -32 biased)."""
    lead = p.shape[:-1]
    v = np.zeros(lead + (4, 4), dtype=np.uint8)
    a0 = p[..., 0:4]
    a1 = p[..., 4:8]
    t = p[..., 8:12]
    v[..., 0, :] = (a0 & 0xF) | (((t >> 0) & 3) << 4)
    v[..., 1, :] = (a1 & 0xF) | (((t >> 2) & 3) << 4)
    v[..., 2, :] = (a0 >> 4) | (((t >> 4) & 3) << 4)
    v[..., 3, :] = (a1 >> 4) | (((t >> 6) & 3) << 4)
    return v.reshape(lead + (16,)).astype(np.int16) - 32


def quantize_q3_K(x: np.ndarray) -> np.ndarray:
    x = np.ascontiguousarray(x, dtype=np.float32)
    lead = x.shape[:-1]
    nb = _check(x.shape[-1])
    b = x.reshape(-1, nb, 16, 16)
    # asymmetric range: q-4 in [-4, 3] — scale so neither side clips
    gscale = np.maximum(b.max(axis=-1) / 3.0, b.min(axis=-1) / -4.0)
    d = _f16(gscale.max(axis=-1) / 31.0).astype(np.float32)
    sc = np.clip(np.rint(gscale * _safe_inv(d)[..., None]), -32, 31
                 ).astype(np.int16)
    dg = d[..., None] * sc.astype(np.float32)
    q = np.clip(np.rint(b * _safe_inv(np.abs(dg))[..., None] *
                        np.sign(dg)[..., None]), -4, 3) + 4.0
    q = q.astype(np.uint8)                 # [0, 7]
    out = np.zeros(b.shape[:2] + (Q3_K_BLOCK_BYTES,), dtype=np.uint8)
    # per 128-half: low 2 bits in qs byte l bits 2j; high bit in
    # hmask[l] bit (half*4 + j)
    qh = q.reshape(b.shape[:2] + (2, 4, 32))
    hm = np.zeros(b.shape[:2] + (32,), dtype=np.uint8)
    qs = np.zeros(b.shape[:2] + (2, 32), dtype=np.uint8)
    for half in range(2):
        for j in range(4):
            w = qh[..., half, j, :]
            qs[..., half, :] |= (w & 3).astype(np.uint8) << (2 * j)
            hm |= ((w >> 2) & 1).astype(np.uint8) << (half * 4 + j)
    out[..., 0:32] = hm
    out[..., 32:96] = qs.reshape(b.shape[:2] + (64,))
    out[..., 96:108] = _pack_scales_q3(sc)
    out[..., 108:110] = _f16(d)[..., None].view(np.uint8).reshape(
        d.shape + (2,))
    return out.reshape(lead + (nb * Q3_K_BLOCK_BYTES,))


def dequantize_q3_K(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    nb = _check(n)
    lead = raw.shape[:-1]
    blk = raw.reshape(-1, nb, Q3_K_BLOCK_BYTES)
    hm = blk[..., 0:32]
    qs = blk[..., 32:96].reshape(blk.shape[:2] + (2, 32))
    sc = _unpack_scales_q3(blk[..., 96:108]).astype(np.float32)
    d = blk[..., 108:110].copy().view(np.float16)[..., 0].astype(
        np.float32)
    q = np.empty(blk.shape[:2] + (2, 4, 32), dtype=np.float32)
    for half in range(2):
        for j in range(4):
            low = ((qs[..., half, :] >> (2 * j)) & 3).astype(np.int16)
            hi = ((hm >> (half * 4 + j)) & 1).astype(np.int16)
            # upstream: q = low2 - (high_bit_set ? 0 : 4)
            q[..., half, j, :] = (low - np.where(hi != 0, 0, 4)).astype(
                np.float32)
    dg = (d[..., None] * sc).reshape(blk.shape[:2] + (2, 4, 2))
    qg = q.reshape(blk.shape[:2] + (2, 4, 2, 16))
    w = dg[..., None] * qg
    return w.reshape(lead + (n,))


CODECS = {
    "q2_K": (Q2_K_BLOCK_BYTES, quantize_q2_K, dequantize_q2_K),
    "q3_K": (Q3_K_BLOCK_BYTES, quantize_q3_K, dequantize_q3_K),
    "q4_K": (Q4_K_BLOCK_BYTES, quantize_q4_K, dequantize_q4_K),
    "q5_K": (Q5_K_BLOCK_BYTES, quantize_q5_K, dequantize_q5_K),
    "q6_K": (Q6_K_BLOCK_BYTES, quantize_q6_K, dequantize_q6_K),
}
