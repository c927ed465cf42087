"""Slice engines: the HIP/CDNA4 production engine and a CPU torch twin.

Both expose the same *stateless* interface (positions and sequence ids are
explicit per-token arrays), which is what makes the decode step
graph-capturable and the pipeline testable on CPU:

    forward(x[T,E] f32, pos[T] i32, seq[T] i32) -> x'   (KV appended)
    embed(tokens[T]) -> x[T,E]
    logits(x[T,E], all_logits=False) -> [T|1, V]

``clear_context`` in the reference (tensor_processor.cpp:1512-1521 destroys
and recreates the llama_context) is here simply "start writing positions
from 0 again" — cache slots are overwritten, nothing to reset.

The HIP engine repacks on-disk q4 AoS blocks into the SoA layout the
kernels stream (scales ‖ nibbles, see csrc/kernels.hip); repacking happens
once at load on the host, uploads via torch pinned copies.
"""
from __future__ import annotations

import math
from typing import Dict, Optional

import numpy as np
import torch

from ..formats import ggml, q4
from ..models.llama import RMS_EPS, ROPE_BASE, rms_norm, rope_interleaved


def _nibbles(t: ggml.GGMLTensor) -> np.ndarray:
    """q4 tensor -> raw nibble values [rows, nb, 32] (u8, order = weights)."""
    rows, cols = t.shape_rows_cols
    nb = cols // 32
    bs = 18 if t.gtype == ggml.GGML_TYPE_Q4_0 else 20
    a = np.frombuffer(t.raw, np.uint8).reshape(rows, nb, bs)
    qs = a[:, :, bs - 16:]
    lo = qs & 0x0F
    hi = qs >> 4
    return np.concatenate([lo, hi], axis=-1)  # weight j / j+16 layout


_BYTE_GTYPES = (ggml.GGML_TYPE_Q5_0, ggml.GGML_TYPE_Q5_1,
                ggml.GGML_TYPE_Q8_0)
# k-quants with per-32 affine sub-blocks ride W_Q8B; per-16 formats
# (q2_K/q3_K/q6_K) ride W_Q8B16 (two scale planes)
_K32_GTYPES = (ggml.GGML_TYPE_Q4_K, ggml.GGML_TYPE_Q5_K)
_K16_GTYPES = (ggml.GGML_TYPE_Q2_K, ggml.GGML_TYPE_Q3_K,
               ggml.GGML_TYPE_Q6_K)
W_Q8B16 = 9  # kernels.h WType


def _kquant_byte_values(t: ggml.GGMLTensor):
    """k-quant tensor -> (u8 values [rows, nb32, 32] for the byte
    kernel's w = alpha*(u-128) + beta form, alpha, beta).

    Per-32 formats return alpha/beta [rows, nb32]; per-16 formats
    return [rows, nb32, 2] (two 16-weight half-planes).
    """
    from ..formats import kquants as KQ
    rows, cols = t.shape_rows_cols
    nsb = cols // KQ.QK_K
    u8 = np.frombuffer(t.raw, np.uint8)
    if t.gtype == ggml.GGML_TYPE_Q4_K:
        b = u8.reshape(rows, nsb, KQ.Q4_K_BLOCK_BYTES)
        d = np.ascontiguousarray(b[:, :, 0:2]).view(np.float16)
        dmin = np.ascontiguousarray(b[:, :, 2:4]).view(np.float16)
        d = d.reshape(rows, nsb).astype(np.float32)
        dmin = dmin.reshape(rows, nsb).astype(np.float32)
        sc, mn = KQ._unpack_scales_k4(b[:, :, 4:16])
        qs = b[:, :, 16:144].reshape(rows, nsb, 4, 32)
        q = np.empty((rows, nsb, 8, 32), dtype=np.uint8)
        q[:, :, 0::2] = qs & 0xF
        q[:, :, 1::2] = qs >> 4
        alpha = (d[..., None] * sc).reshape(rows, nsb * 8)
        beta = (-(dmin[..., None] * mn)).reshape(rows, nsb * 8)
        vals = (q + 128).reshape(rows, nsb * 8, 32)
        return vals, alpha.astype(np.float32), beta.astype(np.float32)
    if t.gtype == ggml.GGML_TYPE_Q5_K:
        b = u8.reshape(rows, nsb, KQ.Q5_K_BLOCK_BYTES)
        d = np.ascontiguousarray(b[:, :, 0:2]).view(np.float16)
        dmin = np.ascontiguousarray(b[:, :, 2:4]).view(np.float16)
        d = d.reshape(rows, nsb).astype(np.float32)
        dmin = dmin.reshape(rows, nsb).astype(np.float32)
        sc, mn = KQ._unpack_scales_k4(b[:, :, 4:16])
        qh = b[:, :, 16:48]
        ql = b[:, :, 48:176].reshape(rows, nsb, 4, 32)
        q = np.empty((rows, nsb, 8, 32), dtype=np.uint8)
        for j in range(4):
            q[:, :, 2 * j] = (ql[:, :, j] & 0xF) | \
                (((qh >> (2 * j)) & 1) << 4)
            q[:, :, 2 * j + 1] = (ql[:, :, j] >> 4) | \
                (((qh >> (2 * j + 1)) & 1) << 4)
        alpha = (d[..., None] * sc).reshape(rows, nsb * 8)
        beta = (-(dmin[..., None] * mn)).reshape(rows, nsb * 8)
        vals = (q + 128).reshape(rows, nsb * 8, 32)
        return vals, alpha.astype(np.float32), beta.astype(np.float32)
    if t.gtype == ggml.GGML_TYPE_Q6_K:
        b = u8.reshape(rows, nsb, KQ.Q6_K_BLOCK_BYTES)
        ql = b[:, :, 0:128].reshape(rows, nsb, 2, 2, 32)
        qh = b[:, :, 128:192].reshape(rows, nsb, 2, 32)
        sc = np.ascontiguousarray(b[:, :, 192:208]).view(np.int8)
        sc = sc.reshape(rows, nsb, 16).astype(np.float32)
        d = np.ascontiguousarray(b[:, :, 208:210]).view(np.float16)
        d = d.reshape(rows, nsb).astype(np.float32)
        q = np.empty((rows, nsb, 2, 4, 32), dtype=np.int16)
        q[:, :, :, 0] = (ql[:, :, :, 0] & 0xF) | (((qh >> 0) & 3) << 4)
        q[:, :, :, 1] = (ql[:, :, :, 1] & 0xF) | (((qh >> 2) & 3) << 4)
        q[:, :, :, 2] = (ql[:, :, :, 0] >> 4) | (((qh >> 4) & 3) << 4)
        q[:, :, :, 3] = (ql[:, :, :, 1] >> 4) | (((qh >> 6) & 3) << 4)
        # u = (q - 32) + 128; alpha16 = d*sc, beta16 = 0
        vals = (q + 96).astype(np.uint8).reshape(rows, nsb * 8, 32)
        alpha16 = (d[..., None] * sc).reshape(rows, nsb, 2, 4, 2)
        alpha16 = alpha16.transpose(0, 1, 2, 3, 4).reshape(
            rows, nsb * 8, 2)
        beta16 = np.zeros_like(alpha16)
        return vals, alpha16.astype(np.float32), beta16.astype(np.float32)
    if t.gtype == ggml.GGML_TYPE_Q2_K:
        b = u8.reshape(rows, nsb, KQ.Q2_K_BLOCK_BYTES)
        sc = (b[:, :, 0:16] & 0xF).astype(np.float32)
        mn = (b[:, :, 0:16] >> 4).astype(np.float32)
        qs = b[:, :, 16:80].reshape(rows, nsb, 2, 32)
        d = np.ascontiguousarray(b[:, :, 80:82]).view(np.float16)
        dmin = np.ascontiguousarray(b[:, :, 82:84]).view(np.float16)
        d = d.reshape(rows, nsb).astype(np.float32)
        dmin = dmin.reshape(rows, nsb).astype(np.float32)
        q = np.empty((rows, nsb, 2, 4, 32), dtype=np.uint8)
        for j in range(4):
            q[:, :, :, j] = (qs >> (2 * j)) & 3
        vals = (q + 128).reshape(rows, nsb * 8, 32)
        alpha16 = (d[..., None] * sc).reshape(rows, nsb * 8, 2)
        beta16 = (-(dmin[..., None] * mn)).reshape(rows, nsb * 8, 2)
        return vals, alpha16.astype(np.float32), beta16.astype(np.float32)
    assert t.gtype == ggml.GGML_TYPE_Q3_K, t.gtype
    b = u8.reshape(rows, nsb, KQ.Q3_K_BLOCK_BYTES)
    hm = b[:, :, 0:32]
    qs = b[:, :, 32:96].reshape(rows, nsb, 2, 32)
    from ..formats.kquants import _unpack_scales_q3
    sc = _unpack_scales_q3(b[:, :, 96:108]).astype(np.float32)
    d = np.ascontiguousarray(b[:, :, 108:110]).view(np.float16)
    d = d.reshape(rows, nsb).astype(np.float32)
    q = np.empty((rows, nsb, 2, 4, 32), dtype=np.int16)
    for half in range(2):
        for j in range(4):
            low = ((qs[:, :, half] >> (2 * j)) & 3).astype(np.int16)
            hi = ((hm >> (half * 4 + j)) & 1).astype(np.int16)
            q[:, :, half, j] = low - np.where(hi != 0, 0, 4)
    vals = (q + 128).astype(np.uint8).reshape(rows, nsb * 8, 32)
    alpha16 = (d[..., None] * sc).reshape(rows, nsb * 8, 2)
    beta16 = np.zeros_like(alpha16)
    return vals, alpha16.astype(np.float32), beta16.astype(np.float32)


def _byte_values(t: ggml.GGMLTensor):
    """q5_0/q5_1/q8_0 tensor -> (u8 values [rows, nb, 32] re-biased for
    the kernel's unified csub=-1152, alpha f32, beta f32).

    The kernel computes w = alpha*((1024+u) - 1152) + beta, so:
      q8_0: u = q + 128          -> alpha=d, beta=0
      q5_0: u = q5 + 112         -> alpha=d, beta=0   (q5-16 exact)
      q5_1: u = q5 + 112         -> alpha=d, beta=m + 16*d
    """
    rows, cols = t.shape_rows_cols
    nb = cols // 32
    u8 = np.frombuffer(t.raw, np.uint8)
    if t.gtype == ggml.GGML_TYPE_Q8_0:
        a = u8.reshape(rows, nb, 34)
        d = np.ascontiguousarray(a[:, :, :2]).view(np.float16)
        alpha = d.reshape(rows, nb).astype(np.float32)
        beta = np.zeros_like(alpha)
        vals = (a[:, :, 2:] ^ 0x80).astype(np.uint8)  # int8 + 128
        return vals, alpha, beta
    bs = 22 if t.gtype == ggml.GGML_TYPE_Q5_0 else 24
    hoff = 2 if t.gtype == ggml.GGML_TYPE_Q5_0 else 4
    a = np.frombuffer(t.raw, np.uint8).reshape(rows, nb, bs)
    q5 = q4._unpack_q5(a[:, :, hoff + 4:],
                       np.ascontiguousarray(a[:, :, hoff:hoff + 4]))
    vals = (q5 + 112).astype(np.uint8)
    d = np.ascontiguousarray(a[:, :, :2]).view(np.float16)
    alpha = d.reshape(rows, nb).astype(np.float32)
    if t.gtype == ggml.GGML_TYPE_Q5_0:
        beta = np.zeros_like(alpha)
    else:
        m = np.ascontiguousarray(a[:, :, 2:4]).view(np.float16)
        beta = (m.reshape(rows, nb).astype(np.float32) + 16.0 * alpha)
    return vals, alpha, beta


def _pack_grouped_scales(alpha, beta, rows, nb, nbp, R):
    """(alpha, beta) f32 [rows, nb] -> grouped f16 layout
    [R][nbp/4][16][4][2] + zero-padded blocks, flat."""
    al = np.zeros((rows, nbp), dtype=np.float32)
    be = np.zeros((rows, nbp), dtype=np.float32)
    al[:, :nb] = alpha
    be[:, :nb] = beta
    ab = np.stack([al, be], axis=-1).astype(np.float16)
    return np.ascontiguousarray(
        ab.reshape(R, 16, nbp // 4, 4, 2).transpose(0, 2, 1, 3, 4))


def _repack_q4_torch(t: ggml.GGMLTensor, device: str):
    """q4_0/q4_1 -> MFMA tile layout with the bit work done by torch
    integer ops ON THE GPU: only the compressed bytes cross PCIe and
    the ~8 expansion passes run at HBM rate (the numpy path measured
    0.2 GB/s on a 30B q4_0 load — BASELINE.md)."""
    rows, cols = t.shape_rows_cols
    R, nb = rows // 16, cols // 32
    nbp = (nb + 3) & ~3
    bs = 18 if t.gtype == ggml.GGML_TYPE_Q4_0 else 20
    host = np.frombuffer(t.raw, np.uint8)
    raw = torch.tensor(host, device=device)  # one host copy + H2D
    a = raw.view(rows, nb, bs)
    qs = a[:, :, bs - 16:]
    # weight j sits at nibble j (low) / j+16 (high) of the 16 qs bytes
    n = torch.cat([(qs & 0xF), (qs >> 4)], dim=-1).to(torch.int32)
    n = n.view(rows, nb, 4, 8)  # [rows][block][ws][j]
    shifts = torch.tensor([(j % 2) * 16 + (j // 2) * 4 for j in range(8)],
                          device=device, dtype=torch.int32)
    # disjoint 4-bit fields -> sum == OR
    qword = (n << shifts).sum(dim=-1, dtype=torch.int64).to(torch.int32)
    qp = torch.zeros(rows, nbp, 4, dtype=torch.int32, device=device)
    qp[:, :nb] = qword
    qs2 = (qp.view(R, 16, nbp // 4, 4, 4)
           .permute(0, 2, 4, 1, 3).contiguous())
    data = torch.cat([qs2.reshape(-1),
                      torch.zeros(256, dtype=torch.int32, device=device)])
    alpha = torch.zeros(rows, nbp, dtype=torch.float32, device=device)
    beta = torch.zeros_like(alpha)
    if t.gtype == ggml.GGML_TYPE_Q4_0:
        d = a[:, :, :2].contiguous().view(torch.half)
        alpha[:, :nb] = d.view(rows, nb).float()
    else:
        dm = a[:, :, :4].contiguous().view(torch.half).view(rows, nb, 2)
        alpha[:, :nb] = dm[:, :, 0].float()
        beta[:, :nb] = dm[:, :, 1].float()
    ab = torch.stack([alpha, beta], dim=-1).to(torch.float16)
    sc = (ab.view(R, 16, nbp // 4, 4, 2)
          .permute(0, 2, 1, 3, 4).contiguous().reshape(-1))
    sc = torch.cat([sc, torch.zeros(128, dtype=torch.float16,
                                    device=device)])
    return data, sc, t.gtype


def _pack_bytes_torch(vals, alpha, beta, rows, cols, per16, device):
    """(u8 values [rows, nb, 32], alpha, beta) -> (data, scales, wtype)
    in the W_Q8B/W_Q8B16 tile layout, all torch ops (device-agnostic,
    bit-identical to the numpy packing in repack_mfma)."""
    R, nb = rows // 16, cols // 32
    nbp = (nb + 3) & ~3
    perm = torch.tensor([0, 2, 1, 3, 4, 6, 5, 7], device=device)
    v = vals.reshape(rows, nb, 4, 8)[:, :, :, perm]
    vp = torch.zeros(rows, nbp, 4, 8, dtype=torch.uint8, device=device)
    vp[:, :nb] = v
    shifts = torch.tensor([0, 8, 16, 24], device=device,
                          dtype=torch.int32)
    q32 = (vp.view(rows, nbp, 4, 2, 4).to(torch.int32) << shifts
           ).sum(dim=-1, dtype=torch.int64).to(torch.int32)
    qs2 = (q32.view(R, 16, nbp // 4, 4, 4, 2)     # [R][i][g4][kb][ws][2]
           .permute(0, 2, 4, 1, 3, 5).contiguous())
    data = torch.cat([qs2.reshape(-1),
                      torch.zeros(512, dtype=torch.int32, device=device)])
    if per16:
        al = torch.zeros(rows, nbp, 2, dtype=torch.float32, device=device)
        be = torch.zeros_like(al)
        al[:, :nb] = alpha
        be[:, :nb] = beta
        ab = torch.stack([al, be], dim=-1).to(torch.float16)
        sc = (ab.view(R, 16, nbp // 4, 4, 2, 2)
              .permute(0, 2, 4, 1, 3, 5).contiguous().reshape(-1))
        wt_out = W_Q8B16
    else:
        al = torch.zeros(rows, nbp, dtype=torch.float32, device=device)
        be = torch.zeros_like(al)
        al[:, :nb] = alpha
        be[:, :nb] = beta
        ab = torch.stack([al, be], dim=-1).to(torch.float16)
        sc = (ab.view(R, 16, nbp // 4, 4, 2)
              .permute(0, 2, 1, 3, 4).contiguous().reshape(-1))
        wt_out = ggml.GGML_TYPE_Q8_0              # = W_Q8B
    sc = torch.cat([sc, torch.zeros(128, dtype=torch.float16,
                                    device=device)])
    return data, sc, wt_out


def _repack_byte_torch(t: ggml.GGMLTensor, device: str):
    """q5_0/q5_1/q8_0 -> W_Q8B byte-stream layout with the expansion done
    by torch integer ops ON THE GPU (bit-identical to the numpy path;
    same rationale as _repack_q4_torch: only compressed bytes cross
    PCIe)."""
    rows, cols = t.shape_rows_cols
    nb = cols // 32
    raw = torch.tensor(np.frombuffer(t.raw, np.uint8), device=device)
    if t.gtype == ggml.GGML_TYPE_Q8_0:
        a = raw.view(rows, nb, 34)
        alpha = a[:, :, :2].contiguous().view(torch.half).view(rows, nb
                                                               ).float()
        beta = torch.zeros_like(alpha)
        vals = a[:, :, 2:] ^ 0x80                     # int8 + 128
    else:
        bs = 22 if t.gtype == ggml.GGML_TYPE_Q5_0 else 24
        hoff = 2 if t.gtype == ggml.GGML_TYPE_Q5_0 else 4
        a = raw.view(rows, nb, bs)
        qs = a[:, :, hoff + 4:]
        qh = a[:, :, hoff:hoff + 4].contiguous().view(torch.int32
                                                      ).view(rows, nb)
        # 5th bits: arithmetic >> keeps bit k at position 0 after & 1
        bits = ((qh.unsqueeze(-1) >>
                 torch.arange(32, device=device, dtype=torch.int32)) & 1
                ).to(torch.uint8)
        lo = (qs & 0xF) | (bits[:, :, :16] << 4)
        hi = (qs >> 4) | (bits[:, :, 16:] << 4)
        vals = torch.cat([lo, hi], dim=-1) + 112      # q5 + 112 (<= 143)
        alpha = a[:, :, :2].contiguous().view(torch.half).view(rows, nb
                                                               ).float()
        if t.gtype == ggml.GGML_TYPE_Q5_0:
            beta = torch.zeros_like(alpha)
        else:
            m = a[:, :, 2:4].contiguous().view(torch.half).view(rows, nb)
            beta = m.float() + 16.0 * alpha
    return _pack_bytes_torch(vals, alpha, beta, rows, cols, False, device)


def _unpack_scales_k4_torch(p):
    """torch port of kquants._unpack_scales_k4: [..., 12] u8 ->
    (sc, mn) u8 [..., 8]."""
    sc = torch.empty(p.shape[:-1] + (8,), dtype=torch.uint8,
                     device=p.device)
    mn = torch.empty_like(sc)
    sc[..., 0:4] = p[..., 0:4] & 63
    mn[..., 0:4] = p[..., 4:8] & 63
    sc[..., 4:8] = (p[..., 8:12] & 0xF) | ((p[..., 0:4] >> 6) << 4)
    mn[..., 4:8] = (p[..., 8:12] >> 4) | ((p[..., 4:8] >> 6) << 4)
    return sc, mn


def _repack_kquant_torch(t: ggml.GGMLTensor, device: str):
    """k-quant (q2_K..q6_K) -> W_Q8B/W_Q8B16 byte-stream layout with the
    super-block expansion done by torch integer ops ON THE GPU —
    bit-identical to the numpy _kquant_byte_values + packing path."""
    from ..formats import kquants as KQ
    rows, cols = t.shape_rows_cols
    nsb = cols // KQ.QK_K
    raw = torch.tensor(np.frombuffer(t.raw, np.uint8), device=device)

    def f16col(b, lo, hi):
        return (b[:, :, lo:hi].contiguous().view(torch.half)
                .view(rows, nsb).float())

    if t.gtype == ggml.GGML_TYPE_Q4_K:
        b = raw.view(rows, nsb, KQ.Q4_K_BLOCK_BYTES)
        d, dmin = f16col(b, 0, 2), f16col(b, 2, 4)
        sc, mn = _unpack_scales_k4_torch(b[:, :, 4:16])
        qs = b[:, :, 16:144].reshape(rows, nsb, 4, 32)
        q = torch.empty(rows, nsb, 8, 32, dtype=torch.uint8,
                        device=device)
        q[:, :, 0::2] = qs & 0xF
        q[:, :, 1::2] = qs >> 4
        alpha = (d.unsqueeze(-1) * sc.float()).reshape(rows, nsb * 8)
        beta = (-(dmin.unsqueeze(-1) * mn.float())).reshape(rows, nsb * 8)
        vals = (q + 128).reshape(rows, nsb * 8, 32)
        return _pack_bytes_torch(vals, alpha, beta, rows, cols, False,
                                 device)
    if t.gtype == ggml.GGML_TYPE_Q5_K:
        b = raw.view(rows, nsb, KQ.Q5_K_BLOCK_BYTES)
        d, dmin = f16col(b, 0, 2), f16col(b, 2, 4)
        sc, mn = _unpack_scales_k4_torch(b[:, :, 4:16])
        qh = b[:, :, 16:48]
        ql = b[:, :, 48:176].reshape(rows, nsb, 4, 32)
        q = torch.empty(rows, nsb, 8, 32, dtype=torch.uint8,
                        device=device)
        for j in range(4):
            q[:, :, 2 * j] = (ql[:, :, j] & 0xF) | \
                (((qh >> (2 * j)) & 1) << 4)
            q[:, :, 2 * j + 1] = (ql[:, :, j] >> 4) | \
                (((qh >> (2 * j + 1)) & 1) << 4)
        alpha = (d.unsqueeze(-1) * sc.float()).reshape(rows, nsb * 8)
        beta = (-(dmin.unsqueeze(-1) * mn.float())).reshape(rows, nsb * 8)
        vals = (q + 128).reshape(rows, nsb * 8, 32)
        return _pack_bytes_torch(vals, alpha, beta, rows, cols, False,
                                 device)
    if t.gtype == ggml.GGML_TYPE_Q6_K:
        b = raw.view(rows, nsb, KQ.Q6_K_BLOCK_BYTES)
        ql = b[:, :, 0:128].reshape(rows, nsb, 2, 2, 32)
        qh = b[:, :, 128:192].reshape(rows, nsb, 2, 32)
        sc = (b[:, :, 192:208].contiguous().view(torch.int8)
              .view(rows, nsb, 16).float())
        d = f16col(b, 208, 210)
        q = torch.empty(rows, nsb, 2, 4, 32, dtype=torch.int16,
                        device=device)
        q[:, :, :, 0] = ((ql[:, :, :, 0] & 0xF) |
                         (((qh >> 0) & 3) << 4)).to(torch.int16)
        q[:, :, :, 1] = ((ql[:, :, :, 1] & 0xF) |
                         (((qh >> 2) & 3) << 4)).to(torch.int16)
        q[:, :, :, 2] = ((ql[:, :, :, 0] >> 4) |
                         (((qh >> 4) & 3) << 4)).to(torch.int16)
        q[:, :, :, 3] = ((ql[:, :, :, 1] >> 4) |
                         (((qh >> 6) & 3) << 4)).to(torch.int16)
        vals = (q + 96).to(torch.uint8).reshape(rows, nsb * 8, 32)
        alpha16 = (d.unsqueeze(-1) * sc).reshape(rows, nsb * 8, 2)
        beta16 = torch.zeros_like(alpha16)
        return _pack_bytes_torch(vals, alpha16, beta16, rows, cols, True,
                                 device)
    if t.gtype == ggml.GGML_TYPE_Q2_K:
        b = raw.view(rows, nsb, KQ.Q2_K_BLOCK_BYTES)
        sc = (b[:, :, 0:16] & 0xF).float()
        mn = (b[:, :, 0:16] >> 4).float()
        qs = b[:, :, 16:80].reshape(rows, nsb, 2, 32)
        d, dmin = f16col(b, 80, 82), f16col(b, 82, 84)
        q = torch.empty(rows, nsb, 2, 4, 32, dtype=torch.uint8,
                        device=device)
        for j in range(4):
            q[:, :, :, j] = (qs >> (2 * j)) & 3
        vals = (q + 128).reshape(rows, nsb * 8, 32)
        alpha16 = (d.unsqueeze(-1) * sc).reshape(rows, nsb * 8, 2)
        beta16 = (-(dmin.unsqueeze(-1) * mn)).reshape(rows, nsb * 8, 2)
        return _pack_bytes_torch(vals, alpha16, beta16, rows, cols, True,
                                 device)
    assert t.gtype == ggml.GGML_TYPE_Q3_K, t.gtype
    b = raw.view(rows, nsb, KQ.Q3_K_BLOCK_BYTES)
    hm = b[:, :, 0:32]
    qs = b[:, :, 32:96].reshape(rows, nsb, 2, 32)
    p = b[:, :, 96:108]
    sc = torch.empty(rows, nsb, 16, dtype=torch.int16, device=device)
    a0, a1, tt = p[..., 0:4], p[..., 4:8], p[..., 8:12]
    sc[..., 0:4] = ((a0 & 0xF) | (((tt >> 0) & 3) << 4)).to(torch.int16)
    sc[..., 4:8] = ((a1 & 0xF) | (((tt >> 2) & 3) << 4)).to(torch.int16)
    sc[..., 8:12] = ((a0 >> 4) | (((tt >> 4) & 3) << 4)).to(torch.int16)
    sc[..., 12:16] = ((a1 >> 4) | (((tt >> 6) & 3) << 4)).to(torch.int16)
    sc = (sc - 32).float()
    d = f16col(b, 108, 110)
    q = torch.empty(rows, nsb, 2, 4, 32, dtype=torch.int16, device=device)
    for half in range(2):
        for j in range(4):
            low = ((qs[:, :, half] >> (2 * j)) & 3).to(torch.int16)
            hi = ((hm >> (half * 4 + j)) & 1).to(torch.int16)
            q[:, :, half, j] = low - torch.where(
                hi != 0, torch.zeros_like(hi), torch.full_like(hi, 4))
    vals = (q + 128).to(torch.uint8).reshape(rows, nsb * 8, 32)
    alpha16 = (d.unsqueeze(-1) * sc).reshape(rows, nsb * 8, 2)
    beta16 = torch.zeros_like(alpha16)
    return _pack_bytes_torch(vals, alpha16, beta16, rows, cols, True,
                             device)


def _repack_f16_torch(t: ggml.GGMLTensor, device: str):
    """f16 -> [R][cols/8][16][8] tile layout, transposed on the GPU."""
    rows, cols = t.shape_rows_cols
    R = rows // 16
    w = torch.tensor(np.frombuffer(t.raw, np.int16), device=device)
    tile = (w.view(R, 16, cols // 8, 8).permute(0, 2, 1, 3)
            .contiguous().reshape(-1))
    data = torch.cat([tile, torch.zeros(2048, dtype=torch.int16,
                                        device=device)])
    return data, torch.empty(0), ggml.GGML_TYPE_F16


def repack_mfma(t: ggml.GGMLTensor, device: str):
    """On-disk tensor -> (data, scales, wtype) in the MFMA tile layout.

    q4_0/q4_1: data u32[R][nbp/4][4 ws][16 rows][4 kb] — word ws of a
    block packs its 8 weights at bit positions (j%2)*16 + (j//2)*4 (the
    arrangement a_frag_q4 unpacks into an f16x8 A-fragment), with 4
    consecutive K-blocks grouped per lane for dwordx4 loads and nb padded
    to a multiple of 4 (zero-scaled pad blocks). scales are (alpha, beta)
    f16 pairs in the matching grouped layout. f16 weights stay f16, tiled
    [R][cols/8][16 rows][8]. Every tensor carries one prefetch batch of
    zero tail slack (kernels overread one batch).
    """
    rows, cols = t.shape_rows_cols
    assert rows % 16 == 0 and cols % 32 == 0, (t.name, rows, cols)
    R = rows // 16
    if t.gtype in (ggml.GGML_TYPE_Q4_0, ggml.GGML_TYPE_Q4_1) and \
            device != "cpu":
        return _repack_q4_torch(t, device)
    if t.gtype in (ggml.GGML_TYPE_Q4_0, ggml.GGML_TYPE_Q4_1):
        nb = cols // 32
        nbp = (nb + 3) & ~3  # padded to the dwordx4 load group of 4 blocks
        n = _nibbles(t).astype(np.uint32).reshape(rows, nb, 4, 8)
        qword = np.zeros((rows, nbp, 4), dtype=np.uint32)
        for j in range(8):
            qword[:, :nb] |= n[:, :, :, j] << ((j % 2) * 16 + (j // 2) * 4)
        # wide-load layout: u32[R][nbp/4][4 ws][16 i][4 kb-in-group] — one
        # dwordx4 per lane covers 4 consecutive K-blocks (1 KiB per wave)
        qs2 = np.ascontiguousarray(
            qword.reshape(R, 16, nbp // 4, 4, 4)       # [R][i][g4][kb][ws]
            .transpose(0, 2, 4, 1, 3))                 # [R][g4][ws][i][kb]
        # per-(row, block) (alpha, beta) f16 pair; the kernel computes
        # w = alpha*((1024+n) + csub) + beta with csub = -1032 (q4_0,
        # giving alpha*(n-8)) or -1024 (q4_1, giving alpha*n + beta) in
        # exact packed-f16 arithmetic (kernels.hip a_frag_q4); pad blocks
        # carry alpha=beta=0 and contribute exact zeros.
        bs = 18 if t.gtype == ggml.GGML_TYPE_Q4_0 else 20
        a = np.frombuffer(t.raw, np.uint8).reshape(rows, nb, bs)
        alpha = np.zeros((rows, nbp), dtype=np.float32)
        beta = np.zeros((rows, nbp), dtype=np.float32)
        if t.gtype == ggml.GGML_TYPE_Q4_0:
            d = np.ascontiguousarray(a[:, :, :2]).view(np.float16)
            alpha[:, :nb] = d.reshape(rows, nb).astype(np.float32)
        else:
            dm = np.ascontiguousarray(a[:, :, :4]).view(np.float16)
            dm = dm.reshape(rows, nb, 2).astype(np.float32)
            alpha[:, :nb] = dm[:, :, 0]
            beta[:, :nb] = dm[:, :, 1]
        ab = np.stack([alpha, beta], axis=-1).astype(np.float16)
        scales = np.ascontiguousarray(
            ab.reshape(R, 16, nbp // 4, 4, 2)          # [R][i][g4][kb][2]
            .transpose(0, 2, 1, 3, 4))                 # [R][g4][i][kb][2]
        # one-prefetch-batch tail slack (kernels overread one batch)
        data = torch.from_numpy(
            np.concatenate([qs2.view(np.int32).reshape(-1),
                            np.zeros(256, np.int32)])).to(device)
        sc = torch.from_numpy(
            np.concatenate([scales.reshape(-1),
                            np.zeros(128, np.float16)])).to(device)
        return data, sc, t.gtype
    if t.gtype in _BYTE_GTYPES and device != "cpu":
        return _repack_byte_torch(t, device)
    if (t.gtype in _K32_GTYPES or t.gtype in _K16_GTYPES) and \
            device != "cpu":
        return _repack_kquant_torch(t, device)
    if t.gtype in _BYTE_GTYPES or t.gtype in _K32_GTYPES or \
            t.gtype in _K16_GTYPES:
        nb = cols // 32
        nbp = (nb + 3) & ~3
        per16 = t.gtype in _K16_GTYPES
        if t.gtype in _BYTE_GTYPES:
            vals, alpha, beta = _byte_values(t)
        else:
            vals, alpha, beta = _kquant_byte_values(t)
        # byte order per lane-block 8 weights: [w0,w2,w1,w3, w4,w6,w5,w7]
        # so the kernel's (q & 0x00FF00FF) / (q>>8 & ..) masks yield the
        # (even, odd) f16 pairs of the A fragment (a_frag_q8)
        perm = [0, 2, 1, 3, 4, 6, 5, 7]
        v = vals.reshape(rows, nb, 4, 8)[:, :, :, perm]
        vp = np.zeros((rows, nbp, 4, 8), dtype=np.uint8)
        vp[:, :nb] = v
        # -> u32[R][nbp/4][4 ws][16 i][4 kb][2] with the block's two u32
        # adjacent per lane (one dwordx4 pair covers a 4-block group)
        q32 = vp.reshape(rows, nbp, 4, 2, 4).copy().view("<u4")[..., 0]
        qs2 = np.ascontiguousarray(
            q32.reshape(R, 16, nbp // 4, 4, 4, 2)   # [R][i][g4][kb][ws][2]
            .transpose(0, 2, 4, 1, 3, 5))           # [R][g4][ws][i][kb][2]
        if per16:
            # two (alpha, beta) half-planes per block:
            # [R][g4][2 hp][16 i][4 kb][2 f16]
            al = np.zeros((rows, nbp, 2), dtype=np.float32)
            be = np.zeros((rows, nbp, 2), dtype=np.float32)
            al[:, :nb] = alpha
            be[:, :nb] = beta
            ab = np.stack([al, be], axis=-1).astype(np.float16)
            scales = np.ascontiguousarray(
                ab.reshape(R, 16, nbp // 4, 4, 2, 2)
                .transpose(0, 2, 4, 1, 3, 5))
            wt_out = W_Q8B16
        else:
            scales = _pack_grouped_scales(alpha, beta, rows, nb, nbp, R)
            wt_out = ggml.GGML_TYPE_Q8_0  # = W_Q8B
        data = torch.from_numpy(
            np.concatenate([qs2.view(np.int32).reshape(-1),
                            np.zeros(512, np.int32)])).to(device)
        sc = torch.from_numpy(
            np.concatenate([scales.reshape(-1),
                            np.zeros(128, np.float16)])).to(device)
        return data, sc, wt_out
    if t.gtype == ggml.GGML_TYPE_F16 and device != "cpu":
        return _repack_f16_torch(t, device)
    if t.gtype == ggml.GGML_TYPE_F16:
        w = np.frombuffer(t.raw, np.float16).reshape(rows, cols)
        tile = np.ascontiguousarray(
            w.reshape(R, 16, cols // 8, 8).transpose(0, 2, 1, 3))
        data = torch.from_numpy(
            np.concatenate([tile.view(np.int16).reshape(-1),
                            np.zeros(2048, np.int16)])).to(device)
        return data, torch.empty(0), ggml.GGML_TYPE_F16
    # f32: legacy scalar path, plain [rows, cols]
    a = np.frombuffer(t.raw, np.float32).reshape(rows, cols)
    return (torch.from_numpy(a.copy()).to(device), torch.empty(0),
            ggml.GGML_TYPE_F32)


def _repack_q4(t: ggml.GGMLTensor):
    rows, cols = t.shape_rows_cols
    nb = cols // 32
    if t.gtype == ggml.GGML_TYPE_Q4_0:
        a = np.frombuffer(t.raw, np.uint8).reshape(rows, nb, 18)
        scales = np.ascontiguousarray(a[:, :, :2]).view(np.float16)
        scales = scales.reshape(rows, nb)
        qs = np.ascontiguousarray(a[:, :, 2:]).reshape(rows, nb * 16)
    else:  # q4_1: 20-byte blocks, (d, m) f16 pairs
        a = np.frombuffer(t.raw, np.uint8).reshape(rows, nb, 20)
        scales = np.ascontiguousarray(a[:, :, :4]).view(np.float16)
        scales = scales.reshape(rows, nb * 2)
        qs = np.ascontiguousarray(a[:, :, 4:]).reshape(rows, nb * 16)
    return scales, qs


def _upload_mat(t: ggml.GGMLTensor, device: str):
    """-> (data, scales, wtype) device tensors in the kernel layout."""
    rows, cols = t.shape_rows_cols
    if t.gtype in (ggml.GGML_TYPE_Q4_0, ggml.GGML_TYPE_Q4_1):
        scales, qs = _repack_q4(t)
        d = torch.from_numpy(qs).to(device)
        s = torch.from_numpy(scales.copy()).to(device)
        return d, s, t.gtype
    if t.gtype in _BYTE_GTYPES or t.gtype in _K32_GTYPES or \
            t.gtype in _K16_GTYPES:
        # embedding gather for q5/q8/k-quant tables: dequantize once to
        # f16 (the gather kernel has no byte path; 2 B/weight)
        a = t.to_f32().astype(np.float16)
        return (torch.from_numpy(a).to(device), torch.empty(0),
                ggml.GGML_TYPE_F16)
    if t.gtype == ggml.GGML_TYPE_F16:
        a = np.frombuffer(t.raw, np.float16).reshape(rows, cols)
        return torch.from_numpy(a.copy()).to(device), torch.empty(0), t.gtype
    a = np.frombuffer(t.raw, np.float32).reshape(rows, cols)
    return torch.from_numpy(a.copy()).to(device), torch.empty(0), t.gtype


def detile_mfma(mat, rows: int, cols: int) -> torch.Tensor:
    """Reference inverse of repack_mfma: tiled weights -> plain f16
    [rows, cols]. Test/debug utility — documents the tile layouts the
    kernels stream (the production prefill path reads tiles directly)."""
    data, sc, wt = mat
    if wt == ggml.GGML_TYPE_F16:
        t = data[:rows * cols].view(torch.half)
        return (t.view(rows // 16, cols // 8, 16, 8)
                .permute(0, 2, 1, 3).reshape(rows, cols))
    R, nb = rows // 16, cols // 32
    nbp = (nb + 3) & ~3
    ab = sc[:R * nbp * 32].view(R, nbp // 4, 16, 4, 2).float()
    alpha = ab[..., 0][:, :, None, :, :, None]  # [R,g4,1,i,kb,1]
    beta = ab[..., 1][:, :, None, :, :, None]
    if wt in (ggml.GGML_TYPE_Q4_0, ggml.GGML_TYPE_Q4_1):
        qw = data[:R * nbp * 64].view(R, nbp // 4, 4, 16, 4)
        shifts = torch.tensor(
            [(j % 2) * 16 + (j // 2) * 4 for j in range(8)],
            device=data.device, dtype=torch.int32)
        n = ((qw.unsqueeze(-1) >> shifts) & 0xF).float()
        if wt == ggml.GGML_TYPE_Q4_0:
            w = alpha * (n - 8.0)
        else:
            w = alpha * n + beta
        # [R,g4,ws,i,kb,8] -> [R,i,g4,kb,ws,8] -> [rows, nbp*32]
        w = w.permute(0, 3, 1, 4, 2, 5).reshape(rows, nbp * 32)
        return w[:, :cols].to(torch.half)
    # W_Q8B byte stream: [R][g4][ws][i][kb][2 u32], u32 bytes
    # [w0,w2,w1,w3] so weight jj sits at byte [0,2,1,3][jj]
    qw = data[:R * nbp * 128].view(R, nbp // 4, 4, 16, 4, 2)
    bshift = torch.tensor([0, 16, 8, 24], device=data.device,
                          dtype=torch.int32)
    u = ((qw.unsqueeze(-1) >> bshift) & 0xFF).float()
    w = alpha.unsqueeze(-1) * (u - 128.0) + beta.unsqueeze(-1)
    # [R,g4,ws,i,kb,2,4] -> [R,i,g4,kb,ws,2,4] -> [rows, nbp*32]
    w = w.permute(0, 3, 1, 4, 2, 5, 6).reshape(rows, nbp * 32)
    return w[:, :cols].to(torch.half)


class HIPSliceEngine:
    """Production engine: CDNA4 kernels, weights resident in HBM3E."""

    def __init__(self, hp: ggml.Hparams, n_layers: int, first_layer: int,
                 n_ctx: int = 2048, max_batch: int = 16,
                 max_prefill: Optional[int] = None):
        from .. import ops
        core = ops.core()
        self.hp = hp
        self.first_layer = first_layer
        self.n_layers = n_layers
        self.n_ctx = n_ctx
        self.max_batch = max_batch
        # prefill token cap per forward call (side channels scale with it;
        # larger prompts tile host-side in max_prefill chunks)
        if max_prefill is None:
            max_prefill = max(64, min(n_ctx, 2048))
        self._eng = core.SliceEngine(
            n_embd=hp.n_embd, n_head=hp.n_head, n_layers=n_layers,
            n_ff=hp.n_ff, n_ctx=n_ctx, max_batch=max_batch,
            eps=RMS_EPS, rope_base=ROPE_BASE, max_prefill=max_prefill,
            n_head_kv=hp.kv_heads if hp.is_gqa else 0)
        self.device = "cuda"
        self.has_extra = False
        # weight-tensor references for clone_shared (filled by .random()
        # or load_layers/attach_extra; the C++ engine holds references to
        # the same device tensors, so clones share HBM)
        self._layers_cache = None
        self._extra_cache = None

    @classmethod
    def from_ggml(cls, f: ggml.GGMLFile, n_ctx: int = 2048,
                  max_batch: int = 16) -> "HIPSliceEngine":
        hp = f.hparams
        first = hp.first_layer if hp.first_layer is not None else 0
        eng = cls(hp, hp.n_layer, first, n_ctx, max_batch)
        eng.load_layers(f)
        return eng

    @classmethod
    def random(cls, hp: ggml.Hparams, n_layers: int, first_layer: int = 0,
               n_ctx: int = 2048, max_batch: int = 16, seed: int = 0,
               with_extra: bool = True,
               max_prefill: Optional[int] = None) -> "HIPSliceEngine":
        """Random-init engine straight on the GPU (synthetic benchmarking).

        Generates weights directly in the repacked kernel layout — identical
        compute and HBM traffic to a real checkpoint of this architecture,
        without materializing a multi-GB GGML file on disk.
        """
        eng = cls(hp, n_layers, first_layer, n_ctx, max_batch,
                  max_prefill=max_prefill)
        g = torch.Generator(device="cuda")
        g.manual_seed(seed)
        E, F, V = hp.n_embd, hp.n_ff, hp.n_vocab
        wt = ggml._FTYPE_TO_GGML[hp.ftype]

        def mat(rows: int, cols: int):
            # random weights directly in the MFMA tile layouts — any random
            # bit pattern is a valid q4 nibble word, so this is byte-for-
            # byte the same compute/HBM traffic as a real checkpoint
            if wt in _K16_GTYPES:
                # byte stream + per-16 (alpha, beta) planes (W_Q8B16)
                R, nb = rows // 16, cols // 32
                nbp = (nb + 3) & ~3
                data = torch.randint(-2**31, 2**31 - 1,
                                     (R * nbp * 128 + 512,),
                                     dtype=torch.int32, device="cuda",
                                     generator=g)
                alpha = ((torch.rand(rows, nbp, 2, device="cuda",
                                     generator=g) * 0.5 + 0.75) * 0.003)
                alpha[:, nb:] = 0.0
                beta = torch.zeros_like(alpha)
                ab = torch.stack([alpha, beta], dim=-1).to(torch.float16)
                ab = (ab.reshape(R, 16, nbp // 4, 4, 2, 2)
                      .permute(0, 2, 4, 1, 3, 5).contiguous().reshape(-1))
                ab = torch.cat([ab, torch.zeros(128, dtype=torch.float16,
                                                device="cuda")])
                return data, ab, W_Q8B16
            if wt in _BYTE_GTYPES or wt in _K32_GTYPES:
                R, nb = rows // 16, cols // 32
                nbp = (nb + 3) & ~3
                data = torch.randint(-2**31, 2**31 - 1,
                                     (R * nbp * 128 + 512,),
                                     dtype=torch.int32, device="cuda",
                                     generator=g)
                alpha = ((torch.rand(rows, nbp, device="cuda",
                                     generator=g) * 0.5 + 0.75) * 0.003)
                alpha[:, nb:] = 0.0
                beta = (alpha * 0.1 if wt == ggml.GGML_TYPE_Q5_1
                        else torch.zeros_like(alpha))
                ab = torch.stack([alpha, beta], dim=-1).to(torch.float16)
                ab = (ab.reshape(R, 16, nbp // 4, 4, 2)
                      .permute(0, 2, 1, 3, 4).contiguous().reshape(-1))
                ab = torch.cat([ab, torch.zeros(128, dtype=torch.float16,
                                                device="cuda")])
                return data, ab, ggml.GGML_TYPE_Q8_0
            if wt in (ggml.GGML_TYPE_Q4_0, ggml.GGML_TYPE_Q4_1):
                R, nb = rows // 16, cols // 32
                nbp = (nb + 3) & ~3
                # any random bits are valid nibble words (the dequant masks
                # them into finite f16 values), so the data stream needs no
                # pad handling; pad scale blocks are (0, 0) => exact zeros
                data = torch.randint(-2**31, 2**31 - 1,
                                     (R * nbp * 64 + 256,),
                                     dtype=torch.int32, device="cuda",
                                     generator=g)
                alpha = ((torch.rand(rows, nbp, device="cuda",
                                     generator=g) * 0.5 + 0.75) * 0.003)
                alpha[:, nb:] = 0.0
                beta = (torch.zeros_like(alpha) if wt == ggml.GGML_TYPE_Q4_0
                        else alpha * 0.1)
                ab = torch.stack([alpha, beta], dim=-1).to(torch.float16)
                ab = (ab.reshape(R, 16, nbp // 4, 4, 2)
                      .permute(0, 2, 1, 3, 4).contiguous().reshape(-1))
                ab = torch.cat([ab, torch.zeros(128, dtype=torch.float16,
                                                device="cuda")])
                return data, ab, wt
            if wt == ggml.GGML_TYPE_F16:
                data = (torch.randn(rows * cols + 2048, device="cuda",
                                    generator=g,
                                    dtype=torch.float32) * 0.02)
                return (data.to(torch.float16).view(torch.int16),
                        torch.empty(0), wt)
            data = (torch.randn(rows, cols, device="cuda", generator=g,
                                dtype=torch.float32) * 0.02)
            return data.contiguous(), torch.empty(0), wt

        def norm_w(n: int):
            return (1.0 + torch.randn(n, device="cuda", generator=g) *
                    0.01).contiguous()

        Ekv = hp.n_embd_kv  # K/V projection rows (GQA: Hkv*D < E)
        shapes = [(E, E), (Ekv, E), (Ekv, E), (E, E), (F, E), (E, F),
                  (F, E)]
        eng._layers_cache = []
        for li in range(n_layers):
            mats = [mat(r, c) for r, c in shapes]
            an, fn = norm_w(E), norm_w(E)
            eng._layers_cache.append((an, fn, mats))
            eng._eng.set_layer(li, an, fn, mats)
        eng._extra_cache = None
        if with_extra:
            # the embedding table uses the legacy SoA layout (gather kernel)
            if wt in _BYTE_GTYPES:
                tok_d = (torch.randn(V, E, device="cuda", generator=g,
                                     dtype=torch.float32) * 0.02).to(
                    torch.float16)
                tok_s, tok_t = torch.empty(0), ggml.GGML_TYPE_F16
            elif wt in (ggml.GGML_TYPE_Q4_0, ggml.GGML_TYPE_Q4_1):
                nb = E // 32
                per = 2 if wt == ggml.GGML_TYPE_Q4_1 else 1
                tok_d = torch.randint(0, 256, (V, nb * 16),
                                      dtype=torch.uint8, device="cuda",
                                      generator=g)
                tok_s = ((torch.rand(V, nb * per, device="cuda",
                                     generator=g) * 0.5 + 0.75) *
                         0.003).to(torch.float16)
                tok_t = wt
            elif wt == ggml.GGML_TYPE_F16:
                tok_d = (torch.randn(V, E, device="cuda", generator=g) *
                         0.02).to(torch.float16)
                tok_s, tok_t = torch.empty(0), wt
            else:
                tok_d = (torch.randn(V, E, device="cuda", generator=g) *
                         0.02)
                tok_s, tok_t = torch.empty(0), wt
            out_d, out_s, out_t = mat(V, E)
            fin = norm_w(E)
            eng._extra_cache = (tok_d, tok_s, tok_t, fin, out_d, out_s,
                                out_t, V)
            eng._eng.set_extra(tok_d, tok_s, tok_t, fin, out_d, out_s,
                               out_t, V)
            eng.has_extra = True
        return eng

    def clone_shared(self) -> "HIPSliceEngine":
        """A second engine over the SAME weight tensors (HBM shared) with
        its own KV cache and side-channel buffers — lets independent
        micro-batches run concurrently on separate HIP streams without
        duplicating the model."""
        assert getattr(self, "_layers_cache", None) is not None, \
            "clone_shared requires an engine built by .random()/.from_ggml"
        twin = HIPSliceEngine(self.hp, self.n_layers, self.first_layer,
                              self.n_ctx, self.max_batch,
                              max_prefill=int(self._eng.max_prefill))
        for li, (an, fn, mats) in enumerate(self._layers_cache):
            twin._eng.set_layer(li, an, fn, mats)
        twin._layers_cache = self._layers_cache  # enables further clones
        if self._extra_cache is not None:
            twin._extra_cache = self._extra_cache
            twin._eng.set_extra(*self._extra_cache)
            twin.has_extra = True
        return twin

    def load_layers(self, f: ggml.GGMLFile) -> None:
        tm = f.tensor_map()
        self._layers_cache = []
        for li in range(self.n_layers):
            gi = li + self.first_layer
            pre = f"layers.{gi}."
            attn_norm = torch.from_numpy(
                tm[pre + "attention_norm.weight"].to_f32()).to(self.device)
            ffn_norm = torch.from_numpy(
                tm[pre + "ffn_norm.weight"].to_f32()).to(self.device)
            mats = []
            for nm in ("attention.wq.weight", "attention.wk.weight",
                       "attention.wv.weight", "attention.wo.weight",
                       "feed_forward.w1.weight", "feed_forward.w2.weight",
                       "feed_forward.w3.weight"):
                mats.append(repack_mfma(tm[pre + nm], self.device))
            self._layers_cache.append((attn_norm, ffn_norm, mats))
            self._eng.set_layer(li, attn_norm, ffn_norm, mats)

    def attach_extra(self, f: ggml.GGMLFile) -> None:
        tm = f.tensor_map()
        # embedding table: legacy SoA layout (row-gather kernel);
        # lm_head: MFMA tiles (it is a GEMV like any other)
        tok_d, tok_s, tok_t = _upload_mat(tm["tok_embeddings.weight"],
                                          self.device)
        out_d, out_s, out_t = repack_mfma(tm["output.weight"], self.device)
        norm = torch.from_numpy(tm["norm.weight"].to_f32()).to(self.device)
        self._extra_cache = (tok_d, tok_s, tok_t, norm, out_d, out_s,
                             out_t, self.hp.n_vocab)
        self._eng.set_extra(*self._extra_cache)
        self.has_extra = True

    def forward(self, x: torch.Tensor, pos: torch.Tensor,
                seq: torch.Tensor, decode: bool = False) -> torch.Tensor:
        """decode=True asserts every token is a distinct sequence
        (batched decode) — enables the qkv-slab + fused-attention path.

        T > 64 (prefill) runs the native large-M path: hand-written
        XCD-grouped dequant-GEMM kernels over all T tokens, the layer
        loop sequenced in C++ (replaced the round-1 rocBLAS-over-
        detiled-f16 path — no f16 weight copy, mixed multi-span
        admission streams handled natively)."""
        T = x.shape[0]
        # the large-M path serves decode too (wide batched decode pays
        # the weight stream + dequant once for the whole batch)
        cap = self._eng.max_prefill if self._mfma_path() \
            else self._eng.max_tokens
        if T <= cap:
            return self._eng.forward(x, pos, seq, decode=decode)
        # token-tile larger inputs; KV order is preserved because tile
        # i's cache rows are written before tile i+1 attends.
        outs = []
        for t0 in range(0, T, cap):
            t1 = min(T, t0 + cap)
            outs.append(self._eng.forward(
                x[t0:t1].contiguous(), pos[t0:t1].contiguous(),
                seq[t0:t1].contiguous(), decode=decode))
        return torch.cat(outs, dim=0)

    def _mfma_path(self) -> bool:
        if not self._layers_cache:
            return False
        return all(m[2] != ggml.GGML_TYPE_F32
                   for (_, _, mats) in self._layers_cache for m in mats)

    def embed(self, tokens: torch.Tensor) -> torch.Tensor:
        return self._eng.embed(tokens.to(self.device, torch.int32))

    def logits(self, x: torch.Tensor, all_logits: bool = False) -> torch.Tensor:
        if not all_logits:
            return self._eng.logits(x, False)
        mt = self._eng.max_tokens
        if x.shape[0] <= mt:
            return self._eng.logits(x, True)
        return torch.cat([self._eng.logits(x[t0:t0 + mt].contiguous(), True)
                          for t0 in range(0, x.shape[0], mt)], dim=0)

    def argmax(self, lg: torch.Tensor) -> torch.Tensor:
        return self._eng.argmax(lg)


class TorchSliceEngine:
    """CPU twin with the identical stateless interface (fp32 torch math).

    Used for CPU tests, the gloo pipeline path, and machines without a GPU.
    Never selected automatically when CUDA is available — the HIP engine is
    the only GPU path.
    """

    def __init__(self, hp: ggml.Hparams, weights: Dict[str, torch.Tensor],
                 n_layers: int, first_layer: int, n_ctx: int = 512,
                 max_batch: int = 4, device: str = "cpu"):
        self.hp = hp
        self.w = weights
        self.first_layer = first_layer
        self.n_layers = n_layers
        self.n_ctx = n_ctx
        self.max_batch = max_batch
        self.device = device
        d = hp.head_dim
        hkv = hp.kv_heads
        self.k_cache = torch.zeros(n_layers, max_batch, n_ctx, hkv, d)
        self.v_cache = torch.zeros(n_layers, max_batch, n_ctx, hkv, d)
        self.has_extra = all(
            k in weights for k in
            ("tok_embeddings.weight", "norm.weight", "output.weight"))

    @classmethod
    def from_ggml(cls, f: ggml.GGMLFile, n_ctx: int = 512,
                  max_batch: int = 4) -> "TorchSliceEngine":
        from ..models.llama import weights_from_ggml
        hp = f.hparams
        first = hp.first_layer if hp.first_layer is not None else 0
        return cls(hp, weights_from_ggml(f), hp.n_layer, first, n_ctx,
                   max_batch)

    def attach_extra(self, f: ggml.GGMLFile) -> None:
        from ..models.llama import weights_from_ggml
        self.w.update(weights_from_ggml(f))
        self.has_extra = True

    def forward(self, x: torch.Tensor, pos: torch.Tensor,
                seq: torch.Tensor, decode: bool = False) -> torch.Tensor:
        hp = self.hp
        T, E = x.shape
        h, d = hp.n_head, hp.head_dim
        hkv = hp.kv_heads
        kv_map = torch.arange(h) // (h // hkv)  # GQA head sharing
        pos_l = pos.tolist()
        seq_l = seq.tolist()
        for li in range(self.n_layers):
            gi = li + self.first_layer
            pre = f"layers.{gi}."
            a = rms_norm(x) * self.w[pre + "attention_norm.weight"]
            q = a @ self.w[pre + "attention.wq.weight"].T
            k = a @ self.w[pre + "attention.wk.weight"].T
            v = a @ self.w[pre + "attention.wv.weight"].T
            o = torch.empty_like(x)
            for t in range(T):  # per-row positions (prefill/decode unified)
                p, b = pos_l[t], seq_l[t]
                qt = rope_interleaved(q[t:t + 1].view(1, h, d), p)[0]
                kt = rope_interleaved(k[t:t + 1].view(1, hkv, d), p)[0]
                self.k_cache[li, b, p] = kt
                self.v_cache[li, b, p] = v[t].view(hkv, d)
                keys = self.k_cache[li, b, :p + 1][:, kv_map]
                vals = self.v_cache[li, b, :p + 1][:, kv_map]
                att = torch.einsum("hd,jhd->hj", qt, keys) / math.sqrt(d)
                prob = torch.softmax(att, dim=-1)
                o[t] = torch.einsum("hj,jhd->hd", prob, vals).reshape(E)
            x = x + o @ self.w[pre + "attention.wo.weight"].T
            f = rms_norm(x) * self.w[pre + "ffn_norm.weight"]
            g = torch.nn.functional.silu(
                f @ self.w[pre + "feed_forward.w1.weight"].T)
            u = f @ self.w[pre + "feed_forward.w3.weight"].T
            x = x + (g * u) @ self.w[pre + "feed_forward.w2.weight"].T
        return x

    def embed(self, tokens: torch.Tensor) -> torch.Tensor:
        idx = tokens.to(torch.long)
        return self.w["tok_embeddings.weight"][idx]

    def logits(self, x: torch.Tensor, all_logits: bool = False) -> torch.Tensor:
        y = rms_norm(x) * self.w["norm.weight"]
        lg = y @ self.w["output.weight"].T
        return lg if all_logits else lg[-1:]

    def argmax(self, lg: torch.Tensor) -> torch.Tensor:
        return lg.argmax(dim=-1).to(torch.int32)


def engine_for_slice(f: ggml.GGMLFile, n_ctx: int, max_batch: int,
                     device: Optional[str] = None):
    """Pick the engine for this machine: HIP when a GPU is present."""
    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cuda":
        return HIPSliceEngine.from_ggml(f, n_ctx=n_ctx, max_batch=max_batch)
    return TorchSliceEngine.from_ggml(f, n_ctx=n_ctx, max_batch=max_batch)
