"""GGJT-v3 model / slice file reader-writer.

The on-disk formats this framework is byte-compatible with:

* **model files** — standard GGJT v3: magic ``'ggjt'`` + version 3, then a
  **7-field** hparams block ``n_vocab, n_embd, n_mult, n_head, n_layer,
  n_rot, ftype`` (reference reads it at slice_model.cpp:167-175), then the
  vocab (``len:u32 ‖ bytes ‖ score:f32`` × n_vocab), then 32-byte-aligned
  tensor records.
* **slice files / extra_layers.bin** — the reference's extended format: an
  extra ``first_layer:u32`` inserted between ``n_rot`` and ``ftype``
  (**8 fields**; written at slice_model.cpp:253-263, read at
  tensor_processor.cpp:179-188).  ``extra_layers.bin`` has ``n_layer=0`` and
  ``first_layer=0xFFFFFFFF`` (slice_model.cpp:380-383).
* tensor record: ``n_dims:u32 ‖ name_len:u32 ‖ type:u32 ‖ ne[n_dims]:u32 ‖
  name`` then zero padding to the next 32-byte file offset, then raw data.
  ``ne[0]`` is the contiguous (row/input) dimension.
* slice tensor names keep their ORIGINAL layer indices; loaders re-base
  with ``first_layer`` (tensor_processor.cpp:1340).

Nothing is copied from the reference — this is a clean-room implementation
of the byte layout described above.
"""
from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import BinaryIO, Dict, List, Optional, Tuple

import numpy as np

from . import kquants, q4

GGJT_MAGIC = 0x67676A74  # bytes 'tjgg' little-endian == "ggjt"
GGJT_VERSION = 3
# This framework's GQA extension: the GGJT v3 header cannot express
# n_head_kv (llama.cpp of the reference's era passed it on the command
# line, `-gqa 8`). Version 4 inserts ``n_head_kv:u32`` directly after
# ``n_head``; it is written ONLY when n_head_kv != n_head, so every MHA
# file stays byte-identical to the reference formats.
GGJT_VERSION_GQA = 4

EXTRA_LAYERS_FIRST_LAYER = 0xFFFFFFFF

# ggml tensor dtypes (on-disk ids)
GGML_TYPE_F32 = 0
GGML_TYPE_F16 = 1
GGML_TYPE_Q4_0 = 2
GGML_TYPE_Q4_1 = 3
GGML_TYPE_Q5_0 = 6   # ids 4/5 were the removed q4_2/q4_3
GGML_TYPE_Q5_1 = 7
GGML_TYPE_Q8_0 = 8
# k-quant super-block types (256-weight blocks; ids of the reference era)
GGML_TYPE_Q2_K = 10
GGML_TYPE_Q3_K = 11
GGML_TYPE_Q4_K = 12
GGML_TYPE_Q5_K = 13
GGML_TYPE_Q6_K = 14

# model-level ftype (llama_ftype ids; 4 = Q4_1_SOME_F16, 5/6 removed)
FTYPE_ALL_F32 = 0
FTYPE_MOSTLY_F16 = 1
FTYPE_MOSTLY_Q4_0 = 2
FTYPE_MOSTLY_Q4_1 = 3
FTYPE_MOSTLY_Q8_0 = 7
FTYPE_MOSTLY_Q5_0 = 8
FTYPE_MOSTLY_Q5_1 = 9
# k-quant ftypes (recognized by the reference's loader era,
# tensor_processor.cpp:846-855); the S/M/L variants differ only in which
# tensors upstream mixes — we map each to its base k-quant type
FTYPE_MOSTLY_Q2_K = 10
FTYPE_MOSTLY_Q3_K_S = 11
FTYPE_MOSTLY_Q3_K_M = 12
FTYPE_MOSTLY_Q3_K_L = 13
FTYPE_MOSTLY_Q4_K_S = 14
FTYPE_MOSTLY_Q4_K_M = 15
FTYPE_MOSTLY_Q5_K_S = 16
FTYPE_MOSTLY_Q5_K_M = 17
FTYPE_MOSTLY_Q6_K = 18

_FTYPE_TO_GGML = {
    FTYPE_ALL_F32: GGML_TYPE_F32,
    FTYPE_MOSTLY_F16: GGML_TYPE_F16,
    FTYPE_MOSTLY_Q4_0: GGML_TYPE_Q4_0,
    FTYPE_MOSTLY_Q4_1: GGML_TYPE_Q4_1,
    FTYPE_MOSTLY_Q8_0: GGML_TYPE_Q8_0,
    FTYPE_MOSTLY_Q5_0: GGML_TYPE_Q5_0,
    FTYPE_MOSTLY_Q5_1: GGML_TYPE_Q5_1,
    FTYPE_MOSTLY_Q2_K: GGML_TYPE_Q2_K,
    FTYPE_MOSTLY_Q3_K_S: GGML_TYPE_Q3_K,
    FTYPE_MOSTLY_Q3_K_M: GGML_TYPE_Q3_K,
    FTYPE_MOSTLY_Q3_K_L: GGML_TYPE_Q3_K,
    FTYPE_MOSTLY_Q4_K_S: GGML_TYPE_Q4_K,
    FTYPE_MOSTLY_Q4_K_M: GGML_TYPE_Q4_K,
    FTYPE_MOSTLY_Q5_K_S: GGML_TYPE_Q5_K,
    FTYPE_MOSTLY_Q5_K_M: GGML_TYPE_Q5_K,
    FTYPE_MOSTLY_Q6_K: GGML_TYPE_Q6_K,
}

TYPE_NAMES = {
    GGML_TYPE_F32: "f32",
    GGML_TYPE_F16: "f16",
    GGML_TYPE_Q4_0: "q4_0",
    GGML_TYPE_Q4_1: "q4_1",
    GGML_TYPE_Q5_0: "q5_0",
    GGML_TYPE_Q5_1: "q5_1",
    GGML_TYPE_Q8_0: "q8_0",
    GGML_TYPE_Q2_K: "q2_K",
    GGML_TYPE_Q3_K: "q3_K",
    GGML_TYPE_Q4_K: "q4_K",
    GGML_TYPE_Q5_K: "q5_K",
    GGML_TYPE_Q6_K: "q6_K",
}

# block codecs for the quantized types: (block bytes, quantize, dequantize)
_BLOCK_CODECS = {
    GGML_TYPE_Q4_0: (q4.Q4_0_BLOCK_BYTES, q4.quantize_q4_0,
                     q4.dequantize_q4_0),
    GGML_TYPE_Q4_1: (q4.Q4_1_BLOCK_BYTES, q4.quantize_q4_1,
                     q4.dequantize_q4_1),
    GGML_TYPE_Q5_0: (q4.Q5_0_BLOCK_BYTES, q4.quantize_q5_0,
                     q4.dequantize_q5_0),
    GGML_TYPE_Q5_1: (q4.Q5_1_BLOCK_BYTES, q4.quantize_q5_1,
                     q4.dequantize_q5_1),
    GGML_TYPE_Q8_0: (q4.Q8_0_BLOCK_BYTES, q4.quantize_q8_0,
                     q4.dequantize_q8_0),
}

# k-quant codecs use 256-weight super-blocks
_KBLOCK_CODECS = {
    GGML_TYPE_Q2_K: kquants.CODECS["q2_K"],
    GGML_TYPE_Q3_K: kquants.CODECS["q3_K"],
    GGML_TYPE_Q4_K: kquants.CODECS["q4_K"],
    GGML_TYPE_Q5_K: kquants.CODECS["q5_K"],
    GGML_TYPE_Q6_K: kquants.CODECS["q6_K"],
}


def tensor_nbytes(gtype: int, ne: Tuple[int, ...]) -> int:
    """Byte size of a tensor's data given its ggml type and dims."""
    n0 = ne[0]
    rows = 1
    for d in ne[1:]:
        rows *= d
    if gtype == GGML_TYPE_F32:
        return 4 * n0 * rows
    if gtype == GGML_TYPE_F16:
        return 2 * n0 * rows
    if gtype in _BLOCK_CODECS:
        if n0 % q4.QK4:
            raise ValueError(
                f"{TYPE_NAMES[gtype]} row length {n0} not a multiple of 32")
        return (n0 // q4.QK4) * _BLOCK_CODECS[gtype][0] * rows
    if gtype in _KBLOCK_CODECS:
        if n0 % kquants.QK_K:
            raise ValueError(
                f"{TYPE_NAMES[gtype]} row length {n0} not a multiple "
                f"of {kquants.QK_K}")
        return (n0 // kquants.QK_K) * _KBLOCK_CODECS[gtype][0] * rows
    raise ValueError(f"unsupported ggml type {gtype}")


@dataclass
class Hparams:
    n_vocab: int
    n_embd: int
    n_mult: int
    n_head: int
    n_layer: int
    n_rot: int
    ftype: int
    first_layer: Optional[int] = None  # present only in the extended format
    n_head_kv: Optional[int] = None    # GQA (v4 header); None == n_head

    @property
    def n_ff(self) -> int:
        """FFN width, the reference's formula (tensor_processor.cpp:1250)."""
        return ((2 * (4 * self.n_embd) // 3 + self.n_mult - 1)
                // self.n_mult) * self.n_mult

    @property
    def head_dim(self) -> int:
        return self.n_embd // self.n_head

    @property
    def kv_heads(self) -> int:
        return self.n_head if self.n_head_kv is None else self.n_head_kv

    @property
    def n_embd_kv(self) -> int:
        """Width of the K/V projections (= n_embd for MHA)."""
        return self.kv_heads * self.head_dim

    @property
    def is_gqa(self) -> bool:
        return self.kv_heads != self.n_head


@dataclass
class GGMLTensor:
    name: str
    ne: Tuple[int, ...]          # ne[0] = contiguous/input dim
    gtype: int
    raw: bytes                   # on-disk bytes (or a zero-copy
                                 # memoryview into an mmap'd file)

    @property
    def nbytes(self) -> int:
        return tensor_nbytes(self.gtype, self.ne)

    @property
    def shape_rows_cols(self) -> Tuple[int, int]:
        """(rows, cols) with cols = ne[0] (contiguous)."""
        rows = 1
        for d in self.ne[1:]:
            rows *= d
        return rows, self.ne[0]

    def to_f32(self) -> np.ndarray:
        """Dequantize/convert to a float32 array of shape [rows, ne0]
        (or [ne0] for 1-D tensors)."""
        rows, cols = self.shape_rows_cols
        if self.gtype == GGML_TYPE_F32:
            a = np.frombuffer(self.raw, dtype=np.float32).astype(np.float32)
        elif self.gtype == GGML_TYPE_F16:
            a = np.frombuffer(self.raw, dtype=np.float16).astype(np.float32)
        elif self.gtype in _BLOCK_CODECS:
            u = np.frombuffer(self.raw, dtype=np.uint8).reshape(rows, -1)
            a = _BLOCK_CODECS[self.gtype][2](u, cols)
        elif self.gtype in _KBLOCK_CODECS:
            u = np.frombuffer(self.raw, dtype=np.uint8).reshape(rows, -1)
            a = _KBLOCK_CODECS[self.gtype][2](u, cols)
        else:
            raise ValueError(f"unsupported ggml type {self.gtype}")
        if len(self.ne) == 1:
            return a.reshape(cols)
        return a.reshape(rows, cols)

    @classmethod
    def from_f32(cls, name: str, a: np.ndarray, gtype: int) -> "GGMLTensor":
        """Quantize/convert a float array into a tensor record."""
        a = np.asarray(a, dtype=np.float32)
        if a.ndim == 1:
            ne: Tuple[int, ...] = (a.shape[0],)
        elif a.ndim == 2:
            ne = (a.shape[1], a.shape[0])  # ne[0] = cols (contiguous)
        else:
            raise ValueError("only 1-D/2-D tensors are supported")
        if gtype == GGML_TYPE_F32:
            raw = a.astype(np.float32).tobytes()
        elif gtype == GGML_TYPE_F16:
            raw = a.astype(np.float16).tobytes()
        elif gtype in _BLOCK_CODECS:
            raw = _BLOCK_CODECS[gtype][1](a).tobytes()
        elif gtype in _KBLOCK_CODECS:
            raw = _KBLOCK_CODECS[gtype][1](a).tobytes()
        else:
            raise ValueError(f"unsupported ggml type {gtype}")
        return cls(name=name, ne=ne, gtype=gtype, raw=raw)


@dataclass
class GGMLFile:
    hparams: Hparams
    vocab: List[Tuple[bytes, float]] = field(default_factory=list)
    tensors: List[GGMLTensor] = field(default_factory=list)

    @property
    def is_extended(self) -> bool:
        return self.hparams.first_layer is not None

    def tensor_map(self) -> Dict[str, GGMLTensor]:
        return {t.name: t for t in self.tensors}

    # ---------------- writing ----------------

    def save(self, path: str) -> None:
        with open(path, "wb") as f:
            self._write(f)

    def _write(self, f: BinaryIO) -> None:
        write_header(f, self.hparams, self.vocab)
        for t in self.tensors:
            write_tensor_record(f, t)

    # ---------------- reading ----------------

    @classmethod
    def load(cls, path: str, extended: bool, with_data: bool = True,
             use_mmap: bool = True) -> "GGMLFile":
        """Read a GGJT v3/v4 file.

        ``extended=True`` for slice/extra_layers files (8-field hparams),
        ``False`` for original model files (7-field).

        ``use_mmap`` (default): tensor bytes are zero-copy views into a
        file mapping — the analog of the reference's mmap load
        (tensor_processor.cpp:996-1074). Host RSS stays bounded by the
        working set (the OS pages tensor data in on first touch and may
        evict it), which is what makes 100 GB-class checkpoints loadable
        through the per-tensor streaming repack. The mapping lives as
        long as the GGMLFile (tensors hold memoryviews into it).
        """
        if use_mmap and with_data:
            import mmap as _mmap
            f = open(path, "rb")
            try:
                mm = _mmap.mmap(f.fileno(), 0, access=_mmap.ACCESS_READ)
            except ValueError:  # empty file
                f.close()
                raise ValueError(f"truncated GGJT file {path!r} (empty)")
            out = cls._parse(path, memoryview(mm), extended, True)
            out._mmap = mm      # keep the mapping alive
            out._mmap_file = f
            return out
        with open(path, "rb") as f:
            data = f.read()
        return cls._parse(path, data, extended, with_data)

    @classmethod
    def _parse(cls, path: str, data, extended: bool,
               with_data: bool) -> "GGMLFile":
        off = 0

        def u32() -> int:
            nonlocal off
            if off + 4 > len(data):
                raise ValueError(
                    f"truncated GGJT file {path!r} (at byte {off})")
            (v,) = struct.unpack_from("<I", data, off)
            off += 4
            return v

        magic = u32()
        if magic != GGJT_MAGIC:
            raise ValueError(f"bad magic 0x{magic:08x}; not a GGJT file")
        version = u32()
        if version not in (GGJT_VERSION, GGJT_VERSION_GQA):
            raise ValueError(f"unsupported GGJT version {version}")
        n_vocab = u32()
        n_embd = u32()
        n_mult = u32()
        n_head = u32()
        n_head_kv = u32() if version == GGJT_VERSION_GQA else None
        n_layer = u32()
        n_rot = u32()
        first_layer = u32() if extended else None
        ftype = u32()
        hp = Hparams(n_vocab=n_vocab, n_embd=n_embd, n_mult=n_mult,
                     n_head=n_head, n_layer=n_layer, n_rot=n_rot,
                     ftype=ftype, first_layer=first_layer,
                     n_head_kv=n_head_kv)

        vocab: List[Tuple[bytes, float]] = []
        for _ in range(n_vocab):
            ln = u32()
            word = bytes(data[off:off + ln])
            off += ln
            (score,) = struct.unpack_from("<f", data, off)
            off += 4
            vocab.append((word, score))

        tensors: List[GGMLTensor] = []
        total = len(data)
        while off < total:
            n_dims = u32()
            name_len = u32()
            gtype = u32()
            if n_dims < 1 or n_dims > 2:
                raise ValueError(f"tensor with {n_dims} dims")
            ne = struct.unpack_from("<%dI" % n_dims, data, off)
            off += 4 * n_dims
            name = bytes(data[off:off + name_len]).decode("utf-8")
            off += name_len
            off += -off & 31
            size = tensor_nbytes(gtype, ne)
            raw = data[off:off + size] if with_data else b""
            if with_data and len(raw) != size:
                raise ValueError(f"truncated tensor data for {name}")
            off += size
            # mmap mode: raw stays a zero-copy view into the mapping;
            # eager mode: a bytes copy as before
            if not isinstance(data, memoryview):
                raw = bytes(raw)
            tensors.append(GGMLTensor(name=name, ne=tuple(ne), gtype=gtype,
                                      raw=raw))
        return cls(hparams=hp, vocab=vocab, tensors=tensors)


def write_header(f: BinaryIO, hp: Hparams,
                 vocab: List[Tuple[bytes, float]]) -> None:
    gqa = hp.is_gqa
    f.write(struct.pack("<II", GGJT_MAGIC,
                        GGJT_VERSION_GQA if gqa else GGJT_VERSION))
    fields = [hp.n_vocab, hp.n_embd, hp.n_mult, hp.n_head]
    if gqa:
        fields.append(hp.kv_heads)
    fields += [hp.n_layer, hp.n_rot]
    if hp.first_layer is not None:
        fields.append(hp.first_layer)
    fields.append(hp.ftype)
    f.write(struct.pack("<%dI" % len(fields), *fields))
    if len(vocab) != hp.n_vocab:
        raise ValueError(
            f"vocab has {len(vocab)} entries, hparams say {hp.n_vocab}")
    for word, score in vocab:
        f.write(struct.pack("<I", len(word)))
        f.write(word)
        f.write(struct.pack("<f", score))


def write_tensor_record(f: BinaryIO, t: GGMLTensor) -> None:
    name_b = t.name.encode("utf-8")
    f.write(struct.pack("<III", len(t.ne), len(name_b), t.gtype))
    f.write(struct.pack("<%dI" % len(t.ne), *t.ne))
    f.write(name_b)
    pad = -f.tell() & 31
    f.write(b"\x00" * pad)
    if len(t.raw) != t.nbytes:
        raise ValueError(
            f"tensor {t.name}: raw {len(t.raw)} B != expected "
            f"{t.nbytes} B")
    f.write(t.raw)


class GGMLWriter:
    """Streaming writer: header first, then one tensor record at a time
    — builds 100 GB-class synthetic checkpoints without materializing
    them in RAM (GGMLFile.save needs every tensor resident)."""

    def __init__(self, path: str, hp: Hparams,
                 vocab: List[Tuple[bytes, float]]):
        self._f = open(path, "wb")
        write_header(self._f, hp, vocab)

    def add(self, t: GGMLTensor) -> None:
        write_tensor_record(self._f, t)

    def close(self) -> None:
        self._f.close()

    def __enter__(self) -> "GGMLWriter":
        return self

    def __exit__(self, *exc) -> None:
        self.close()


def sniff_extended(path: str) -> bool:
    """Heuristic: is this file in the extended (8-field) header format?

    The extended header is valid iff interpreting field 7 as ``first_layer``
    and field 8 as ``ftype`` yields a known ftype AND the resulting vocab
    walk lands exactly on a tensor record. We use the cheap test: ftype
    value plausibility in both interpretations, preferring the explicit one
    when unambiguous.
    """
    with open(path, "rb") as f:
        head = f.read(4 * 11)
    if len(head) < 4 * 11:
        raise ValueError(f"truncated GGJT file {path!r} "
                         f"({len(head)} bytes)")
    vals = struct.unpack_from("<11I", head, 0)
    if vals[0] != GGJT_MAGIC:
        raise ValueError("not a GGJT file")
    # v4 (GQA) shifts every index after n_head by one
    shift = 1 if vals[1] == GGJT_VERSION_GQA else 0
    # indices (v3): 0 magic, 1 version, 2 n_vocab, 3 n_embd, 4 n_mult,
    # 5 n_head, 6 n_layer, 7 n_rot, then [8]=ftype (7-field) or
    # first_layer (8-field), [9]=ftype (8-field).
    seven_ok = vals[8 + shift] in _FTYPE_TO_GGML
    eight_ok = vals[9 + shift] in _FTYPE_TO_GGML and (
        vals[8 + shift] == EXTRA_LAYERS_FIRST_LAYER or
        vals[8 + shift] < 4096)
    if eight_ok and not seven_ok:
        return True
    if seven_ok and not eight_ok:
        return False
    # Ambiguous (both parse): treat files with the extra_layers sentinel or a
    # small first_layer AND a valid trailing ftype as extended.
    return eight_ok
