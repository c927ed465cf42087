"""Model slicer: cut a GGJT v3 model into layer-range slice files.

Native-equivalent of the reference's ``slice_model`` CLI
(/root/reference/distllm/slice_model.cpp:361-446), byte-compatible:

* ``slice <model> <from> <to> [out]`` → extended-header file with
  ``n_layer = to-from+1``, ``first_layer = from``, full vocab, and every
  tensor whose name selects layer i ∈ [from, to] (names keep their ORIGINAL
  indices — loaders re-base with first_layer).
* ``extra_layers <model> [out]`` → tensors named norm*/output*/
  tok_embeddings* with ``n_layer=0``, ``first_layer=0xFFFFFFFF``.

Raw tensor bytes are copied verbatim (no requantization), so slicing is
lossless and idempotent, as in the reference.
"""
from __future__ import annotations

import re
from typing import Optional

from . import ggml

_LAYER_RE = re.compile(r"^layers\.(\d+)\.")


def layer_index(name: str) -> Optional[int]:
    m = _LAYER_RE.match(name)
    return int(m.group(1)) if m else None


def make_slice(src: ggml.GGMLFile, idx_from: int, idx_to: int) -> ggml.GGMLFile:
    if idx_from < 0 or idx_to < idx_from:
        raise ValueError(f"bad layer range [{idx_from}, {idx_to}]")
    if idx_to >= src.hparams.n_layer:
        raise ValueError(
            f"range [{idx_from},{idx_to}] exceeds n_layer={src.hparams.n_layer}")
    hp = src.hparams
    new_hp = ggml.Hparams(n_vocab=hp.n_vocab, n_embd=hp.n_embd,
                          n_mult=hp.n_mult, n_head=hp.n_head,
                          n_layer=idx_to - idx_from + 1, n_rot=hp.n_rot,
                          ftype=hp.ftype, first_layer=idx_from,
                          n_head_kv=hp.n_head_kv)
    tensors = [t for t in src.tensors
               if (li := layer_index(t.name)) is not None
               and idx_from <= li <= idx_to]
    return ggml.GGMLFile(hparams=new_hp, vocab=list(src.vocab),
                         tensors=tensors)


def make_extra_layers(src: ggml.GGMLFile) -> ggml.GGMLFile:
    hp = src.hparams
    new_hp = ggml.Hparams(n_vocab=hp.n_vocab, n_embd=hp.n_embd,
                          n_mult=hp.n_mult, n_head=hp.n_head, n_layer=0,
                          n_rot=hp.n_rot, ftype=hp.ftype,
                          first_layer=ggml.EXTRA_LAYERS_FIRST_LAYER,
                          n_head_kv=hp.n_head_kv)
    tensors = [t for t in src.tensors
               if t.name.startswith(("norm", "output", "tok_embeddings"))]
    return ggml.GGMLFile(hparams=new_hp, vocab=list(src.vocab),
                         tensors=tensors)


def slice_model_file(src_path: str, idx_from: int, idx_to: int,
                     out_path: str) -> None:
    src = ggml.GGMLFile.load(src_path, extended=False)
    make_slice(src, idx_from, idx_to).save(out_path)


def extract_extra_layers_file(src_path: str, out_path: str) -> None:
    src = ggml.GGMLFile.load(src_path, extended=False)
    make_extra_layers(src).save(out_path)
