"""Random-init GGML model generator for synthetic benchmarking.

There is no network access for datasets or checkpoints, so
benchmarks and end-to-end tests run on random-init weights of the real
architectures. This module writes standard GGJT-v3 model files (7-field
header) that the slicer/provisioner then cuts exactly like real ones.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np

from ..models.llama import PRESETS, EXTRA_TENSOR_NAMES, LlamaPreset
from . import ggml


def synthetic_vocab(n_vocab: int) -> List[Tuple[bytes, float]]:
    """A functional SentencePiece-style vocab.

    ids 0-2: <unk>, <s>, </s>; ids 3-258: byte tokens (the byte-fallback
    range the tokenizer maps bytes into, tensor_processor.cpp:1660); the
    rest: synthetic word pieces with increasing scores so the greedy-BPE
    merge loop has something to merge.
    """
    vocab: List[Tuple[bytes, float]] = [(b"<unk>", 0.0), (b"<s>", 0.0),
                                        (b"</s>", 0.0)]
    for b in range(256):
        vocab.append((f"<0x{b:02X}>".encode(), 0.0))
    words = ["▁the", "▁of", "▁and", "▁a", "▁to",
             "▁in", "▁is", "▁for", "he", "at", "on", "er",
             "an", "ing", "ion", "es", "en", "▁hello", "▁world",
             "ll", "lo", "wor", "ld", "▁he", "▁wo"]
    i = len(vocab)
    wi = 0
    while i < n_vocab:
        if wi < len(words):
            tok = words[wi].encode()
        else:
            tok = f"▁w{wi}".encode()
        vocab.append((tok, -float(wi)))
        wi += 1
        i += 1
    return vocab[:n_vocab]


def build_model(preset: str | LlamaPreset, ftype: int = ggml.FTYPE_MOSTLY_Q4_0,
                seed: int = 0, scale: float = 0.02,
                n_layer: Optional[int] = None) -> ggml.GGMLFile:
    """Random-init model in the standard (7-field) GGJT v3 layout.

    2-D tensors take the model ftype; 1-D norm weights stay f32 (near 1.0),
    matching how real checkpoints are laid out.
    """
    p = PRESETS[preset] if isinstance(preset, str) else preset
    L = p.n_layer if n_layer is None else n_layer
    hp = ggml.Hparams(n_vocab=p.n_vocab, n_embd=p.n_embd, n_mult=p.n_mult,
                      n_head=p.n_head, n_layer=L, n_rot=p.n_rot, ftype=ftype,
                      n_head_kv=p.n_head_kv)
    wt = ggml._FTYPE_TO_GGML[ftype]
    rng = np.random.default_rng(seed)
    E, F, V = p.n_embd, p.n_ff, p.n_vocab
    Ekv = p.n_embd_kv  # K/V projection width (= E for MHA)

    def mat(rows: int, cols: int) -> np.ndarray:
        return rng.standard_normal((rows, cols), dtype=np.float32) * scale

    def norm_w(n: int) -> np.ndarray:
        return (1.0 + rng.standard_normal(n).astype(np.float32) * 0.01)

    tensors: List[ggml.GGMLTensor] = [
        ggml.GGMLTensor.from_f32("tok_embeddings.weight", mat(V, E), wt),
        ggml.GGMLTensor.from_f32("norm.weight", norm_w(E), ggml.GGML_TYPE_F32),
        ggml.GGMLTensor.from_f32("output.weight", mat(V, E), wt),
    ]
    for i in range(L):
        pre = f"layers.{i}."
        tensors += [
            ggml.GGMLTensor.from_f32(pre + "attention_norm.weight",
                                     norm_w(E), ggml.GGML_TYPE_F32),
            ggml.GGMLTensor.from_f32(pre + "attention.wq.weight", mat(E, E), wt),
            ggml.GGMLTensor.from_f32(pre + "attention.wk.weight",
                                     mat(Ekv, E), wt),
            ggml.GGMLTensor.from_f32(pre + "attention.wv.weight",
                                     mat(Ekv, E), wt),
            ggml.GGMLTensor.from_f32(pre + "attention.wo.weight", mat(E, E), wt),
            ggml.GGMLTensor.from_f32(pre + "ffn_norm.weight",
                                     norm_w(E), ggml.GGML_TYPE_F32),
            ggml.GGMLTensor.from_f32(pre + "feed_forward.w1.weight",
                                     mat(F, E), wt),
            ggml.GGMLTensor.from_f32(pre + "feed_forward.w2.weight",
                                     mat(E, F), wt),
            ggml.GGMLTensor.from_f32(pre + "feed_forward.w3.weight",
                                     mat(F, E), wt),
        ]
    return ggml.GGMLFile(hparams=hp, vocab=synthetic_vocab(V),
                         tensors=tensors)
