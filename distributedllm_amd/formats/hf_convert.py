"""HF LLaMA checkpoint directory → GGML model file.

Capability-parity with the vendored `convert.py` the reference's
provisioning imports (/root/reference/distllm/cli_api/provision.py:204-210):
reads an HF LLaMA directory (config.json + safetensors or pytorch .bin
shards + sentencepiece tokenizer.model) and produces a standard GGJT-v3
`GGMLFile` in f16 (requantization to q4 happens in cluster.provision).

Key conversion details (must match what the engine expects):

* HF stores Q/K projections in rotate-half RoPE order; GGML (and our HIP
  kernels — rope_interleaved in models/llama.py) use interleaved order, so
  wq/wk rows are permuted with the classic llama.cpp permutation,
* HF name map: model.layers.{i}.self_attn.{q,k,v,o}_proj → layers.{i}.
  attention.w{q,k,v,o}; mlp.{gate,down,up}_proj → feed_forward.w{1,2,3};
  input_layernorm → attention_norm; post_attention_layernorm → ffn_norm,
* vocab comes from tokenizer.model via sentencepiece (piece text + score,
  byte tokens kept in their "<0xXX>" spelling, which the Tokenizer's
  byte-fallback understands); if the file is absent a synthetic byte-level
  vocab is used so conversion still yields a loadable model.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Tuple

import numpy as np

from . import ggml
from .synthetic import synthetic_vocab

_LAYER_MAP = {
    "self_attn.q_proj.weight": "attention.wq.weight",
    "self_attn.k_proj.weight": "attention.wk.weight",
    "self_attn.v_proj.weight": "attention.wv.weight",
    "self_attn.o_proj.weight": "attention.wo.weight",
    "mlp.gate_proj.weight": "feed_forward.w1.weight",
    "mlp.down_proj.weight": "feed_forward.w2.weight",
    "mlp.up_proj.weight": "feed_forward.w3.weight",
    "input_layernorm.weight": "attention_norm.weight",
    "post_attention_layernorm.weight": "ffn_norm.weight",
}
_TOP_MAP = {
    "model.embed_tokens.weight": "tok_embeddings.weight",
    "model.norm.weight": "norm.weight",
    "lm_head.weight": "output.weight",
}


def permute_rotary(w: np.ndarray, n_head: int) -> np.ndarray:
    """Rotate-half → interleaved head layout (llama.cpp convert permute)."""
    rows = w.shape[0]
    return (w.reshape(n_head, 2, rows // n_head // 2, *w.shape[1:])
             .swapaxes(1, 2).reshape(w.shape))


def find_n_mult(n_embd: int, n_ff: int) -> int:
    """Recover the GGML n_mult hparam from the actual FFN width."""
    for n_mult in range(1, 16384):
        if ((2 * (4 * n_embd) // 3 + n_mult - 1) // n_mult) * n_mult == n_ff:
            return n_mult
    raise ValueError(f"no n_mult reproduces n_ff={n_ff} for E={n_embd}")


def _load_state_dict(hf_dir: str) -> Dict[str, np.ndarray]:
    """All weights as fp32 numpy, from safetensors or torch shards."""
    out: Dict[str, np.ndarray] = {}

    st_files = sorted(f for f in os.listdir(hf_dir)
                      if f.endswith(".safetensors"))
    if st_files:
        from safetensors import safe_open
        for fn in st_files:
            with safe_open(os.path.join(hf_dir, fn), framework="np") as f:
                for k in f.keys():
                    t = f.get_tensor(k)
                    out[k] = np.asarray(t, dtype=np.float32) \
                        if t.dtype != np.float32 else t
        return out

    import torch
    pt_files = sorted(f for f in os.listdir(hf_dir)
                      if f.startswith("pytorch_model") and f.endswith(".bin"))
    if not pt_files:
        raise FileNotFoundError(f"no safetensors/bin weights in {hf_dir}")
    for fn in pt_files:
        sd = torch.load(os.path.join(hf_dir, fn), map_location="cpu",
                        weights_only=True)
        for k, v in sd.items():
            out[k] = v.float().numpy()
    return out


def load_hf_vocab(hf_dir: str,
                  n_vocab: int) -> List[Tuple[bytes, float]]:
    tok_path = os.path.join(hf_dir, "tokenizer.model")
    if not os.path.exists(tok_path):
        return synthetic_vocab(n_vocab)
    import sentencepiece as spm
    sp = spm.SentencePieceProcessor(model_file=tok_path)
    vocab: List[Tuple[bytes, float]] = []
    for i in range(sp.vocab_size()):
        vocab.append((sp.id_to_piece(i).encode("utf-8"),
                      float(sp.get_score(i))))
    # pad to the model's vocab size (some checkpoints round up)
    while len(vocab) < n_vocab:
        vocab.append((f"<pad{len(vocab)}>".encode(), -1e9))
    return vocab[:n_vocab]


def convert_hf_dir(hf_dir: str) -> ggml.GGMLFile:
    with open(os.path.join(hf_dir, "config.json")) as f:
        cfg = json.load(f)
    E = cfg["hidden_size"]
    H = cfg["num_attention_heads"]
    L = cfg["num_hidden_layers"]
    V = cfg["vocab_size"]
    F = cfg["intermediate_size"]
    # GQA (llama-2 70B class): k/v projections carry num_key_value_heads
    HKV = cfg.get("num_key_value_heads", H)
    hp = ggml.Hparams(n_vocab=V, n_embd=E, n_mult=find_n_mult(E, F),
                      n_head=H, n_layer=L, n_rot=E // H,
                      ftype=ggml.FTYPE_MOSTLY_F16,
                      n_head_kv=HKV if HKV != H else None)

    sd = _load_state_dict(hf_dir)
    tensors: List[ggml.GGMLTensor] = []

    def emit(name: str, a: np.ndarray) -> None:
        gtype = ggml.GGML_TYPE_F32 if a.ndim == 1 else ggml.GGML_TYPE_F16
        tensors.append(ggml.GGMLTensor.from_f32(name, a, gtype))

    for hf_name, g_name in _TOP_MAP.items():
        if hf_name not in sd:
            raise KeyError(f"missing tensor {hf_name!r} in {hf_dir}")
        emit(g_name, sd[hf_name])
    for i in range(L):
        for hf_suffix, g_suffix in _LAYER_MAP.items():
            k = f"model.layers.{i}.{hf_suffix}"
            if k not in sd:
                raise KeyError(f"missing tensor {k!r} in {hf_dir}")
            w = sd[k]
            if hf_suffix == "self_attn.q_proj.weight":
                w = permute_rotary(w, H)
            elif hf_suffix == "self_attn.k_proj.weight":
                w = permute_rotary(w, HKV)  # Hkv rows under GQA
            emit(f"layers.{i}.{g_suffix}", w)

    return ggml.GGMLFile(hparams=hp, vocab=load_hf_vocab(hf_dir, V),
                         tensors=tensors)
