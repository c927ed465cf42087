"""LLaMA-v1-family architecture presets and a pure-PyTorch fp32 reference.

The reference implementation's forward graph (what every HIP kernel in
``ops/csrc`` must reproduce) is ``llama_eval_internal`` at
/root/reference/distllm/tensor_processor.cpp:474-809:

    per layer:  x  -> RMSNorm(eps=1e-6) * attn_norm_w
                   -> wq/wk/wv projections -> RoPE (GGML mode 0,
                      interleaved adjacent pairs, theta base 10000,
                      position offset n_past) on q,k
                   -> append k,v to per-layer KV cache
                   -> scores = k·q / sqrt(head_dim), causal mask, softmax
                   -> out = v·p, head-merge, wo
                   -> residual add
                   -> RMSNorm * ffn_norm_w -> silu(w1·x) * (w3·x) -> w2
                   -> residual add
    final (extra layers): RMSNorm * norm_w -> output matmul -> logits

``LlamaRefModel`` here is a clean fp32 torch implementation of that math,
used (a) as the CPU ground truth in kernel/engine parity tests and (b) as a
CPU fallback execution path. It is NOT the production path — that is the
HIP engine in ``distributedllm_amd.ops``.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, List, Optional

import numpy as np
import torch

from ..formats import ggml

RMS_EPS = 1e-6  # ggml_rms_norm eps of the reference's llama.cpp era
ROPE_BASE = 10000.0


@dataclass(frozen=True)
class LlamaPreset:
    name: str
    n_vocab: int
    n_embd: int
    n_mult: int
    n_head: int
    n_layer: int
    n_head_kv: Optional[int] = None  # GQA (llama-v2 70B class); None = MHA

    @property
    def n_rot(self) -> int:
        return self.n_embd // self.n_head

    @property
    def kv_heads(self) -> int:
        return self.n_head if self.n_head_kv is None else self.n_head_kv

    @property
    def n_embd_kv(self) -> int:
        return self.kv_heads * (self.n_embd // self.n_head)

    @property
    def n_ff(self) -> int:
        return ((2 * (4 * self.n_embd) // 3 + self.n_mult - 1)
                // self.n_mult) * self.n_mult

    def hparams(self, ftype: int,
                first_layer: Optional[int] = None) -> ggml.Hparams:
        return ggml.Hparams(n_vocab=self.n_vocab, n_embd=self.n_embd,
                            n_mult=self.n_mult, n_head=self.n_head,
                            n_layer=self.n_layer, n_rot=self.n_rot,
                            ftype=ftype, first_layer=first_layer,
                            n_head_kv=self.n_head_kv)


# n_ff sanity: 3B=8640, 7B=11008, 13B=13824, 30B=17920, 65B=22016;
# llama2_70b: n_mult 28672 makes the era formula land on 28672 exactly
PRESETS: Dict[str, LlamaPreset] = {
    "open_llama_3b": LlamaPreset("open_llama_3b", 32000, 3200, 216, 32, 26),
    "llama_7b": LlamaPreset("llama_7b", 32000, 4096, 256, 32, 32),
    "llama_13b": LlamaPreset("llama_13b", 32000, 5120, 256, 40, 40),
    "llama_30b": LlamaPreset("llama_30b", 32000, 6656, 256, 52, 60),
    "llama_65b": LlamaPreset("llama_65b", 32000, 8192, 256, 64, 80),
    # llama-v2 family (7b/13b are MHA with the same dims as v1)
    "llama2_70b": LlamaPreset("llama2_70b", 32000, 8192, 28672, 64, 80,
                              n_head_kv=8),
    # tiny configs for tests (head_dim 8 resp. 100-like non-pow2 = 20)
    "tiny": LlamaPreset("tiny", 256, 64, 32, 4, 3),
    "tiny_oddhead": LlamaPreset("tiny_oddhead", 256, 96, 32, 4, 2),
    # GQA test config: 8 query heads sharing 2 kv heads (group 4)
    "tiny_gqa": LlamaPreset("tiny_gqa", 256, 128, 32, 8, 3, n_head_kv=2),
    "small_gqa": LlamaPreset("small_gqa", 512, 512, 64, 8, 2, n_head_kv=2),
    # mid-size test config: large enough that the split-K/RT kernel paths
    # run with realistic grids (E/16=32 tiles, F=1408)
    "small": LlamaPreset("small", 512, 512, 64, 8, 2),
    # k-quant test config: E and F both multiples of QK_K=256 (F=1536)
    "small_k": LlamaPreset("small_k", 512, 512, 512, 8, 2),
}


def layer_tensor_names(i: int) -> List[str]:
    return [
        f"layers.{i}.attention_norm.weight",
        f"layers.{i}.attention.wq.weight",
        f"layers.{i}.attention.wk.weight",
        f"layers.{i}.attention.wv.weight",
        f"layers.{i}.attention.wo.weight",
        f"layers.{i}.ffn_norm.weight",
        f"layers.{i}.feed_forward.w1.weight",
        f"layers.{i}.feed_forward.w2.weight",
        f"layers.{i}.feed_forward.w3.weight",
    ]


EXTRA_TENSOR_NAMES = ["tok_embeddings.weight", "norm.weight", "output.weight"]


def rms_norm(x: torch.Tensor, eps: float = RMS_EPS) -> torch.Tensor:
    return x * torch.rsqrt(x.pow(2).mean(dim=-1, keepdim=True) + eps)


def rope_interleaved(x: torch.Tensor, n_past: int,
                     base: float = ROPE_BASE) -> torch.Tensor:
    """GGML mode-0 RoPE: rotate adjacent pairs (x[2i], x[2i+1]).

    x: [N, H, D]; position of token t is n_past + t; all D dims rotated
    (n_rot == D for LLaMA v1).
    """
    n, h, d = x.shape
    half = d // 2
    pos = torch.arange(n_past, n_past + n, dtype=torch.float64)
    inv = base ** (-2.0 * torch.arange(half, dtype=torch.float64) / d)
    theta = pos[:, None] * inv[None, :]              # [N, half]
    cos = torch.cos(theta).to(x.dtype)[:, None, :]   # [N,1,half]
    sin = torch.sin(theta).to(x.dtype)[:, None, :]
    x0 = x[..., 0::2]
    x1 = x[..., 1::2]
    out = torch.empty_like(x)
    out[..., 0::2] = x0 * cos - x1 * sin
    out[..., 1::2] = x0 * sin + x1 * cos
    return out


class LlamaSliceRef:
    """fp32 torch reference for one contiguous layer slice with KV cache."""

    def __init__(self, hp: ggml.Hparams, weights: Dict[str, torch.Tensor],
                 first_layer: int, n_layers: int, n_ctx: int = 512):
        self.hp = hp
        self.w = weights
        self.first_layer = first_layer
        self.n_layers = n_layers
        self.n_ctx = n_ctx
        self.n_past = 0
        d = hp.head_dim
        hkv = hp.kv_heads
        self.k_cache = torch.zeros(n_layers, n_ctx, hkv, d)
        self.v_cache = torch.zeros(n_layers, n_ctx, hkv, d)

    def clear_context(self) -> None:
        self.n_past = 0
        self.k_cache.zero_()
        self.v_cache.zero_()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: [N, E] fp32 activations; returns [N, E]. Advances n_past."""
        hp = self.hp
        n, e = x.shape
        h, d = hp.n_head, hp.head_dim
        p = self.n_past
        assert p + n <= self.n_ctx, "context overflow"
        for li in range(self.n_layers):
            gi = li + self.first_layer  # global layer index (tensor names)
            pre = f"layers.{gi}."
            a = rms_norm(x) * self.w[pre + "attention_norm.weight"]
            q = a @ self.w[pre + "attention.wq.weight"].T
            k = a @ self.w[pre + "attention.wk.weight"].T
            v = a @ self.w[pre + "attention.wv.weight"].T
            hkv = hp.kv_heads
            q = rope_interleaved(q.view(n, h, d), p)
            k = rope_interleaved(k.view(n, hkv, d), p)
            v = v.view(n, hkv, d)
            self.k_cache[li, p:p + n] = k
            self.v_cache[li, p:p + n] = v
            # GQA: q head i attends kv head i // (H/Hkv)
            kv_map = torch.arange(h) // (h // hkv)
            keys = self.k_cache[li, :p + n][:, kv_map]   # [P+N, H, D]
            vals = self.v_cache[li, :p + n][:, kv_map]
            # scores[t, j] over j<=p+t
            att = torch.einsum("nhd,jhd->hnj", q, keys) / math.sqrt(d)
            mask = torch.arange(p + n)[None, :] > (
                p + torch.arange(n)[:, None])     # [N, P+N]
            att = att.masked_fill(mask[None], float("-inf"))
            prob = torch.softmax(att, dim=-1)
            o = torch.einsum("hnj,jhd->nhd", prob, vals).reshape(n, e)
            x = x + o @ self.w[pre + "attention.wo.weight"].T
            f = rms_norm(x) * self.w[pre + "ffn_norm.weight"]
            g = torch.nn.functional.silu(
                f @ self.w[pre + "feed_forward.w1.weight"].T)
            u = f @ self.w[pre + "feed_forward.w3.weight"].T
            x = x + (g * u) @ self.w[pre + "feed_forward.w2.weight"].T
        self.n_past = p + n
        return x


class LlamaExtraRef:
    """Embedding lookup + final norm / lm_head from extra-layer weights.

    On a machine with a GPU the lm_head matmul runs there (a [V, E] fp32
    matmul per generated token is 20-30 ms on CPU at 3B scale and
    dominated the TCP client's single-stream latency); results come back
    as CPU tensors either way."""

    def __init__(self, weights: Dict[str, torch.Tensor],
                 device: Optional[str] = None):
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = device
        self.tok = weights["tok_embeddings.weight"].to(device)   # [V, E]
        self.norm = weights["norm.weight"].to(device)            # [E]
        self.out = weights["output.weight"].to(device)           # [V, E]

    def embed(self, tokens: List[int]) -> torch.Tensor:
        idx = torch.tensor(tokens, dtype=torch.long, device=self.device)
        return self.tok[idx].cpu()

    def logits(self, x: torch.Tensor, all_logits: bool = False) -> torch.Tensor:
        x = x.to(self.device)
        y = rms_norm(x) * self.norm
        lg = y @ self.out.T
        return (lg if all_logits else lg[-1:]).cpu()


def weights_from_ggml(f: ggml.GGMLFile) -> Dict[str, torch.Tensor]:
    """Dequantize every tensor of a GGML file to fp32 torch tensors."""
    return {t.name: torch.from_numpy(np.ascontiguousarray(t.to_f32()))
            for t in f.tensors}
