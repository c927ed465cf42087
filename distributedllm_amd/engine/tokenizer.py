"""SentencePiece-style greedy-BPE tokenizer over a GGML vocab.

Behavioral parity with the reference's `llama_tokenizer`
(/root/reference/distllm/tensor_processor.cpp:1596-1697):

* text splits into UTF-8 characters; the work queue greedily merges the
  highest-scoring adjacent pair found in the vocab until no merge applies,
* symbols with no vocab entry fall back to byte tokens ``id = byte + 3``,
* BOS = 1 prepended when requested; EOS = 2; UNK = 0,
* detokenization maps ``▁`` to space and ``<0xXX>`` byte tokens to their
  byte (llama.cpp token_to_str behavior of that era).

Pure Python: tokenization is not a hot path (one call per generation).
"""
from __future__ import annotations

import heapq
import re
from typing import Dict, List, Sequence, Tuple

BOS_ID = 1
EOS_ID = 2
UNK_ID = 0

_BYTE_RE = re.compile(r"^<0x([0-9A-Fa-f]{2})>$")
SPM_SPACE = "▁"  # ▁


def _utf8_char_lens(b: bytes) -> List[int]:
    lens = []
    i = 0
    while i < len(b):
        c = b[i]
        if c < 0x80:
            n = 1
        elif c >> 5 == 0b110:
            n = 2
        elif c >> 4 == 0b1110:
            n = 3
        elif c >> 3 == 0b11110:
            n = 4
        else:
            n = 1  # invalid lead byte: treat as single byte
        n = min(n, len(b) - i)
        lens.append(n)
        i += n
    return lens


class Tokenizer:
    def __init__(self, vocab: Sequence[Tuple[bytes, float]]):
        self.id_to_token: List[bytes] = [w for w, _ in vocab]
        self.scores: List[float] = [s for _, s in vocab]
        self.token_to_id: Dict[bytes, int] = {}
        for i, (w, _) in enumerate(vocab):
            # first occurrence wins (dict insert order mirrors the
            # reference's map insert semantics closely enough: duplicates
            # in real vocabs don't exist)
            self.token_to_id.setdefault(w, i)

    @property
    def n_vocab(self) -> int:
        return len(self.id_to_token)

    def encode(self, text: str, bos: bool = True) -> List[int]:
        out: List[int] = [BOS_ID] if bos else []
        if not text:
            return out if bos else []
        raw = text.encode("utf-8")
        # symbol chain over utf-8 chars
        lens = _utf8_char_lens(raw)
        pieces: List[bytes] = []
        off = 0
        for n in lens:
            pieces.append(raw[off:off + n])
            off += n
        prev = list(range(-1, len(pieces) - 1))
        nxt = list(range(1, len(pieces) + 1))
        nxt[-1] = -1
        alive = [True] * len(pieces)

        heap: List[Tuple[float, int, int, int]] = []

        def try_add(left: int, right: int) -> None:
            if left == -1 or right == -1:
                return
            merged = pieces[left] + pieces[right]
            tid = self.token_to_id.get(merged)
            if tid is None:
                return
            # max-heap on score; tie-break on left index for determinism
            heapq.heappush(heap, (-self.scores[tid], left, right,
                                  len(merged)))

        for i in range(1, len(pieces)):
            try_add(i - 1, i)

        while heap:
            _, left, right, size = heapq.heappop(heap)
            if not alive[left] or not alive[right]:
                continue
            if len(pieces[left]) + len(pieces[right]) != size:
                continue
            pieces[left] = pieces[left] + pieces[right]
            alive[right] = False
            nxt[left] = nxt[right]
            if nxt[right] != -1:
                prev[nxt[right]] = left
            try_add(prev[left] if prev[left] != -1 else -1, left)
            if nxt[left] != -1:
                try_add(left, nxt[left])

        i = 0
        while i != -1:
            tid = self.token_to_id.get(pieces[i])
            if tid is None:
                for b in pieces[i]:
                    out.append(b + 3)  # byte fallback
            else:
                out.append(tid)
            i = nxt[i]
        return out

    def decode_token(self, tid: int) -> str:
        if tid < 0 or tid >= len(self.id_to_token):
            return ""
        tok = self.id_to_token[tid].decode("utf-8", errors="replace")
        m = _BYTE_RE.match(tok)
        if m:
            return chr(int(m.group(1), 16))
        return tok.replace(SPM_SPACE, " ")

    def decode(self, ids: Sequence[int]) -> str:
        return "".join(self.decode_token(t) for t in ids
                       if t not in (BOS_ID, EOS_ID))
