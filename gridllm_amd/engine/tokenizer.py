"""Tokenization.

Default is a self-contained byte-level tokenizer (ids 0..255 = raw bytes +
specials) so the framework runs with no downloaded assets (this image has
no network). If a HuggingFace tokenizer.json path is supplied, the
`tokenizers` wheel (installed offline) is used instead.
"""

from __future__ import annotations

import os
from typing import List, Optional


_PSEUDO_CACHE: dict = {}


def _pseudo_word(tid: int) -> str:
    """Deterministic pseudo-word for an unassigned vocab id. Real
    tokenizers decode every non-special id to text; mapping the synthetic
    vocab's unassigned ids (258..V) to "" made 99.8% of sampled tokens
    invisible on a random-init model — streaming looked silent and TTFT
    was unmeasurable (gpurun_out/r02ab traces). Memoized: this runs on
    the batcher thread once per generated token."""
    w = _PSEUDO_CACHE.get(tid)
    if w is None:
        i = tid - 258
        n = 3 + (i % 3)
        out = []
        for _ in range(n):
            out.append(chr(97 + i % 26))
            i //= 26
        w = _PSEUDO_CACHE[tid] = "".join(out) + " "
    return w


class ByteTokenizer:
    BOS = 256
    EOS = 257

    def __init__(self, vocab_size: int = 512):
        self.vocab_size = max(vocab_size, 258)

    def encode(self, text: str, bos: bool = True) -> List[int]:
        ids = list(text.encode("utf-8"))
        return ([self.BOS] if bos else []) + ids

    def decode(self, ids: List[int]) -> str:
        parts: List[str] = []
        run: List[int] = []            # contiguous raw-byte run
        for i in ids:
            if 0 <= i < 256:
                run.append(i)
                continue
            if 256 <= i < 258:
                continue               # BOS/EOS: text-invisible, and they
                                       # must NOT split a byte run (stream
                                       # decode equivalence — fuzz found
                                       # [0xC2, BOS, 0x80] diverging)
            if run:
                parts.append(bytes(run).decode("utf-8", errors="replace"))
                run = []
            parts.append(_pseudo_word(i))
        if run:
            parts.append(bytes(run).decode("utf-8", errors="replace"))
        return "".join(parts)

    def decode_stream(self) -> "ByteStreamDecoder":
        return ByteStreamDecoder()

    @property
    def eos_id(self) -> int:
        return self.EOS


class ByteStreamDecoder:
    """Incremental UTF-8-safe decoding for token streaming, built on the
    stdlib incremental codec: its output over any chunking is IDENTICAL
    to batch-decoding the concatenated bytes (a hand-rolled 4-byte
    bailout was measured splitting a code point whose lead byte arrived
    inside the bailout window — hypothesis fuzz in
    tests/test_fuzz_decoders.py)."""

    def __init__(self) -> None:
        import codecs
        self._dec = codecs.getincrementaldecoder("utf-8")("replace")

    def push(self, token_id: int) -> str:
        if token_id >= 258:            # unassigned id -> pseudo-word
            held = self.flush()        # emit any partial byte run first
            return held + _pseudo_word(token_id)
        if not (0 <= token_id < 256):
            return ""                  # BOS/EOS
        return self._dec.decode(bytes((token_id,)))

    def flush(self) -> str:
        out = self._dec.decode(b"", True)
        self._dec.reset()
        return out


def _byte_level_decoder() -> dict:
    """Invert the GPT-2/byte-level-BPE printable-alphabet mapping:
    token-string char -> original raw byte."""
    bs = list(range(ord("!"), ord("~") + 1)) + \
        list(range(0xA1, 0xAC + 1)) + list(range(0xAE, 0xFF + 1))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return {chr(c): b for b, c in zip(bs, cs)}


class HFTokenizer:
    def __init__(self, path: str):
        from tokenizers import Tokenizer  # offline wheel
        self._tok = Tokenizer.from_file(path)
        self.vocab_size = self._tok.get_vocab_size()
        self.eos_id = self.vocab_size - 1
        for cand in ("<|end_of_text|>", "</s>", "<|eot_id|>"):
            tid = self._tok.token_to_id(cand)
            if tid is not None:
                self.eos_id = tid
                break
        self._tb: Optional[list] = None      # id -> raw bytes (lazy)
        self._byte_tok: Optional[dict] = None

    def _build_token_bytes(self) -> None:
        dec = _byte_level_decoder()
        special: set = set()
        try:
            special = {i for i, t in
                       self._tok.get_added_tokens_decoder().items()
                       if t.special}
        except Exception:  # noqa: BLE001 — older tokenizers versions
            pass
        tb: list = []
        for i in range(self.vocab_size):
            s = self._tok.id_to_token(i)
            if s is None or i in special:
                tb.append(None)
                continue
            try:
                tb.append(bytes(dec[ch] for ch in s))
            except KeyError:
                # special token or non-byte-level alphabet: approximate
                # with the decoded text (empty -> not representable)
                t = self._tok.decode([i])
                tb.append(t.encode("utf-8") if t else None)
        self._tb = tb
        self._byte_tok = {}
        for i, bs in enumerate(tb):
            if bs is not None and len(bs) == 1 and bs[0] not in \
                    self._byte_tok:
                self._byte_tok[bs[0]] = i

    def token_bytes(self, tid: int):
        """Raw bytes the token contributes to the output stream (None for
        specials) — token-level grammar constraints check these against
        the byte automaton (engine/constraints.py)."""
        if self._tb is None:
            self._build_token_bytes()
        if 0 <= tid < len(self._tb):
            return self._tb[tid]
        return None

    def byte_token_id(self, b: int):
        """Token id of the single-byte fallback token for byte b (byte-
        level BPE vocabs contain all 256) — the guaranteed-progress path
        when no high-probability token is grammar-legal."""
        if self._byte_tok is None:
            self._build_token_bytes()
        return self._byte_tok.get(b)

    def encode(self, text: str, bos: bool = True) -> List[int]:
        return self._tok.encode(text).ids

    def decode(self, ids: List[int]) -> str:
        return self._tok.decode(ids)

    def decode_stream(self) -> "HFStreamDecoder":
        return HFStreamDecoder(self._tok)


class HFStreamDecoder:
    """Streams decoded text token-by-token, holding back spans whose tail
    is an incomplete UTF-8 sequence. Decodes only the UNFLUSHED tail ids
    (not the whole history — a full re-decode per token is O(n^2) over a
    generation), emitting and resetting at each clean boundary."""

    # Longest hold before emitting with replacement chars. A trained
    # model emits invalid-UTF-8 spans of at most a few tokens; without a
    # bound, a pathological stream (e.g. random byte-fragment tokens)
    # whose tail KEEPS ending mid-codepoint holds the entire generation
    # until flush() — the stream goes silent and the span grows without
    # limit (measured: TTFT == total on random-init weights).
    MAX_HOLD = 16

    def __init__(self, tok) -> None:
        self._tok = tok
        self._tail: List[int] = []

    def push(self, token_id: int) -> str:
        self._tail.append(token_id)
        text = self._tok.decode(self._tail)
        if text.endswith("�") and len(self._tail) < self.MAX_HOLD:
            return ""                   # mid-codepoint: hold the span
        self._tail.clear()
        return text

    def flush(self) -> str:
        if not self._tail:
            return ""
        text = self._tok.decode(self._tail)
        self._tail.clear()
        return text


def load_tokenizer(weights_path: str = "", vocab_size: int = 512):
    if weights_path:
        tok_json = os.path.join(weights_path, "tokenizer.json")
        if os.path.exists(tok_json):
            return HFTokenizer(tok_json)
    return ByteTokenizer(vocab_size)
