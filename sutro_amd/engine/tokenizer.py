"""Byte-level BPE tokenizer (real multi-byte vocab) + legacy byte tokenizer.

The reference cloud serves BPE-tokenized models and the client's structured
outputs are produced against that vocab (`/root/reference/sutro/sdk.py:220`,
`common.py:170-181`). There is no network for tokenizer files, so the merge
table is trained offline on local text (tools/train_tokenizer.py) and shipped
as a compact artifact: ``sutro_amd/data/bpe_merges.npz`` holds rank-ordered
merges as (left_id, right_id) pairs.

Fixed id layout (shared with every truncation):

    0 PAD, 1 BOS, 2 EOS, 3..258 raw bytes, 259+i = merge rank i

Because merges are strictly rank-ordered, the tokenizer for ANY model vocab V
is the prefix of the first V-259 merges — the 512-vocab CPU-test models get a
real multi-byte-token BPE exactly like the 151,936-vocab Qwen3 configs, so
every guided-decoding/stop-string/sampling path is exercised at both scales.

Encoding runs through the Rust `tokenizers` BPE with our vocab/merges (id
layout preserved); decoding is a direct bytes join over the id->bytes table.
"""

from __future__ import annotations

import os
import threading
from typing import Dict, List, Optional

import numpy as np

PAD_ID = 0
BOS_ID = 1
EOS_ID = 2
BYTE_OFFSET = 3
# legacy constant: the byte tokenizer's vocab (kept for ByteTokenizer users)
TOKENIZER_VOCAB = BYTE_OFFSET + 256  # 259

_DATA_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                         "data")
_MERGES_PATH = os.path.join(_DATA_DIR, "bpe_merges.npz")


def _bytes_to_unicode() -> Dict[int, str]:
    """GPT-2 byte<->unicode printable mapping (standard byte-level BPE)."""
    bs = (list(range(ord("!"), ord("~") + 1)) + list(range(0xA1, 0xAD))
          + list(range(0xAE, 0x100)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, (chr(c) for c in cs)))


class BPETokenizer:
    """Byte-level BPE over the fixed id layout; exact UTF-8 round trip
    (single-byte tokens 3..258 are always available as fallback)."""

    pad_id = PAD_ID
    bos_id = BOS_ID
    eos_id = EOS_ID

    def __init__(self, merges: np.ndarray, vocab_size: Optional[int] = None):
        full = BYTE_OFFSET + 256 + len(merges)
        if vocab_size is None:
            vocab_size = full
        vocab_size = min(vocab_size, full)
        if vocab_size < BYTE_OFFSET + 256:
            raise ValueError(f"vocab_size {vocab_size} < minimum "
                             f"{BYTE_OFFSET + 256} (specials + bytes)")
        self.vocab_size = vocab_size
        n_merges = vocab_size - BYTE_OFFSET - 256
        merges = merges[:n_merges]

        # id -> bytes table (specials map to b"")
        tb: List[bytes] = [b""] * vocab_size
        for b in range(256):
            tb[BYTE_OFFSET + b] = bytes([b])
        base = BYTE_OFFSET + 256
        for i in range(n_merges):
            l, r = int(merges[i, 0]), int(merges[i, 1])
            tb[base + i] = tb[l] + tb[r]
        self._token_bytes = tb

        # Rust BPE for encoding, built with OUR ids
        from tokenizers import Tokenizer, models, pre_tokenizers

        b2u = _bytes_to_unicode()

        def s(bs: bytes) -> str:
            return "".join(b2u[x] for x in bs)

        vocab = {s(bytes([b])): BYTE_OFFSET + b for b in range(256)}
        merge_strs = []
        for i in range(n_merges):
            l, r = int(merges[i, 0]), int(merges[i, 1])
            merge_strs.append((s(tb[l]), s(tb[r])))
            vocab[s(tb[base + i])] = base + i
        model = models.BPE(vocab=vocab, merges=merge_strs)
        self._hf = Tokenizer(model)
        self._hf.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)

    # ---- core API ----

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = self._hf.encode(text, add_special_tokens=False).ids
        return [BOS_ID] + ids if add_bos else ids

    def encode_batch(self, texts: List[str], add_bos: bool = False) -> List[List[int]]:
        encs = self._hf.encode_batch(texts, add_special_tokens=False)
        if add_bos:
            return [[BOS_ID] + e.ids for e in encs]
        return [e.ids for e in encs]

    def decode(self, ids: List[int]) -> str:
        tb = self._token_bytes
        n = self.vocab_size
        data = b"".join(tb[i] for i in ids if 0 <= i < n)
        return data.decode("utf-8", errors="replace")

    def token_bytes(self, token_id: int) -> bytes:
        """Raw bytes a token contributes to output (b'' for specials)."""
        if 0 <= token_id < self.vocab_size:
            return self._token_bytes[token_id]
        return b""

    @property
    def token_bytes_table(self) -> List[bytes]:
        """The full id->bytes table (guided-decoding mask construction)."""
        return self._token_bytes

    def render_prompt(self, user: str, system: Optional[str] = None) -> List[int]:
        """Chat-template render: system + user -> prompt token ids."""
        parts = []
        if system:
            parts.append(f"<|system|>\n{system}\n")
        parts.append(f"<|user|>\n{user}\n<|assistant|>\n")
        return self.encode("".join(parts), add_bos=True)

    def render_prompts(self, users: List[str],
                       system: Optional[str] = None) -> List[List[int]]:
        """Batch chat-template render (one Rust batch-encode call — large
        job admissions tokenize thousands of rows)."""
        pre = f"<|system|>\n{system}\n" if system else ""
        texts = [f"{pre}<|user|>\n{u}\n<|assistant|>\n" for u in users]
        return self.encode_batch(texts, add_bos=True)


class ByteTokenizer:
    """Legacy lossless byte tokenizer (ids 3..258 = bytes). Kept for unit
    tests that pin byte-level semantics; the engine default is BPE."""

    pad_id = PAD_ID
    bos_id = BOS_ID
    eos_id = EOS_ID
    vocab_size = TOKENIZER_VOCAB

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = [BYTE_OFFSET + b for b in text.encode("utf-8")]
        return [BOS_ID] + ids if add_bos else ids

    def encode_batch(self, texts: List[str], add_bos: bool = False) -> List[List[int]]:
        return [self.encode(t, add_bos) for t in texts]

    def decode(self, ids: List[int]) -> str:
        data = bytes(i - BYTE_OFFSET for i in ids if BYTE_OFFSET <= i < BYTE_OFFSET + 256)
        return data.decode("utf-8", errors="replace")

    def token_bytes(self, token_id: int) -> bytes:
        if BYTE_OFFSET <= token_id < BYTE_OFFSET + 256:
            return bytes([token_id - BYTE_OFFSET])
        return b""

    @property
    def token_bytes_table(self) -> List[bytes]:
        return [self.token_bytes(i) for i in range(self.vocab_size)]

    def render_prompt(self, user: str, system: Optional[str] = None) -> List[int]:
        parts = []
        if system:
            parts.append(f"<|system|>\n{system}\n")
        parts.append(f"<|user|>\n{user}\n<|assistant|>\n")
        return self.encode("".join(parts), add_bos=True)

    def render_prompts(self, users: List[str],
                       system: Optional[str] = None) -> List[List[int]]:
        return [self.render_prompt(u, system) for u in users]


_LOCK = threading.Lock()
_MERGES: Optional[np.ndarray] = None
_CACHE: Dict[int, BPETokenizer] = {}


def _load_merges() -> np.ndarray:
    global _MERGES
    if _MERGES is None:
        if not os.path.exists(_MERGES_PATH):
            raise FileNotFoundError(
                f"BPE merge table missing at {_MERGES_PATH}; run "
                f"tools/train_tokenizer.py (one-time, offline) or restore the "
                f"committed artifact")
        with np.load(_MERGES_PATH) as z:
            _MERGES = np.ascontiguousarray(z["merges"], dtype=np.int32)
    return _MERGES


def full_vocab_size() -> int:
    """Vocab of the untruncated shipped tokenizer."""
    return BYTE_OFFSET + 256 + len(_load_merges())


def get_tokenizer(vocab_size: Optional[int] = None) -> BPETokenizer:
    """The BPE tokenizer truncated to ``vocab_size`` (None = full). Cached
    per size; pass the model spec's vocab_size so sampled ids always fit the
    embedding table (model vocabs larger than the shipped tokenizer get the
    full tokenizer; the sampler masks the dead tail)."""
    merges = _load_merges()
    full = BYTE_OFFSET + 256 + len(merges)
    v = full if vocab_size is None else min(vocab_size, full)
    with _LOCK:
        t = _CACHE.get(v)
        if t is None:
            t = BPETokenizer(merges, v)
            _CACHE[v] = t
        return t
