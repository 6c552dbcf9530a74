"""Deterministic byte-level tokenizer.

There is no network access for real tokenizer files, so the engine ships a
self-contained byte-level tokenizer: ids 0..2 are specials, 3..258 are raw
bytes. It is exact (lossless UTF-8 round trip), fast, and makes the guided-
decoding FSM (JSON-schema -> DFA over bytes) trivially correct: one token = one
byte of output, so schema-valid JSON is produced even with random weights.

Models with larger vocab_size simply never see ids >= 259 from the tokenizer;
the sampler masks them out (or the FSM mask does).
"""

from __future__ import annotations

from typing import List, Optional

PAD_ID = 0
BOS_ID = 1
EOS_ID = 2
BYTE_OFFSET = 3
TOKENIZER_VOCAB = BYTE_OFFSET + 256  # 259


class ByteTokenizer:
    """Lossless byte tokenizer with a minimal chat template."""

    pad_id = PAD_ID
    bos_id = BOS_ID
    eos_id = EOS_ID
    vocab_size = TOKENIZER_VOCAB

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = [BYTE_OFFSET + b for b in text.encode("utf-8")]
        return [BOS_ID] + ids if add_bos else ids

    def decode(self, ids: List[int]) -> str:
        data = bytes(i - BYTE_OFFSET for i in ids if BYTE_OFFSET <= i < BYTE_OFFSET + 256)
        return data.decode("utf-8", errors="replace")

    def token_bytes(self, token_id: int) -> bytes:
        """Raw bytes a token contributes to output ('' for specials/out-of-range)."""
        if BYTE_OFFSET <= token_id < BYTE_OFFSET + 256:
            return bytes([token_id - BYTE_OFFSET])
        return b""

    def render_prompt(self, user: str, system: Optional[str] = None) -> List[int]:
        """Chat-template render: system + user -> prompt token ids."""
        parts = []
        if system:
            parts.append(f"<|system|>\n{system}\n")
        parts.append(f"<|user|>\n{user}\n<|assistant|>\n")
        return self.encode("".join(parts), add_bos=True)


_TOKENIZER = ByteTokenizer()


def get_tokenizer() -> ByteTokenizer:
    return _TOKENIZER
