"""Train the framework's byte-level BPE merge list (one-time, offline).

The engine needs a real BPE tokenizer at production vocab scale (the reference
cloud serves BPE-tokenized models; `/root/reference/sutro/sdk.py:220` implies
schema-guided decoding over that vocab). There is no network for downloading
tokenizer files, so the merge table is trained here on local text (installed
Python sources = real English + code + punctuation, plus synthetic JSON and
Zipf-sampled pseudo-words for tail diversity) and committed as a compact
artifact.

Layout (fixed, see sutro_amd/engine/tokenizer.py): ids 0-2 specials,
3-258 raw bytes, 259+ merges in rank order. Because merges are strictly
rank-ordered, a tokenizer for ANY model vocab V is the prefix of the first
V-259 merges — one artifact serves the 512-vocab test models and the 151,936
Qwen3 vocab alike.

Output: sutro_amd/data/bpe_merges.npz  (int32 [N, 2] of (left_id, right_id))

Usage:  python tools/train_tokenizer.py [--vocab 151936] [--corpus-mb 260]
"""

from __future__ import annotations

import argparse
import glob
import io
import json
import os
import random

import numpy as np

BYTE_OFFSET = 3
FULL_VOCAB_DEFAULT = 151936


def bytes_to_unicode():
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


def iter_source_text(cap_bytes: int):
    """Stream text from installed Python sources (deterministic order)."""
    roots = [
        "/usr/lib/python3.10",
        "/usr/local/lib/python3.10/dist-packages",
    ]
    total = 0
    for root in roots:
        files = sorted(glob.glob(os.path.join(root, "**", "*.py"),
                                 recursive=True))
        for f in files:
            if "test" in f:   # keep it lighter / less repetitive
                continue
            try:
                with io.open(f, "r", encoding="utf-8", errors="ignore") as fh:
                    txt = fh.read()
            except OSError:
                continue
            if not txt:
                continue
            total += len(txt)
            yield txt
            if total >= cap_bytes:
                return


def iter_synthetic(seed: int, n_json: int, n_words: int):
    """JSON rows (structural chars + field names), numbers, pseudo-words."""
    rng = random.Random(seed)
    cats = ["news", "review", "spam", "other", "positive", "negative",
            "neutral", "question", "statement", "toxic", "safe"]
    fields = ["name", "category", "sentiment", "score", "tags", "summary",
              "title", "label", "confidence", "reasoning", "content", "id"]
    syll = ["ba", "be", "bi", "bo", "bu", "da", "de", "di", "do", "du",
            "ka", "ke", "ki", "ko", "ku", "la", "le", "li", "lo", "lu",
            "ma", "me", "mi", "mo", "mu", "na", "ne", "ni", "no", "nu",
            "ra", "re", "ri", "ro", "ru", "sa", "se", "si", "so", "su",
            "ta", "te", "ti", "to", "tu", "va", "ve", "vi", "vo", "vu",
            "cha", "she", "thi", "pro", "ter", "ing", "ion", "ent", "ard",
            "ova", "ism", "est", "ory", "and", "ex", "un", "im", "al"]

    def word():
        k = rng.choice((2, 3, 3, 4))
        return "".join(rng.choice(syll) for _ in range(k))

    # Zipf-sampled pseudo-word text: gives the merge tail real repeated units
    vocab_words = [word() for _ in range(220_000)]
    buf = []
    for i in range(n_words):
        # Zipf via inverse-power index
        idx = int(len(vocab_words) * (rng.random() ** 2.2))
        w = vocab_words[min(idx, len(vocab_words) - 1)]
        if rng.random() < 0.08:
            w = w.capitalize()
        buf.append(w)
        if len(buf) >= 4000:
            yield " ".join(buf) + ".\n"
            buf = []
    if buf:
        yield " ".join(buf) + ".\n"

    for i in range(n_json):
        obj = {
            rng.choice(fields): rng.choice(cats),
            "score": rng.randint(0, 100),
            "tags": [rng.choice(cats) for _ in range(rng.randint(1, 4))],
            rng.choice(fields): word(),
        }
        yield json.dumps(obj) + "\n"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--vocab", type=int, default=FULL_VOCAB_DEFAULT)
    ap.add_argument("--corpus-mb", type=int, default=260)
    ap.add_argument("--out", type=str,
                    default="sutro_amd/data/bpe_merges.npz")
    args = ap.parse_args()

    from tokenizers import Tokenizer, models, pre_tokenizers, trainers

    n_merges = args.vocab - BYTE_OFFSET - 256
    tok = Tokenizer(models.BPE())
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    trainer = trainers.BpeTrainer(
        vocab_size=256 + n_merges,
        min_frequency=2,
        show_progress=True,
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
        special_tokens=[],
    )

    def corpus():
        yield from iter_source_text(args.corpus_mb * 1_000_000)
        yield from iter_synthetic(1234, n_json=400_000, n_words=14_000_000)

    tok.train_from_iterator(corpus(), trainer)
    blob = json.loads(tok.to_str())
    merges = blob["model"]["merges"]  # rank-ordered
    print(f"trained merges: {len(merges)} (target {n_merges})")

    b2u = bytes_to_unicode()
    u2b = {c: b for b, c in b2u.items()}

    def tok_bytes(s: str) -> bytes:
        return bytes(u2b[ch] for ch in s)

    # map token strings -> our ids: bytes 3..258, merges appended in rank order
    id_of = {bytes([b]): BYTE_OFFSET + b for b in range(256)}
    out = np.empty((len(merges), 2), dtype=np.int32)
    next_id = BYTE_OFFSET + 256
    for i, m in enumerate(merges):
        if isinstance(m, str):
            a, b = m.split(" ", 1)
        else:
            a, b = m
        ab, bb = tok_bytes(a), tok_bytes(b)
        out[i, 0] = id_of[ab]
        out[i, 1] = id_of[bb]
        id_of[ab + bb] = next_id
        next_id += 1

    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    np.savez_compressed(args.out, merges=out)
    print(f"wrote {args.out}: {out.shape[0]} merges, "
          f"{os.path.getsize(args.out)} bytes")


if __name__ == "__main__":
    main()
