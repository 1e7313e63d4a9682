"""Tokenizers: HF tokenizer.json (via the `tokenizers` lib) when a local
checkpoint provides one; a deterministic byte-level tokenizer otherwise.

There is no network in this deployment for downloading tokenizer files, so
the byte tokenizer is the offline default — it makes the full serve path
(encode → engine → decode → stream) real end-to-end on synthetic models.
"""
from __future__ import annotations

import os
from typing import List, Optional


class ByteTokenizer:
    """ids 0..255 = raw bytes shifted by n_special; specials at the front."""

    def __init__(self, vocab_size: int = 512, bos_id: int = 1, eos_id: int = 2) -> None:
        self.n_special = 4  # pad, bos, eos, unk
        self.vocab_size = max(vocab_size, 256 + self.n_special)
        self.bos_token_id = bos_id if bos_id < self.n_special else 1
        self.eos_token_id = eos_id if eos_id < self.n_special else 2
        self.pad_token_id = 0

    def encode(self, text: str, add_bos: bool = True) -> List[int]:
        ids = [b + self.n_special for b in text.encode("utf-8")]
        return ([self.bos_token_id] if add_bos else []) + ids

    def decode(self, ids: List[int]) -> str:
        # total decode: ids past the byte range fold back onto bytes, so
        # every sampled id yields text (a real tokenizer maps every id to a
        # string; synthetic serving benchmarks rely on that for streaming)
        data = bytes(
            (i - self.n_special) % 256 for i in ids if i >= self.n_special
        )
        return data.decode("utf-8", errors="replace")


class HFTokenizer:
    """Wraps a local tokenizer.json via the `tokenizers` package."""

    def __init__(self, path: str) -> None:
        from tokenizers import Tokenizer

        self.tk = Tokenizer.from_file(path)
        self.vocab_size = self.tk.get_vocab_size()
        self.bos_token_id = self._special_id(("<|begin_of_text|>", "<s>", "<bos>"))
        self.eos_token_id = self._special_id(("<|eot_id|>", "<|end_of_text|>", "</s>", "<eos>"))
        self.pad_token_id = 0

    def _special_id(self, names) -> Optional[int]:
        for n in names:
            tid = self.tk.token_to_id(n)
            if tid is not None:
                return tid
        return None

    def encode(self, text: str, add_bos: bool = True) -> List[int]:
        ids = self.tk.encode(text).ids
        if add_bos and self.bos_token_id is not None and (not ids or ids[0] != self.bos_token_id):
            ids = [self.bos_token_id] + ids
        return ids

    def decode(self, ids: List[int]) -> str:
        return self.tk.decode(ids)


class BPETokenizer:
    """Older GPT-2-style checkpoints ship vocab.json + merges.txt instead
    of tokenizer.json; build a byte-level BPE from them."""

    def __init__(self, vocab_path: str, merges_path: str,
                 eos_id: int = 50256) -> None:
        from tokenizers import Tokenizer, decoders, models, pre_tokenizers

        tk = Tokenizer(models.BPE.from_file(vocab_path, merges_path))
        tk.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
        tk.decoder = decoders.ByteLevel()
        self.tk = tk
        self.vocab_size = tk.get_vocab_size()
        tid = tk.token_to_id("<|endoftext|>")
        self.bos_token_id = tid if tid is not None else eos_id
        self.eos_token_id = self.bos_token_id
        self.pad_token_id = self.bos_token_id

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        return self.tk.encode(text).ids  # GPT-2 adds no BOS

    def decode(self, ids: List[int]) -> str:
        return self.tk.decode(ids)


def load_tokenizer(model_path: Optional[str], vocab_size: int, bos_id: int, eos_id: int):
    if model_path:
        tj = os.path.join(model_path, "tokenizer.json")
        if os.path.isfile(tj):
            return HFTokenizer(tj)
        vj = os.path.join(model_path, "vocab.json")
        mg = os.path.join(model_path, "merges.txt")
        if os.path.isfile(vj) and os.path.isfile(mg):
            return BPETokenizer(vj, mg, eos_id=eos_id)
    return ByteTokenizer(vocab_size=vocab_size, bos_id=bos_id, eos_id=eos_id)
