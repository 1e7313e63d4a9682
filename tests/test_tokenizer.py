"""Tokenizer stack: byte tokenizer totality + HF tokenizer.json loading
(reference loads AutoTokenizer; we load tokenizer.json offline via the
`tokenizers` package)."""
import os

import pytest

from bee2bee_amd.models.tokenizer import ByteTokenizer, load_tokenizer


def test_byte_tokenizer_total_roundtrip():
    tok = ByteTokenizer(vocab_size=512, bos_id=1, eos_id=2)
    text = "héllo wörld ✓"
    ids = tok.encode(text)
    assert tok.decode(ids) == text
    # totality: EVERY id decodes to something (no dropped tokens in streams)
    assert isinstance(tok.decode(list(range(0, 300))), str)


def _build_tokenizer_json(path):
    tokenizers = pytest.importorskip("tokenizers")
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers

    tk = Tokenizer(models.BPE(unk_token="<unk>"))
    tk.pre_tokenizer = pre_tokenizers.Whitespace()
    trainer = trainers.BpeTrainer(
        vocab_size=200,
        special_tokens=["<unk>", "<s>", "</s>"],
    )
    tk.train_from_iterator(
        ["the mesh serves tokens", "tokens stream over the mesh"] * 20,
        trainer,
    )
    tk.save(path)


def test_hf_tokenizer_json_loaded(tmp_path):
    tj = str(tmp_path / "tokenizer.json")
    _build_tokenizer_json(tj)
    tok = load_tokenizer(str(tmp_path), 512, 1, 2)
    assert type(tok).__name__ == "HFTokenizer"
    ids = tok.encode("the mesh serves tokens")
    assert ids and ids[0] == tok.bos_token_id  # "<s>" resolved
    text = tok.decode([i for i in ids if i != tok.bos_token_id])
    assert "mesh" in text


def test_missing_tokenizer_json_falls_back(tmp_path):
    tok = load_tokenizer(str(tmp_path), 512, 1, 2)
    assert isinstance(tok, ByteTokenizer)


def test_bpe_vocab_merges_fallback(tmp_path):
    """GPT-2-era checkpoints with vocab.json + merges.txt (no
    tokenizer.json) load as a byte-level BPE."""
    import json

    vocab = {"<|endoftext|>": 0, "h": 1, "e": 2, "l": 3, "o": 4,
             "he": 5, "ll": 6, "hell": 7, "hello": 8, "Ġ": 9}
    json.dump(vocab, open(tmp_path / "vocab.json", "w"))
    (tmp_path / "merges.txt").write_text(
        "#version: 0.2\nh e\nl l\nhe ll\nhell o\n")
    from bee2bee_amd.models.tokenizer import load_tokenizer

    tk = load_tokenizer(str(tmp_path), 512, 1, 2)
    ids = tk.encode("hello")
    assert tk.decode(ids) == "hello"
    assert tk.eos_token_id == 0  # <|endoftext|>


def test_stream_decoder_with_real_byte_level_bpe(tmp_path):
    """TextStreamDecoder's prefix-slicing contract against a REAL
    byte-level BPE (the tokenizer.json family llama3/gpt2 use): decode of a
    growing id list must extend monotonically, and streamed deltas must
    reassemble the text exactly."""
    tokenizers = pytest.importorskip("tokenizers")
    from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers

    tk = Tokenizer(models.BPE(unk_token=None))
    tk.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tk.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(vocab_size=300, special_tokens=["<bos>"])
    tk.train_from_iterator(
        ["hello mesh world", "héllo wörld", "aaa bbb aaab", "😀 emoji test"],
        trainer)
    path = tmp_path / "tokenizer.json"
    tk.save(str(path))

    from bee2bee_amd.engine.engine import TextStreamDecoder
    from bee2bee_amd.models.tokenizer import HFTokenizer

    tok = HFTokenizer(str(path))
    for text in ["hello mesh world", "héllo wörld 😀", "aaab aaa bbb"]:
        ids = tok.encode(text, add_bos=False)
        # prefix monotonicity of decode (byte-level property the streaming
        # decoder relies on)
        prev = ""
        for n in range(1, len(ids) + 1):
            cur = tok.decode(ids[:n])
            assert cur.startswith(prev) or cur.rstrip("�").startswith(
                prev.rstrip("�")), (prev, cur)
            prev = cur
        dec = TextStreamDecoder(tok, flush_every=1)
        out = ""
        for n in range(1, len(ids) + 1):
            out += dec.delta(ids[:n])
        out += dec.delta(ids, final=True)
        assert out == tok.decode(ids)


def test_engine_serves_with_real_tokenizer_json(tmp_path):
    """End to end: a checkpoint dir carrying a real tokenizer.json serves
    through the engine with text in/out via that tokenizer (streamed, with
    a stop string)."""
    tokenizers = pytest.importorskip("tokenizers")
    import torch
    from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers

    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.weights import ModelWeights, save_hf

    tk = Tokenizer(models.BPE(unk_token=None))
    tk.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tk.decoder = decoders.ByteLevel()
    tk.train_from_iterator(["the mesh serves tokens"],
                           trainers.BpeTrainer(vocab_size=280))
    d = tmp_path / "ckpt"
    w = ModelWeights(PRESETS["tiny"], torch.device("cpu"),
                     torch.float32).random_init(2)
    save_hf(w, str(d))
    tk.save(str(d / "tokenizer.json"))

    from bee2bee_amd.engine.engine import InferenceEngine

    eng = InferenceEngine("tiny", device="cpu", model_path=str(d),
                          max_batch=2, max_seq_len=128, seed=2)
    try:
        assert type(eng.tokenizer).__name__ == "HFTokenizer"
        chunks = []
        res = eng.generate_text("the mesh", max_new_tokens=12,
                                temperature=0.0, on_text=chunks.append)
        assert res["text"] == "".join(chunks)
        assert res["tokens"] == 12
        # deterministic rerun matches
        res2 = eng.generate_text("the mesh", max_new_tokens=12,
                                 temperature=0.0)
        assert res2["text"] == res["text"]
    finally:
        eng.shutdown()
