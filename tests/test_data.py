"""Dataset preprocessing parity (reference bee2bee/datasets.py + hf.py
preprocess_examples) — offline, via in-memory datasets."""
import pytest

datasets = pytest.importorskip("datasets")

from bee2bee_amd.data import (
    build_preprocess_config,
    preprocess_examples,
    prompts_from_dataset,
    tokenize_batch,
)
from bee2bee_amd.models.tokenizer import load_tokenizer


def _ds():
    return datasets.Dataset.from_dict(
        {"text": ["Hello Mesh", "a longer example sentence for padding", "x"]}
    )


def test_config_shape():
    cfg = build_preprocess_config("bytes", max_length=16, lower_case=True)
    assert cfg == {
        "tokenizer_name": "bytes",
        "text_field": "text",
        "max_length": 16,
        "lower_case": True,
    }


def test_preprocess_examples_fixed_length():
    out = preprocess_examples(_ds(), None, max_length=12)
    assert out.column_names >= ["text"]
    for ids, mask in zip(out["input_ids"], out["attention_mask"]):
        assert len(ids) == 12 and len(mask) == 12
        assert sum(mask) <= 12
    # roundtrip: unpadded prefix decodes back to the original text
    tok = load_tokenizer(None, 512, 1, 2)
    row = out[0]
    n = sum(row["attention_mask"])
    assert "Hello Mesh".startswith(tok.decode(row["input_ids"][:n])[:5])


def test_tokenize_batch_truncates_and_lowers():
    tok = load_tokenizer(None, 512, 1, 2)
    enc = tokenize_batch(["ABCDEF" * 50], tok, max_length=8, lower_case=True)
    assert len(enc["input_ids"][0]) == 8
    assert enc["attention_mask"][0] == [1] * 8
    dec = tok.decode(enc["input_ids"][0])
    assert dec == dec.lower()


def test_prompts_from_dataset():
    assert prompts_from_dataset(_ds(), limit=2) == [
        "Hello Mesh",
        "a longer example sentence for padding",
    ]
