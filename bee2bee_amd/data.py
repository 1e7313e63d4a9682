"""Dataset loading + tokenization helpers (parity with the reference's
dataset preprocessing aids: bee2bee/datasets.py:1-24 and
bee2bee/hf.py:161-176).

The reference wraps HF `datasets.load_dataset` + an AutoTokenizer map for
its legacy training coordinator. Here the same capability targets OUR
tokenizers (models/tokenizer.py) and the engine's prompt format, works
fully offline (local datasets / in-memory dicts — this deployment has no
hub egress), and pads/truncates to fixed length for batched prefill.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence


def build_preprocess_config(
    tokenizer_name: str,
    text_field: str = "text",
    max_length: int = 128,
    lower_case: bool = False,
) -> Dict[str, Any]:
    """Same shape as the reference's config (bee2bee/datasets.py:5-16)."""
    return {
        "tokenizer_name": tokenizer_name,
        "text_field": text_field,
        "max_length": int(max_length),
        "lower_case": bool(lower_case),
    }


def load_dataset(name_or_path: str, split: str = "train", streaming: bool = False, **kwargs):
    """HF `datasets` passthrough; accepts local paths (offline) as well as
    hub names (when egress exists)."""
    from datasets import load_dataset as _ld

    return _ld(name_or_path, split=split, streaming=streaming, **kwargs)


def _resolve_tokenizer(tokenizer_name: Optional[str], vocab_size: int = 512):
    """Our tokenizer stack: a model dir with tokenizer.json -> HFTokenizer,
    otherwise the byte tokenizer (total, never drops ids)."""
    from .models.tokenizer import load_tokenizer

    return load_tokenizer(tokenizer_name, vocab_size, 1, 2)


def tokenize_batch(
    texts: Sequence[str],
    tokenizer,
    max_length: int = 128,
    lower_case: bool = False,
    pad_id: int = 0,
) -> Dict[str, List[List[int]]]:
    """Fixed-length encode: truncation + right padding + attention mask
    (the reference's padding='max_length' behavior, bee2bee/hf.py:174)."""
    input_ids: List[List[int]] = []
    attention_mask: List[List[int]] = []
    for t in texts:
        if lower_case:
            t = t.lower()
        ids = tokenizer.encode(t)[:max_length]
        mask = [1] * len(ids)
        if len(ids) < max_length:
            pad = max_length - len(ids)
            ids = ids + [pad_id] * pad
            mask = mask + [0] * pad
        input_ids.append(ids)
        attention_mask.append(mask)
    return {"input_ids": input_ids, "attention_mask": attention_mask}


def preprocess_examples(
    dataset,
    tokenizer_name: Optional[str],
    text_field: str = "text",
    max_length: int = 128,
    lower_case: bool = False,
):
    """dataset.map over the text field -> input_ids/attention_mask columns
    (reference: bee2bee/hf.py:167-176, with AutoTokenizer replaced by our
    offline tokenizer stack)."""
    tok = _resolve_tokenizer(tokenizer_name)

    def _proc(batch):
        return tokenize_batch(
            batch[text_field], tok, max_length=max_length, lower_case=lower_case
        )

    return dataset.map(_proc, batched=True)


def load_and_preprocess(
    dataset_name: str, split: str, config: Dict[str, Any], streaming: bool = False, **kwargs
):
    """Reference entry point (bee2bee/datasets.py:19-24)."""
    ds = load_dataset(dataset_name, split=split, streaming=streaming, **kwargs)
    return preprocess_examples(
        ds,
        config["tokenizer_name"],
        text_field=config.get("text_field", "text"),
        max_length=config.get("max_length", 128),
        lower_case=config.get("lower_case", False),
    )


def prompts_from_dataset(dataset, text_field: str = "text", limit: Optional[int] = None) -> List[str]:
    """Engine-facing helper: pull raw prompt strings for bench/serve runs."""
    out: List[str] = []
    for i, row in enumerate(dataset):
        if limit is not None and i >= limit:
            break
        out.append(row[text_field])
    return out
