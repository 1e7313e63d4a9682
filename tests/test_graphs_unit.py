"""DecodeGraphs bucket logic (pure; capture itself is GPU-tested via the
engine)."""
import torch

from bee2bee_amd.engine.graphs import DecodeGraphs, decode_slot_mapping


class _DummyRunner:
    device = torch.device("cpu")


def _graphs(max_batch):
    return DecodeGraphs(_DummyRunner(), max_batch, max_blocks_per_seq=4)


def test_buckets_cover_power_of_two_and_max():
    g = _graphs(48)
    assert g.buckets() == [1, 2, 4, 8, 16, 32, 48]
    assert g.bucket_for(1) == 1
    assert g.bucket_for(3) == 4
    assert g.bucket_for(33) == 48
    assert g.bucket_for(48) == 48
    assert g.bucket_for(500) == 48  # clamped


def test_buckets_exact_power_of_two():
    g = _graphs(64)
    assert g.buckets()[-1] == 64
    assert g.bucket_for(64) == 64
    assert g.bucket_for(17) == 32


def test_decode_slot_mapping_matches_python():
    bt = torch.tensor([[3, 7, 1], [5, 0, 0]], dtype=torch.int32)
    pos = torch.tensor([33, 2], dtype=torch.int32)
    slots = decode_slot_mapping(bt, pos, 32)
    # seq0: pos 33 -> block idx 1 (=7), offset 1 -> 7*32+1
    # seq1: pos 2  -> block idx 0 (=5), offset 2 -> 5*32+2
    assert slots.tolist() == [7 * 32 + 1, 5 * 32 + 2]
