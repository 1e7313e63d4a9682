import os

from bee2bee_amd.mesh.pieces import (
    bitfield_from_pieces,
    load_pieces,
    piece_hashes,
    save_pieces,
    split_pieces,
    verify_and_reassemble,
)


def test_split_verify_reassemble():
    data = os.urandom(10_000)
    pieces = split_pieces(data, 1024)
    assert len(pieces) == 10
    hashes = piece_hashes(pieces)
    assert verify_and_reassemble(pieces, hashes) == data


def test_corruption_detected():
    import pytest

    data = b"x" * 4096
    pieces = split_pieces(data, 1024)
    hashes = piece_hashes(pieces)
    pieces[2] = b"corrupt" + pieces[2][7:]
    with pytest.raises(ValueError, match="hash_mismatch_at_2"):
        verify_and_reassemble(pieces, hashes)


def test_bitfield():
    assert bitfield_from_pieces(5, [0, 3, 99]) == [1, 0, 0, 1, 0]


def test_persistence_roundtrip(tmp_path):
    data = os.urandom(5000)
    pieces = split_pieces(data, 2048)
    hashes = piece_hashes(pieces)
    save_pieces(str(tmp_path), "deadbeef", pieces)
    loaded = load_pieces(str(tmp_path), "deadbeef")
    assert verify_and_reassemble(loaded, hashes) == data
