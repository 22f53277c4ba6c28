"""Preprocessing tool tests (reference: utils/preprocessing/)."""

import json
import os

import torch

from tools.preprocessing import build_vocab, json_to_packed, tsv_to_json


def test_tsv_json_packed_vocab_roundtrip(tmp_path):
    tsv = tmp_path / "raw.tsv"
    tsv.write_text("alice\thello world\nbob\tfoo bar\nalice\thello again\n")
    blob = tsv_to_json(str(tsv), str(tmp_path / "blob.json"))
    assert blob["users"] == ["alice", "bob"]
    assert blob["num_samples"] == [2, 1]

    out = json_to_packed(str(tmp_path / "blob.json"), str(tmp_path / "b.pt"))
    loaded = torch.load(out, weights_only=False)
    assert loaded["user_data"]["alice"]["x"] == ["hello world", "hello again"]

    vocab = build_vocab(str(tmp_path / "blob.json"),
                        str(tmp_path / "vocab.json"), vocab_size=3)
    assert list(vocab)[0] == "hello"  # most frequent first

    # the packed blob loads through the engine's generic loader
    from msrflute_amd.models.generic_data import load_blob
    assert load_blob(str(tmp_path / "b.pt"))["users"] == ["alice", "bob"]
