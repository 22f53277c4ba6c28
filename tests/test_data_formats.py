"""Data ingest formats: LEAF/FedML JSON normalization (per-user {x,y}
dicts, missing num_samples) and the gated HDF5 reader (VERDICT round-1
item 8 — real-data ingest behind the blob contract)."""

import json

import numpy as np
import pytest

from msrflute_amd.models.generic_data import (ArrayDataset, load_blob,
                                              load_hdf5_blob)


def _leaf_blob():
    return {
        "users": ["u0", "u1"],
        "user_data": {
            "u0": {"x": [[0.1] * 4, [0.2] * 4], "y": [1, 2]},
            "u1": {"x": [[0.3] * 4], "y": [3]},
        },
    }


def test_leaf_json_normalization(tmp_path):
    p = tmp_path / "leaf.json"
    p.write_text(json.dumps(_leaf_blob()))
    blob = load_blob(str(p))
    assert blob["num_samples"] == [2, 1]
    assert blob["user_data"]["u0"] == [[0.1] * 4, [0.2] * 4]
    assert blob["user_data_label"]["u1"] == [3]
    ds = ArrayDataset(blob, user_idx=1, x_shape=(4,))
    assert len(ds) == 1
    x, y = ds[0]
    assert float(y) == 3


def test_fedml_dict_direct():
    ds = ArrayDataset(_leaf_blob(), user_idx=0, x_shape=(4,))
    assert len(ds) == 2
    assert ds.user_list == ["u0", "u1"]


def test_hdf5_requires_h5py_with_actionable_error(tmp_path):
    try:
        import h5py  # noqa: F401
        have_h5py = True
    except ImportError:
        have_h5py = False
    if not have_h5py:
        p = tmp_path / "d.hdf5"
        p.write_bytes(b"\x89HDF\r\n")
        with pytest.raises(RuntimeError, match="h5py"):
            load_hdf5_blob(str(p))
    else:  # networked machines with h5py: full round-trip
        import h5py
        p = tmp_path / "d.hdf5"
        with h5py.File(p, "w") as f:
            f.create_dataset("users", data=[b"u0", b"u1"])
            f.create_dataset("num_samples", data=[2, 1])
            g = f.create_group("user_data")
            g.create_dataset("u0", data=np.zeros((2, 4)))
            g.create_dataset("u1", data=np.ones((1, 4)))
            gl = f.create_group("user_data_label")
            gl.create_dataset("u0", data=[0, 1])
            gl.create_dataset("u1", data=[2])
        blob = load_hdf5_blob(str(p))
        assert blob["users"] == ["u0", "u1"]
        assert blob["num_samples"] == [2, 1]
