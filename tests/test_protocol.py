import numpy as np
import pytest
import torch

from polyrl_amd.protocol import TensorBatch


def make_batch(n=8, L=4):
    return TensorBatch.from_dict(
        tensors={"x": torch.arange(n * L).reshape(n, L).float(),
                 "y": torch.arange(n)},
        non_tensors={"uid": np.array([f"u{i}" for i in range(n)], dtype=object)},
        meta_info={"tag": 1},
    )


def test_len_and_keys():
    b = make_batch()
    assert len(b) == 8
    assert set(b.keys()) == {"x", "y", "uid"}
    assert "x" in b and "uid" in b


def test_slice_and_getitem():
    b = make_batch()
    s = b[2:5]
    assert len(s) == 3
    assert torch.equal(s["y"], torch.tensor([2, 3, 4]))
    assert list(s["uid"]) == ["u2", "u3", "u4"]


def test_split_concat_roundtrip():
    b = make_batch()
    parts = b.split(3)
    assert [len(p) for p in parts] == [3, 3, 2]
    r = TensorBatch.concat(parts)
    assert torch.equal(r["x"], b["x"])
    assert list(r["uid"]) == list(b["uid"])


def test_chunk_requires_divisibility():
    b = make_batch()
    with pytest.raises(AssertionError):
        b.chunk(3)
    parts = b.chunk(4)
    assert all(len(p) == 2 for p in parts)


def test_union_and_select_pop():
    b = make_batch()
    other = TensorBatch.from_dict(tensors={"z": torch.ones(8)})
    b.union(other)
    assert "z" in b
    sel = b.select(tensor_keys=["x"], non_tensor_keys=[])
    assert set(sel.keys()) == {"x"}
    popped = b.pop(tensor_keys=["z"])
    assert "z" not in b and "z" in popped


def test_repeat_interleave():
    b = make_batch(n=2)
    r = b.repeat(3)
    assert len(r) == 6
    assert list(r["uid"]) == ["u0", "u0", "u0", "u1", "u1", "u1"]
    assert torch.equal(r["y"], torch.tensor([0, 0, 0, 1, 1, 1]))


def test_pad_unpad():
    b = make_batch(n=6)
    p = b.pad_to_divisor(4)
    assert len(p) == 8 and p.meta_info["_pad_size"] == 2
    u = p.unpad()
    assert len(u) == 6
    assert torch.equal(u["y"], b["y"])


def test_index_tensor_slice():
    b = make_batch()
    idx = torch.tensor([5, 1, 3])
    s = b.slice(idx)
    assert torch.equal(s["y"], torch.tensor([5, 1, 3]))
    assert list(s["uid"]) == ["u5", "u1", "u3"]


def test_consistency_check():
    with pytest.raises(AssertionError):
        TensorBatch.from_dict(tensors={"a": torch.zeros(3), "b": torch.zeros(4)})
