"""Data-model tests (mirrors reference lib.rs inline tests for MessageBatch,
split_batch, metadata columns)."""
import pytest
import torch

from arkflow_amd.batch import (
    Column,
    DEFAULT_BINARY_VALUE_FIELD,
    MessageBatch,
    concat_batches,
    split_batch,
)


def test_binary_roundtrip():
    payloads = [b"hello", b"", b"world!!", "unicodeé".encode()]
    b = MessageBatch.from_binary(payloads, input_name="t")
    assert b.num_rows == 4
    assert b.binary_values() == payloads
    assert b.input_name == "t"


def test_numeric_columns_and_rows():
    b = MessageBatch.from_dict({
        "a": [1, 2, 3],
        "b": [1.5, 2.5, 3.5],
        "s": ["x", "y", "z"],
    })
    rows = b.to_rows()
    assert rows[1]["a"] == 2
    assert rows[2]["b"] == 3.5
    assert rows[0]["s"] == b"x"


def test_mismatched_length_raises():
    with pytest.raises(ValueError):
        MessageBatch.from_dict({"a": [1, 2], "b": [1]})


def test_slice_and_split():
    b = MessageBatch.from_dict({
        "v": list(range(100)),
        "s": [f"row{i}" for i in range(100)],
    })
    parts = split_batch(b, 30)
    assert [p.num_rows for p in parts] == [30, 30, 30, 10]
    assert parts[1].to_rows()[0]["v"] == 30
    assert parts[3].to_rows()[-1]["s"] == b"row99"


def test_concat_batches():
    b1 = MessageBatch.from_dict({"v": [1, 2], "s": ["a", "b"]})
    b2 = MessageBatch.from_dict({"v": [3], "s": ["c"]})
    out = concat_batches([b1, b2])
    assert out.num_rows == 3
    assert out.column("v").to_pylist() == [1, 2, 3]
    assert out.column("s").to_pylist() == [b"a", b"b", b"c"]


def test_concat_schema_mismatch():
    b1 = MessageBatch.from_dict({"v": [1]})
    b2 = MessageBatch.from_dict({"w": [1]})
    with pytest.raises(ValueError):
        concat_batches([b1, b2])


def test_take_numeric_and_binary():
    b = MessageBatch.from_dict({
        "v": [10, 20, 30, 40],
        "s": ["aa", "b", "cccc", ""],
    })
    idx = torch.tensor([3, 1, 2])
    t = b.take(idx)
    assert t.column("v").to_pylist() == [40, 20, 30]
    assert t.column("s").to_pylist() == [b"", b"b", b"cccc"]


def test_take_empty():
    b = MessageBatch.from_dict({"v": [1, 2], "s": ["a", "bb"]})
    t = b.take(torch.tensor([], dtype=torch.int64))
    assert t.num_rows == 0
    assert t.column("s").to_pylist() == []


def test_json_lines():
    b = MessageBatch.from_dict({"v": [1], "s": ["hi"]})
    lines = b.to_json_lines()
    assert lines == [b'{"v":1,"s":"hi"}']


def test_value_column_convention():
    b = MessageBatch.from_binary([b"x"])
    assert DEFAULT_BINARY_VALUE_FIELD in b.columns


def test_column_take_preserves_device_dtype():
    c = Column.from_numeric(torch.tensor([1.0, 2.0, 3.0], dtype=torch.float32))
    t = c.take(torch.tensor([2, 0]))
    assert t.data.dtype == torch.float32
    assert t.to_pylist() == [3.0, 1.0]
