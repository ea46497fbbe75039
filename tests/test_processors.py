"""Processor/codec/component-library tests (reference arkflow-plugin inline
tests: processor/json.rs, protobuf.rs, batch.rs, python.rs, vrl.rs round-trip,
codec tests, temporary joins)."""
import asyncio
import json

import pytest
import torch

from arkflow_amd.batch import DEFAULT_BINARY_VALUE_FIELD, MessageBatch


def test_json_to_arrow_roundtrip(run):
    from arkflow_amd.processors.json_proc import (
        ArrowToJsonProcessor,
        JsonToArrowProcessor,
    )
    payloads = [json.dumps({"a": i, "b": i * 1.5, "s": f"x{i}"}).encode()
                for i in range(100)]
    b = MessageBatch.from_binary(payloads, input_name="t")
    out = run(JsonToArrowProcessor({}, None).process(b))[0]
    assert out.num_rows == 100
    assert out.column("a").to_pylist()[:3] == [0, 1, 2]
    assert out.column("b").to_pylist()[2] == 3.0
    assert out.column("s").to_strlist()[5] == "x5"
    back = run(ArrowToJsonProcessor({}, None).process(out))[0]
    row0 = json.loads(back.binary_values()[0])
    assert row0 == {"a": 0, "b": 0.0, "s": "x0"}


def test_json_projection(run):
    from arkflow_amd.processors.json_proc import JsonToArrowProcessor
    b = MessageBatch.from_binary([b'{"a": 1, "b": 2}'])
    out = run(JsonToArrowProcessor({"columns": ["b"]}, None).process(b))[0]
    assert out.column_names == ["b"]


def test_batch_processor(run):
    from arkflow_amd.processors.batch_proc import BatchProcessor
    p = BatchProcessor({"count": 3, "timeout_ms": 60_000})
    b = MessageBatch.from_dict({"v": [1]})
    assert run(p.process(b)) == []
    assert run(p.process(b)) == []
    out = run(p.process(b))
    assert len(out) == 1 and out[0].num_rows == 3


def test_python_processor_inline_code(run):
    from arkflow_amd.processors.python_proc import PythonProcessor
    p = PythonProcessor({
        "code": "def process(batch):\n"
                "    import torch\n"
                "    from arkflow_amd.batch import Column\n"
                "    return batch.with_columns("
                "{'v2': Column('numeric', batch.column('v').data * 2)})",
    })
    b = MessageBatch.from_dict({"v": [1, 2, 3]})
    out = run(p.process(b))[0]
    assert out.column("v2").to_pylist() == [2, 4, 6]


def test_python_processor_pyarrow(run):
    from arkflow_amd.processors.python_proc import PythonProcessor
    p = PythonProcessor({
        "convert": "pyarrow",
        "code": "def process(table):\n"
                "    return table.select(['v'])",
    })
    b = MessageBatch.from_dict({"v": [1.0, 2.0], "w": [0.0, 0.0]})
    out = run(p.process(b))[0]
    assert out.column_names == ["v"]


def test_expr_processor(run):
    from arkflow_amd.processors.expr_proc import ExprProcessor
    p = ExprProcessor({
        "assignments": {"total": "price * qty",
                        "flag": "CASE WHEN price > 10 THEN 1 ELSE 0 END"},
        "drop": ["qty"],
    })
    b = MessageBatch.from_dict({"price": [5.0, 20.0], "qty": [2, 3]})
    out = run(p.process(b))[0]
    assert out.column("total").to_pylist() == [10.0, 60.0]
    assert out.column("flag").to_pylist() == [0.0, 1.0]
    assert "qty" not in out.columns


PROTO = """
syntax = "proto3";
message Telemetry {
  double temp = 1;
  float press = 2;
  int64 ts = 3;
  sint32 delta = 4;
  bool ok = 5;
  string tag = 6;
  fixed32 fx = 7;
}
"""


def test_proto_wire_roundtrip():
    from arkflow_amd.processors.proto_wire import (
        ProtoSchema,
        decode_message,
        encode_message,
    )
    schema = ProtoSchema.parse(PROTO)
    row = {"temp": 21.5, "press": 1.25, "ts": 171234, "delta": -7,
           "ok": True, "tag": "dev1", "fx": 42}
    buf = encode_message(row, schema)
    back = decode_message(buf, schema)
    assert back["temp"] == 21.5
    assert abs(back["press"] - 1.25) < 1e-6
    assert back["ts"] == 171234
    assert back["delta"] == -7
    assert back["ok"] is True
    assert back["tag"] == "dev1"
    assert back["fx"] == 42
    # absent fields → proto3 defaults
    empty = decode_message(b"", schema)
    assert empty["temp"] == 0.0 and empty["tag"] == ""


def test_protobuf_processors_cpu(run):
    from arkflow_amd.processors.proto_wire import ProtoSchema, encode_message
    from arkflow_amd.processors.protobuf_proc import (
        ArrowToProtobufProcessor,
        ProtobufToArrowProcessor,
    )
    schema = ProtoSchema.parse(PROTO)
    payloads = [
        encode_message({"temp": float(i), "press": 0.5 * i, "ts": i,
                        "delta": -i, "ok": i % 2 == 0, "tag": f"t{i}",
                        "fx": i}, schema)
        for i in range(50)
    ]
    b = MessageBatch.from_binary(payloads)
    p2a = ProtobufToArrowProcessor({"proto": PROTO}, None)
    out = run(p2a.process(b))[0]
    assert out.column("temp").to_pylist()[:3] == [0.0, 1.0, 2.0]
    assert out.column("delta").to_pylist()[3] == -3
    assert out.column("tag").to_strlist()[7] == "t7"
    a2p = ArrowToProtobufProcessor({"proto": PROTO}, None)
    back = run(a2p.process(out))[0]
    assert back.binary_values()[5] == payloads[5]


def test_json_codec():
    from arkflow_amd.codecs.json_codec import JsonCodec
    c = JsonCodec({}, None)
    b = MessageBatch.from_dict({"v": [1, 2], "s": ["a", "b"]})
    enc = c.encode(b)
    dec = c.decode(enc)
    assert dec.column("v").to_pylist() == [1, 2]
    assert dec.column("s").to_strlist() == ["a", "b"]


def test_protobuf_codec():
    from arkflow_amd.codecs.protobuf_codec import ProtobufCodec
    c = ProtobufCodec({"proto": PROTO}, None)
    b = MessageBatch.from_dict({
        "temp": [1.0], "press": [2.0], "ts": [3], "delta": [-1],
        "ok": [True], "tag": ["z"], "fx": [9],
    })
    dec = c.decode(c.encode(b))
    assert dec.column("temp").to_pylist() == [1.0]
    assert dec.column("tag").to_strlist() == ["z"]


def test_file_input_output(tmp_path, run):
    from arkflow_amd.inputs.file import FileInput
    from arkflow_amd.outputs.file import FileOutput
    # write parquet then read back
    out = FileOutput({"path": str(tmp_path / "o.parquet")})
    b = MessageBatch.from_dict({"v": list(range(10)), "s": [f"r{i}" for i in
                                                            range(10)]})
    run(out.connect())
    run(out.write(b))
    run(out.close())
    inp = FileInput({"path": str(tmp_path / "o.parquet"), "batch_size": 4})
    batches = []
    from arkflow_amd.errors import EOFError_
    async def drain():
        while True:
            try:
                batch, _ = await inp.read()
            except EOFError_:
                return
            batches.append(batch)
    run(drain())
    assert sum(x.num_rows for x in batches) == 10
    assert batches[0].column("v").to_pylist() == [0, 1, 2, 3]
    # csv path
    out2 = FileOutput({"path": str(tmp_path / "o.csv")})
    run(out2.connect())
    run(out2.write(b))
    run(out2.close())
    inp2 = FileInput({"path": str(tmp_path / "o.csv")})
    async def one():
        return await inp2.read()
    batch, _ = run(one())
    assert batch.num_rows == 10


def test_file_input_with_query(tmp_path, run):
    from arkflow_amd.inputs.file import FileInput
    import pyarrow as pa
    import pyarrow.parquet as pq
    pq.write_table(pa.table({"v": list(range(100))}),
                   str(tmp_path / "d.parquet"))
    inp = FileInput({"path": str(tmp_path / "d.parquet"),
                     "query": "SELECT v FROM flow WHERE v >= 95"})
    async def one():
        return await inp.read()
    batch, _ = run(one())
    assert batch.column("v").to_pylist() == [95, 96, 97, 98, 99]


def test_http_input_output_loopback(run):
    """HTTP input server + HTTP output client, full loopback."""
    from arkflow_amd.inputs.http import HttpInput
    from arkflow_amd.outputs.http import HttpOutput

    async def main():
        inp = HttpInput({"address": "127.0.0.1:0", "token": "sek"})
        await inp.connect()
        out = HttpOutput({
            "url": f"http://127.0.0.1:{inp.port}/ingest",
            "token": "sek", "raw_value": True,
        })
        await out.connect()
        b = MessageBatch.from_binary([b'{"x": 1}'])
        await out.write(b)
        got, _ = await asyncio.wait_for(inp.read(), 5)
        assert got.binary_values() == [b'{"x": 1}']
        # wrong token rejected
        import aiohttp
        async with aiohttp.ClientSession() as s:
            async with s.post(f"http://127.0.0.1:{inp.port}/ingest",
                              data=b"x") as resp:
                assert resp.status == 401
        await out.close()
        await inp.close()

    run(main())


def test_multiple_inputs_fanin(run):
    from arkflow_amd.inputs.multiple import MultipleInputs
    from arkflow_amd.spi import Resource
    from arkflow_amd.errors import EOFError_
    res = Resource()
    mi = MultipleInputs({
        "inputs": {
            "a": {"type": "generate", "count": 3, "batch_size": 1,
                  "fields": {"v": {"dtype": "float32"}}},
            "b": {"type": "generate", "count": 2, "batch_size": 1,
                  "fields": {"v": {"dtype": "float32"}}},
        },
    }, res)
    assert res.input_names == ["a", "b"]

    async def main():
        await mi.connect()
        names = []
        try:
            while True:
                batch, _ = await asyncio.wait_for(mi.read(), 5)
                names.append(batch.input_name)
        except EOFError_:
            pass
        await mi.close()
        return names

    names = run(main())
    assert sorted(names) == ["a", "a", "a", "b", "b"]


def test_sql_temporary_join(run):
    """SQL processor joining a temporary lookup table
    (reference processor/sql.rs:148-183)."""
    from arkflow_amd.processors.sql import SqlProcessor
    from arkflow_amd.spi import Resource
    from arkflow_amd.temporary.memory_table import MemoryTemporary
    res = Resource()
    temp = MemoryTemporary({"key_column": "uid",
                            "rows": [{"uid": 1, "name": "ann"},
                                     {"uid": 2, "name": "bob"}]})
    res.temporaries = {"users": temp}
    p = SqlProcessor({
        "query": "SELECT flow.uid, users.name FROM flow "
                 "JOIN users ON flow.uid = users.uid ORDER BY flow.uid",
        "temporaries": [{"name": "users", "key": "uid"}],
    }, res)
    b = MessageBatch.from_dict({"uid": [2, 1, 2]})
    out = run(p.process(b))[0]
    assert out.column("uid").to_pylist() == [1, 2, 2]
    assert out.column("name").to_strlist() == ["ann", "bob", "bob"]


def test_schema_registry_codec():
    from arkflow_amd.codecs.schema_registry import SchemaRegistryCodec
    proto = "message M { double v = 1; string tag = 2; }"
    c = SchemaRegistryCodec({"schemas": {"7": proto}, "default_schema_id": 7},
                            None)
    b = MessageBatch.from_dict({"v": [1.5], "tag": ["x"]})
    enc = c.encode(b)
    assert enc[0][0] == 0 and enc[0][1:5] == (7).to_bytes(4, "big")
    dec = c.decode(enc)
    assert dec.column("v").to_pylist() == [1.5]
    assert dec.column("tag").to_strlist() == ["x"]


def test_inference_mlp_cpu(run):
    from arkflow_amd.processors.inference import InferenceProcessor
    p = InferenceProcessor({"model": "mlp_anomaly", "columns": ["a", "b"],
                            "hidden": [32], "device": "cpu"})
    b = MessageBatch.from_dict({"a": [0.1, 0.9], "b": [0.2, 0.8],
                                "k": [1, 2]})
    out = run(p.process(b))[0]
    assert "score" in out.columns
    assert out.num_rows == 2
    assert out.column("k").to_pylist() == [1, 2]  # passthrough cols kept
    # deterministic per seed
    p2 = InferenceProcessor({"model": "mlp_anomaly", "columns": ["a", "b"],
                             "hidden": [32], "device": "cpu"})
    out2 = run(p2.process(b))[0]
    assert out.column("score").to_pylist() == out2.column("score").to_pylist()


def test_inference_bert_cpu_tiny(run):
    from arkflow_amd.processors.inference import InferenceProcessor
    p = InferenceProcessor({"model": "bert_base", "layers": 1,
                            "hidden_size": 64, "heads": 2, "ff": 128,
                            "seq_len": 8, "device": "cpu"})
    b = MessageBatch.from_dict({"token": list(range(16))})
    out = run(p.process(b))[0]
    assert out.num_rows == 2  # 16 tokens / seq_len 8 → 2 sequences
    assert "logit_0" in out.columns and "score" in out.columns


def test_vrl_statement_translation(run):
    """VRL-style source (vrl.rs config shape) translated to columnar ops."""
    from arkflow_amd.processors.expr_proc import ExprProcessor

    async def main():
        proc = ExprProcessor({"statement": """
            .total = .price * .qty       # arithmetic on fields
            .name_up = upcase(.name)
            .price_i = to_int(.price)
            .note = to_string(.qty)
            del(.tmp)
        """})
        b = MessageBatch.from_dict({
            "price": [1.5, 2.0], "qty": [2, 3],
            "name": ["ab", "cd"], "tmp": [0, 0]})
        out = (await proc.process(b))[0]
        assert out.column("total").to_pylist() == [3.0, 6.0]
        assert out.column("name_up").to_strlist() == ["AB", "CD"]
        assert out.column("price_i").to_pylist() == [1, 2]
        assert out.column("note").to_strlist() == ["2", "3"]
        assert "tmp" not in out.columns

    run(main())


def test_vrl_statement_coalesce_and_errors():
    from arkflow_amd.errors import ConfigError
    from arkflow_amd.processors.expr_proc import ExprProcessor, translate_vrl
    a, _ = translate_vrl(".v = .maybe ?? 0")
    assert a == [("v", "coalesce(maybe, 0)")]
    with pytest.raises(ConfigError):
        ExprProcessor({"statement": "if .a > 1 { .b = 2 }"})


def test_json_schema_host_fallback_nested(run):
    """Fixed-schema host decode produces the same columns (incl. dotted
    nested paths and validity) as the GPU kernel path."""
    import json as _json
    from arkflow_amd.processors.json_proc import JsonToArrowProcessor

    async def main():
        docs = [{"region": "eu", "user": {"tier": 2}, "amount": 1.5},
                {"region": 7, "amount": 2.0},  # wrong type + missing nested
                {"user": {"tier": 0}, "amount": 3.0}]
        b = MessageBatch.from_binary([_json.dumps(d).encode() for d in docs])
        proc = JsonToArrowProcessor({"schema": {
            "region": "str", "user.tier": "int", "amount": "float"}})
        out = (await proc.process(b))[0]
        region = [x.decode() if isinstance(x, bytes) else x
                  for x in out.column("region").to_pylist()]
        assert region == ["eu", None, None]
        assert out.column("user.tier").to_pylist() == [2, None, 0]
        assert out.column("amount").to_pylist() == [1.5, 2.0, 3.0]

    run(main())


def test_json_inferred_schema_matches_fixed(run):
    """Schemaless json_to_arrow infers from the FIRST record (reference
    component/json.rs:27) and takes the same typed decode path as a
    configured schema — including one level of nesting and validity for
    fields missing in later records."""
    import json as _json

    from arkflow_amd.batch import MessageBatch
    from arkflow_amd.processors.json_proc import JsonToArrowProcessor

    docs = [
        {"a": 1, "b": 0.5, "ok": True, "s": "x", "u": {"id": 7}},
        {"a": 2, "b": 1.5, "ok": False, "s": "yy", "u": {"id": 8}},
        {"a": 3, "b": 2.5, "s": "z", "u": {"id": 9}},  # ok missing
    ]
    payloads = [_json.dumps(d).encode() for d in docs]
    batch = MessageBatch.from_binary(payloads)
    proc = JsonToArrowProcessor({}, None)
    out = run(proc.process(batch))[0]
    assert proc._inferred == {"a": "int", "b": "float", "ok": "bool",
                              "s": "str", "u.id": "int"}
    assert out.column("a").to_pylist() == [1, 2, 3]
    assert out.column("u.id").to_pylist() == [7, 8, 9]
    assert out.column("s").to_strlist() == ["x", "yy", "z"]
    assert out.column("ok").validity.tolist() == [True, True, False]
    # unsupported first record (array value) → host pyarrow path
    proc2 = JsonToArrowProcessor({}, None)
    b2 = MessageBatch.from_binary(
        [_json.dumps({"xs": [1, 2]}).encode()])
    out2 = run(proc2.process(b2))[0]
    assert proc2._inferred is False
    assert out2.num_rows == 1
