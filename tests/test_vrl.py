"""VRL interpreter tests — mirrors the reference's vrl.rs inline suite
(processor/vrl.rs:583-765: round-trip, type preservation, nested paths,
fallibility)."""
import asyncio

import pytest

from arkflow_amd.batch import MessageBatch
from arkflow_amd.errors import ProcessError
from arkflow_amd.processors.expr_proc import VrlProcessor, _build_vrl
from arkflow_amd.processors.vrl_lang import VrlAbort, VrlError, VrlProgram


def remap(src, event):
    return VrlProgram(src).remap(dict(event))


def test_basic_assign_and_types():
    ev = remap('.x = .a + 1\n.s = "hi"\n.f = 1.5\n.b = true\n.n = null',
               {"a": 2})
    assert ev == {"a": 2, "x": 3, "s": "hi", "f": 1.5, "b": True, "n": None}
    assert isinstance(ev["x"], int) and not isinstance(ev["x"], bool)


def test_nested_paths_create_objects():
    ev = remap('.user.name = "ada"\n.user.id = 7\n.tags[1] = "b"', {})
    assert ev["user"] == {"name": "ada", "id": 7}
    assert ev["tags"] == [None, "b"]
    assert remap('.id = .user.id', {"user": {"id": 9}})["id"] == 9


def test_del_and_variables():
    ev = remap('tmp = .a * 2\n.b = tmp + 1\ndel(.a)', {"a": 5})
    assert ev == {"b": 11}


def test_string_ops():
    ev = remap('.u = upcase(.s)\n.l = downcase(.s)\n'
               '.parts = split(.s, "-")\n.j = join(split(.s, "-"), "/")\n'
               '.has = contains(.s, "b-")\n.len = length(.s)',
               {"s": "aB-cD"})
    assert ev["u"] == "AB-CD" and ev["l"] == "ab-cd"
    assert ev["parts"] == ["aB", "cD"] and ev["j"] == "aB/cD"
    assert ev["has"] is False and ev["len"] == 5


def test_if_else_and_comparisons():
    src = 'if .v > 10 { .size = "big" } else if .v > 3 { .size = "mid" } ' \
          'else { .size = "small" }'
    assert remap(src, {"v": 20})["size"] == "big"
    assert remap(src, {"v": 5})["size"] == "mid"
    assert remap(src, {"v": 1})["size"] == "small"


def test_error_coalescing_and_fallibility():
    # fallible to_int caught by ??
    assert remap('.x = to_int(.s) ?? -1', {"s": "nope"})["x"] == -1
    assert remap('.x = to_int(.s) ?? -1', {"s": "42"})["x"] == 42
    # missing field caught by ??
    assert remap('.x = .missing ?? "dflt"', {})["x"] == "dflt"
    # uncaught fallible raises VrlError
    with pytest.raises(VrlError):
        remap('.x = to_int(.s)', {"s": "nope"})
    # bang form aborts (uncatchable by ??)
    with pytest.raises(VrlAbort):
        remap('.x = to_int!(.s) ?? 0', {"s": "nope"})


def test_parse_encode_json_roundtrip():
    ev = remap('.doc = parse_json!(.raw)\n.out = encode_json(.doc)\n'
               '.id = .doc.id', {"raw": '{"id": 3, "ok": true}'})
    assert ev["doc"] == {"id": 3, "ok": True}
    assert ev["id"] == 3
    assert '"id":3' in ev["out"]


def test_arith_and_division_semantics():
    ev = remap('.q = .a / .b\n.m = .a % .b\n.neg = -.a', {"a": 7, "b": 2})
    assert ev["q"] == 3.5 and ev["m"] == 1 and ev["neg"] == -7
    with pytest.raises(VrlError):
        remap('.x = .a / 0', {"a": 1})


def test_processor_batch_roundtrip(run):
    """Type-preserving batch→events→batch conversion (vrl.rs:153,358)."""
    proc = VrlProcessor({"source": '''
.score = .a * 2 + 1
.name = upcase(.name)
if .a > 1 { .cls = "hi" } else { .cls = "lo" }
del(.drop_me)
'''})
    batch = MessageBatch.from_dict({
        "a": [1, 2, 3],
        "name": ["x", "y", "z"],
        "drop_me": [9, 9, 9],
    })
    out = run(proc.process(batch))[0]
    assert out.column("score").to_pylist() == [3, 5, 7]
    assert out.column("name").to_strlist() == ["X", "Y", "Z"]
    assert out.column("cls").to_strlist() == ["lo", "hi", "hi"]
    assert "drop_me" not in out.columns


def test_processor_on_error_policies(run):
    batch = MessageBatch.from_dict({"s": ["1", "x", "3"]})
    keep = VrlProcessor({"source": '.v = to_int!(.s)', "on_error": "keep"})
    out = run(keep.process(batch))[0]
    assert out.num_rows == 3  # bad row kept unchanged (no .v)
    assert out.column("v").validity.tolist() == [True, False, True]
    skip = VrlProcessor({"source": '.v = to_int!(.s)', "on_error": "skip"})
    out2 = run(skip.process(batch))[0]
    assert out2.num_rows == 2
    fail = VrlProcessor({"source": '.v = to_int!(.s)', "on_error": "fail"})
    with pytest.raises(ProcessError):
        run(fail.process(batch))


def test_builder_routes_flat_to_columnar_and_rest_to_interpreter():
    from arkflow_amd.processors.expr_proc import ExprProcessor
    flat = _build_vrl({"source": '.v2 = .v * 2\ndel(.old)'})
    assert isinstance(flat, ExprProcessor)
    full = _build_vrl({"source": 'if .a > 1 { .b = "x" }'})
    assert isinstance(full, VrlProcessor)


def test_stdlib_hash_encode_time_regex():
    """Vector-stdlib parity batch 2: hashing, base64, parse_int, truncate,
    timestamps, regex match/capture, uuid (vrl.rs stdlib surface)."""
    import hashlib

    from arkflow_amd.processors.vrl_lang import VrlProgram

    out = VrlProgram('''
.h = sha256(.name)
.m5 = md5(.name)
.b = encode_base64(.name)
.rt = decode_base64(.b)
.n = parse_int("ff", 16)
.t = truncate(.name, 3)
.ts = format_timestamp(1700000000, "%Y-%m-%d")
.back = parse_timestamp("2023-11-14", "%Y-%m-%d")
.m = match(.name, "^al")
.g = parse_regex(.name, "(?P<first>a.)")
.u = uuid_v4()
''').remap({"name": "alice"})
    assert out["h"] == hashlib.sha256(b"alice").hexdigest()
    assert out["m5"] == hashlib.md5(b"alice").hexdigest()
    assert out["rt"] == "alice" and out["n"] == 255 and out["t"] == "ali"
    assert out["ts"] == "2023-11-14" and out["back"] == 1699920000
    assert out["m"] is True and out["g"] == {"first": "al"}
    assert len(out["u"]) == 36


def test_stdlib_parse_regex_abort_on_no_match():
    from arkflow_amd.processors.vrl_lang import VrlError, VrlProgram

    try:
        VrlProgram('.g = parse_regex(.name, "(?P<x>zz)")').remap(
            {"name": "alice"})
        raise AssertionError("expected VrlError")
    except VrlError:
        pass
