"""VRL (Vector Remap Language) interpreter — row-wise event remapping.

Mirrors reference crates/arkflow-plugin/src/processor/vrl.rs (769 LoC),
which embeds the vrl crate: programs transform one event (a JSON-like
object) at a time with type-preserving conversion both directions. This is
a from-scratch interpreter for the VRL core:

  - paths: ``.a``, ``.a.b``, ``.items[0]``; nested assignment creates
    objects; ``del(.x)`` removes fields; bare names are local variables
  - literals: strings/ints/floats/bools/null/arrays/objects
  - operators: ``+ - * / %``, comparisons, ``&& || !``, and error/null
    coalescing ``??``
  - control flow: ``if cond { } else if { } else { }``; blocks yield their
    last expression
  - fallible functions: ``to_int!(.x)`` aborts the event on error,
    ``to_int(.x) ?? 0`` catches; uncaught errors follow the processor's
    ``on_error`` policy (keep original event / skip event / fail batch)
  - stdlib: type conversion, string ops, parse_json/encode_json,
    exists/is_* predicates, math, length/contains/split/join/...

The ``vrl`` processor routes flat ``.col = expr`` programs to the columnar
expression engine (tensor ops, expr_proc.py) and everything else here.
"""
from __future__ import annotations

import json
import math
import re
import time
from typing import Any, Dict, List, Optional, Tuple

from ..errors import ConfigError


class VrlError(Exception):
    """Runtime error inside a VRL program (catchable with ``??`` / ``!``)."""


# ------------------------------------------------------------------- lexer
_TOKEN = re.compile(r"""
  (?P<ws>[ \t\r]+)
| (?P<comment>\#[^\n]*)
| (?P<nl>\n)
| (?P<float>\d+\.\d+)
| (?P<int>\d+)
| (?P<str>"(?:\\.|[^"\\])*")
| (?P<op>\?\?|==|!=|<=|>=|&&|\|\||[-+*/%<>=!(){}\[\],.;:])
| (?P<name>[A-Za-z_][A-Za-z0-9_]*(?:!(?!=))?)
""", re.X)

_KEYWORDS = {"if", "else", "true", "false", "null", "del"}


def _lex(src: str) -> List[Tuple[str, str]]:
    toks: List[Tuple[str, str]] = []
    pos = 0
    while pos < len(src):
        m = _TOKEN.match(src, pos)
        if not m:
            raise ConfigError(f"vrl: bad character {src[pos]!r} at {pos}")
        pos = m.end()
        kind = m.lastgroup
        if kind in ("ws", "comment"):
            continue
        val = m.group()
        if kind == "nl":
            toks.append(("nl", "\n"))
        elif kind == "name" and val in _KEYWORDS:
            toks.append((val, val))
        else:
            toks.append((kind, val))
    toks.append(("eof", ""))
    return toks


# ------------------------------------------------------------------ parser
# AST: tuples — ("int",v) ("float",v) ("str",v) ("bool",v) ("null",)
# ("array",[e]) ("object",[(k,e)]) ("path",[segs]) ("var",name)
# ("bin",op,l,r) ("not",e) ("coalesce",l,r) ("call",name,bang,[args])
# ("assign",target,e) ("del",path) ("if",[(cond,block)],else_block)
# ("block",[stmts])
class _Parser:
    def __init__(self, toks: List[Tuple[str, str]]):
        self.toks = toks
        self.i = 0

    def peek(self) -> Tuple[str, str]:
        return self.toks[self.i]

    def next(self) -> Tuple[str, str]:
        t = self.toks[self.i]
        self.i += 1
        return t

    def skip_nl(self) -> None:
        while self.peek()[0] == "nl" or (self.peek()[0] == "op"
                                         and self.peek()[1] == ";"):
            self.next()

    def expect(self, kind: str, val: Optional[str] = None) -> str:
        k, v = self.next()
        if k != kind or (val is not None and v != val):
            raise ConfigError(f"vrl: expected {val or kind}, got {v!r}")
        return v

    def program(self) -> list:
        stmts = []
        self.skip_nl()
        while self.peek()[0] != "eof":
            stmts.append(self.statement())
            self.skip_nl()
        return stmts

    def statement(self):
        k, v = self.peek()
        if k == "del":
            self.next()
            self.expect("op", "(")
            path = self.path_expr()
            if path[0] != "path":
                raise ConfigError("vrl: del() needs a field path")
            self.expect("op", ")")
            return ("del", path[1])
        if k == "if":
            return self.if_stmt()
        # assignment or bare expression
        start = self.i
        if k == "op" and v == ".":
            target = self.path_expr()
            if self.peek() == ("op", "="):
                self.next()
                return ("assign", target, self.expr())
            self.i = start
        elif k == "name" and not v.endswith("!"):
            nxt = self.toks[self.i + 1]
            if nxt == ("op", "="):
                self.next()
                self.next()
                return ("assign", ("var", v), self.expr())
        return self.expr()

    def if_stmt(self):
        arms = []
        self.expect("if")
        cond = self.expr()
        arms.append((cond, self.block()))
        else_block = None
        while self.peek()[0] == "else":
            self.next()
            if self.peek()[0] == "if":
                self.next()
                arms.append((self.expr(), self.block()))
            else:
                else_block = self.block()
                break
        return ("if", arms, else_block)

    def block(self):
        self.expect("op", "{")
        stmts = []
        self.skip_nl()
        while self.peek() != ("op", "}"):
            stmts.append(self.statement())
            self.skip_nl()
        self.next()
        return ("block", stmts)

    # expression precedence: ?? < || < && < cmp < addsub < muldiv < unary
    def expr(self):
        e = self.or_expr()
        while self.peek() == ("op", "??"):
            self.next()
            e = ("coalesce", e, self.or_expr())
        return e

    def or_expr(self):
        e = self.and_expr()
        while self.peek() == ("op", "||"):
            self.next()
            e = ("bin", "||", e, self.and_expr())
        return e

    def and_expr(self):
        e = self.cmp_expr()
        while self.peek() == ("op", "&&"):
            self.next()
            e = ("bin", "&&", e, self.cmp_expr())
        return e

    def cmp_expr(self):
        e = self.add_expr()
        while self.peek()[0] == "op" and self.peek()[1] in (
                "==", "!=", "<", "<=", ">", ">="):
            op = self.next()[1]
            e = ("bin", op, e, self.add_expr())
        return e

    def add_expr(self):
        e = self.mul_expr()
        while self.peek()[0] == "op" and self.peek()[1] in "+-":
            op = self.next()[1]
            e = ("bin", op, e, self.mul_expr())
        return e

    def mul_expr(self):
        e = self.unary()
        while self.peek()[0] == "op" and self.peek()[1] in ("*", "/", "%"):
            op = self.next()[1]
            e = ("bin", op, e, self.unary())
        return e

    def unary(self):
        k, v = self.peek()
        if (k, v) == ("op", "!"):
            self.next()
            return ("not", self.unary())
        if (k, v) == ("op", "-"):
            self.next()
            return ("bin", "-", ("int", 0), self.unary())
        return self.postfix()

    def postfix(self):
        k, v = self.next()
        if k == "int":
            return ("int", int(v))
        if k == "float":
            return ("float", float(v))
        if k == "str":
            return ("str", json.loads(v))
        if k == "true":
            return ("bool", True)
        if k == "false":
            return ("bool", False)
        if k == "null":
            return ("null",)
        if (k, v) == ("op", "("):
            e = self.expr()
            self.expect("op", ")")
            return e
        if (k, v) == ("op", "["):
            items = []
            while self.peek() != ("op", "]"):
                items.append(self.expr())
                if self.peek() == ("op", ","):
                    self.next()
            self.next()
            return ("array", items)
        if (k, v) == ("op", "{"):
            pairs = []
            self.skip_nl()
            while self.peek() != ("op", "}"):
                kk, kv = self.next()
                if kk == "str":
                    key = json.loads(kv)
                elif kk == "name":
                    key = kv
                else:
                    raise ConfigError("vrl: bad object key")
                self.expect("op", ":")
                pairs.append((key, self.expr()))
                if self.peek() == ("op", ","):
                    self.next()
                self.skip_nl()
            self.next()
            return ("object", pairs)
        if (k, v) == ("op", "."):
            self.i -= 1
            return self.path_expr()
        if k == "name":
            if self.peek() == ("op", "("):
                self.next()
                args = []
                while self.peek() != ("op", ")"):
                    args.append(self.expr())
                    if self.peek() == ("op", ","):
                        self.next()
                self.next()
                bang = v.endswith("!")
                return ("call", v.rstrip("!"), bang, args)
            if v.endswith("!"):
                raise ConfigError(f"vrl: {v} must be called")
            return ("var", v)
        raise ConfigError(f"vrl: unexpected token {v!r}")

    def path_expr(self):
        segs = []
        while self.peek() == ("op", "."):
            self.next()
            k, v = self.next()
            if k not in ("name", "str"):
                raise ConfigError("vrl: bad path segment")
            segs.append(json.loads(v) if k == "str" else v)
            while self.peek() == ("op", "["):
                self.next()
                idx = self.next()
                if idx[0] != "int":
                    raise ConfigError("vrl: path index must be an int")
                segs.append(int(idx[1]))
                self.expect("op", "]")
        return ("path", segs)


def parse_vrl(src: str) -> list:
    return _Parser(_lex(src)).program()


# --------------------------------------------------------------- evaluator
def _num(v):
    if isinstance(v, bool) or not isinstance(v, (int, float)):
        raise VrlError(f"expected number, got {type(v).__name__}")
    return v


_STDLIB = {}


def _fn(name):
    def deco(f):
        _STDLIB[name] = f
        return f
    return deco


@_fn("to_int")
def _to_int(v):
    if isinstance(v, bool):
        return int(v)
    try:
        return int(v)
    except (TypeError, ValueError) as e:
        raise VrlError(f"to_int: {e}")


@_fn("to_float")
def _to_float(v):
    try:
        return float(v)
    except (TypeError, ValueError) as e:
        raise VrlError(f"to_float: {e}")


@_fn("to_string")
def _to_string(v):
    if v is None:
        return ""
    if isinstance(v, bool):
        return "true" if v else "false"
    if isinstance(v, (dict, list)):
        return json.dumps(v, separators=(",", ":"))
    return str(v)


@_fn("to_bool")
def _to_bool(v):
    if isinstance(v, bool):
        return v
    if isinstance(v, (int, float)):
        return v != 0
    if isinstance(v, str):
        if v.lower() in ("true", "yes", "1"):
            return True
        if v.lower() in ("false", "no", "0", ""):
            return False
    raise VrlError(f"to_bool: cannot coerce {v!r}")


_STDLIB.update({
    "upcase": lambda s: _s(s).upper(),
    "downcase": lambda s: _s(s).lower(),
    "trim": lambda s: _s(s).strip(),
    "length": lambda v: len(v) if isinstance(v, (str, list, dict))
    else _err("length: not a collection"),
    "contains": lambda s, sub: _s(sub) in _s(s),
    "starts_with": lambda s, p: _s(s).startswith(_s(p)),
    "ends_with": lambda s, p: _s(s).endswith(_s(p)),
    "split": lambda s, sep: _s(s).split(_s(sep)),
    "join": lambda xs, sep="": _s(sep).join(_to_string(x) for x in xs),
    "replace": lambda s, old, new: _s(s).replace(_s(old), _s(new)),
    "slice": lambda v, start, end=None: v[int(start):
                                          None if end is None else int(end)],
    "abs": lambda v: abs(_num(v)),
    "round": lambda v, p=0: round(_num(v), int(p)),
    "floor": lambda v: math.floor(_num(v)),
    "ceil": lambda v: math.ceil(_num(v)),
    "merge": lambda a, b: {**a, **b} if isinstance(a, dict)
    and isinstance(b, dict) else _err("merge: need objects"),
    "encode_json": lambda v: json.dumps(v, separators=(",", ":")),
    "now": lambda: time.time(),
    # --- hashing / encoding (vector stdlib: md5, sha1, sha2, base64) ---
    "md5": lambda s: __import__("hashlib").md5(
        _s(s).encode()).hexdigest(),
    "sha1": lambda s: __import__("hashlib").sha1(
        _s(s).encode()).hexdigest(),
    "sha2": lambda s: __import__("hashlib").sha256(
        _s(s).encode()).hexdigest(),
    "sha256": lambda s: __import__("hashlib").sha256(
        _s(s).encode()).hexdigest(),
    "encode_base64": lambda s: __import__("base64").b64encode(
        s if isinstance(s, (bytes, bytearray)) else _s(s).encode()
    ).decode(),
    "decode_base64": lambda s: __import__("base64").b64decode(
        _s(s)).decode("utf-8", "replace"),
    "uuid_v4": lambda: str(__import__("uuid").uuid4()),
    # --- numbers / strings ---
    "parse_int": lambda s, base=10: int(_s(s), int(base)),
    "truncate": lambda s, n, suffix=False: (
        _s(s) if len(_s(s)) <= int(n)
        else _s(s)[: int(n)] + ("..." if suffix else "")),
    "strip_whitespace": lambda s: _s(s).strip(),
    # --- timestamps (float unix-seconds representation, like now()) ---
    "to_unix_timestamp": lambda t: float(_num(t)),
    "from_unix_timestamp": lambda t: float(_num(t)),
    "format_timestamp": lambda t, fmt="%Y-%m-%dT%H:%M:%SZ": time.strftime(
        fmt.replace("%f", "{us:06d}").format(
            us=int((float(_num(t)) % 1) * 1e6))
        if "%f" in fmt else fmt, time.gmtime(float(_num(t)))),
    "parse_timestamp": lambda s, fmt="%Y-%m-%dT%H:%M:%SZ":
        __import__("calendar").timegm(time.strptime(_s(s), fmt)),
    # --- regex ---
    "match": lambda s, pat: __import__("re").search(_s(pat), _s(s))
    is not None,
    "parse_regex": lambda s, pat: (
        lambda m: m.groupdict() if m and m.groupdict()
        else (dict(enumerate(m.groups(), 1)) and
              {str(i): g for i, g in enumerate(m.groups(), 1)}) if m
        else _err("parse_regex: no match"))(
        __import__("re").search(_s(pat), _s(s))),
    "is_null": lambda v: v is None,
    "is_string": lambda v: isinstance(v, str),
    "is_int": lambda v: isinstance(v, int) and not isinstance(v, bool),
    "is_float": lambda v: isinstance(v, float),
    "is_bool": lambda v: isinstance(v, bool),
    "is_object": lambda v: isinstance(v, dict),
    "is_array": lambda v: isinstance(v, list),
    "string": lambda v: v if isinstance(v, str)
    else _err("string: not a string"),
    "int": lambda v: v if isinstance(v, int) and not isinstance(v, bool)
    else _err("int: not an int"),
    "float": lambda v: v if isinstance(v, float)
    else _err("float: not a float"),
})


def _s(v):
    if not isinstance(v, str):
        raise VrlError(f"expected string, got {type(v).__name__}")
    return v


def _err(msg):
    raise VrlError(msg)


@_fn("parse_json")
def _parse_json(v):
    try:
        return json.loads(v)
    except (TypeError, ValueError) as e:
        raise VrlError(f"parse_json: {e}")


class VrlProgram:
    """Compiled program; ``remap(event) -> event`` runs it on one row."""

    def __init__(self, source: str):
        self.stmts = parse_vrl(source)

    # ---- paths ---------------------------------------------------------
    @staticmethod
    def _get_path(event, segs):
        cur = event
        for s in segs:
            if isinstance(s, int):
                if not isinstance(cur, list) or s >= len(cur):
                    raise VrlError(f"path index {s} out of range")
                cur = cur[s]
            else:
                if not isinstance(cur, dict) or s not in cur:
                    raise VrlError(f"field .{s} does not exist")
                cur = cur[s]
        return cur

    @staticmethod
    def _set_path(event, segs, value):
        cur = event
        for i, s in enumerate(segs[:-1]):
            if isinstance(s, int):
                while isinstance(cur, list) and len(cur) <= s:
                    cur.append(None)
                if cur[s] is None:
                    cur[s] = [] if isinstance(segs[i + 1], int) else {}
                cur = cur[s]
            else:
                if not isinstance(cur.get(s), (dict, list)):
                    cur[s] = [] if isinstance(segs[i + 1], int) else {}
                cur = cur[s]
        last = segs[-1]
        if isinstance(last, int):
            while isinstance(cur, list) and len(cur) <= last:
                cur.append(None)
            cur[last] = value
        else:
            cur[last] = value

    # ---- eval ----------------------------------------------------------
    def _eval(self, node, event, scope):
        t = node[0]
        if t in ("int", "float", "str", "bool"):
            return node[1]
        if t == "null":
            return None
        if t == "array":
            return [self._eval(e, event, scope) for e in node[1]]
        if t == "object":
            return {k: self._eval(e, event, scope) for k, e in node[1]}
        if t == "path":
            return self._get_path(event, node[1])
        if t == "var":
            if node[1] not in scope:
                raise VrlError(f"undefined variable {node[1]}")
            return scope[node[1]]
        if t == "not":
            return not _truthy(self._eval(node[1], event, scope))
        if t == "coalesce":
            try:
                v = self._eval(node[1], event, scope)
                return v if v is not None else self._eval(node[2], event,
                                                          scope)
            except VrlError:
                return self._eval(node[2], event, scope)
        if t == "bin":
            return self._bin(node[1], node[2], node[3], event, scope)
        if t == "call":
            name, bang, args = node[1], node[2], node[3]
            fn = _STDLIB.get(name)
            if fn is None:
                raise ConfigError(f"vrl: unknown function {name}")
            vals = [self._eval(a, event, scope) for a in args]
            # `f!(x)` aborts the event on error (uncatchable by ??);
            # plain `f(x)` raises VrlError, catchable with ??
            if bang:
                try:
                    return fn(*vals)
                except VrlError as e:
                    raise VrlAbort(str(e))
            return fn(*vals)
        if t == "block":
            out = None
            for s in node[1]:
                out = self._exec(s, event, scope)
            return out
        if t == "if":
            for cond, blk in node[1]:
                if _truthy(self._eval(cond, event, scope)):
                    return self._eval(blk, event, scope)
            if node[2] is not None:
                return self._eval(node[2], event, scope)
            return None
        raise ConfigError(f"vrl: bad node {t}")

    def _bin(self, op, ln, rn, event, scope):
        if op == "&&":
            return _truthy(self._eval(ln, event, scope)) and \
                _truthy(self._eval(rn, event, scope))
        if op == "||":
            return _truthy(self._eval(ln, event, scope)) or \
                _truthy(self._eval(rn, event, scope))
        lv = self._eval(ln, event, scope)
        rv = self._eval(rn, event, scope)
        if op == "==":
            return lv == rv
        if op == "!=":
            return lv != rv
        if op == "+":
            if isinstance(lv, str) and isinstance(rv, str):
                return lv + rv
            if isinstance(lv, list) and isinstance(rv, list):
                return lv + rv
            return _num(lv) + _num(rv)
        if op == "-":
            return _num(lv) - _num(rv)
        if op == "*":
            return _num(lv) * _num(rv)
        if op == "/":
            if _num(rv) == 0:
                raise VrlError("division by zero")
            return _num(lv) / rv
        if op == "%":
            if _num(rv) == 0:
                raise VrlError("mod by zero")
            return _num(lv) % rv
        try:
            if op == "<":
                return lv < rv
            if op == "<=":
                return lv <= rv
            if op == ">":
                return lv > rv
            if op == ">=":
                return lv >= rv
        except TypeError as e:
            raise VrlError(f"bad comparison: {e}")
        raise ConfigError(f"vrl: bad operator {op}")

    def _exec(self, stmt, event, scope):
        t = stmt[0]
        if t == "assign":
            v = self._eval(stmt[2], event, scope)
            target = stmt[1]
            if target[0] == "path":
                self._set_path(event, target[1], v)
            else:
                scope[target[1]] = v
            return v
        if t == "del":
            segs = stmt[1]
            try:
                parent = self._get_path(event, segs[:-1]) if len(segs) > 1 \
                    else event
                if isinstance(parent, dict):
                    return parent.pop(segs[-1], None)
            except VrlError:
                return None
            return None
        if t == "if":
            return self._eval(stmt, event, scope)
        return self._eval(stmt, event, scope)

    def remap(self, event: Dict[str, Any]) -> Dict[str, Any]:
        scope: Dict[str, Any] = {}
        for stmt in self.stmts:
            self._exec(stmt, event, scope)
        return event


class VrlAbort(Exception):
    """`fn!()` failed — the event aborts (reference: abort on `!` errors)."""


def _truthy(v) -> bool:
    if isinstance(v, bool):
        return v
    raise VrlError(f"condition must be a boolean, got {type(v).__name__}")
