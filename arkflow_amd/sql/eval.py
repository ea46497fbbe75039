"""Columnar expression evaluator.

Evaluates parsed SQL expressions over an environment of named columns.
Numeric/bool expressions are torch tensors (device-resident on GPU — each
operator is a kernel on the batch's device); string/binary values stay
:class:`~arkflow_amd.batch.Column` (binary) with vectorized compare paths.

Replaces DataFusion's PhysicalExpr evaluation (reference expr/mod.rs:29-210
cached-PhysicalExpr + processor/sql.rs execution).
"""
from __future__ import annotations

from typing import Dict, Optional, Union

import numpy as np
import torch

from ..batch import Column
from .parser import (
    AGGREGATE_FUNCS,
    Between,
    BinaryOp,
    Case,
    Cast,
    ColumnRef,
    FuncCall,
    InList,
    IsNull,
    Like,
    Literal,
    SqlError,
    Star,
    UnaryOp,
)

Value = Union[torch.Tensor, Column, int, float, str, bool, None]


class Env:
    """Column environment: unqualified + qualified names, plus precomputed
    aggregate results keyed by expression text."""

    def __init__(self, columns: Dict[str, Column], n_rows: int,
                 device: torch.device,
                 agg_results: Optional[Dict[str, torch.Tensor]] = None):
        self.columns = columns
        self.n_rows = n_rows
        self.device = device
        self.agg_results = agg_results or {}

    def lookup(self, ref: ColumnRef) -> Column:
        if ref.table:
            qual = f"{ref.table}.{ref.name}"
            if qual in self.columns:
                return self.columns[qual]
        if ref.name in self.columns:
            return self.columns[ref.name]
        raise SqlError(f"unknown column {ref.table + '.' if ref.table else ''}"
                       f"{ref.name}; have {sorted(self.columns)}")


def expr_name(e) -> str:
    """Readable output-column name for an unaliased projection."""
    if isinstance(e, ColumnRef):
        return e.name
    if isinstance(e, Literal):
        return repr(e.value)
    if isinstance(e, FuncCall):
        args = ", ".join(expr_name(a) for a in e.args)
        base = f"{e.name}({'DISTINCT ' if e.distinct else ''}{args})"
        if getattr(e, "filter", None) is not None:
            base += f" filter({expr_name(e.filter)})"
        if e.over is not None:
            part = ", ".join(expr_name(p) for p in e.over.partition_by)
            ob = ", ".join(expr_name(o) + ("" if asc else " desc")
                           for o, asc in e.over.order_by)
            base += f" over(partition by {part} order by {ob})"
        return base
    if isinstance(e, Star):
        return "*"
    if isinstance(e, BinaryOp):
        return f"{expr_name(e.left)} {e.op} {expr_name(e.right)}"
    if isinstance(e, UnaryOp):
        return f"{e.op} {expr_name(e.operand)}"
    if isinstance(e, Cast):
        return f"cast({expr_name(e.expr)} as {e.to_type})"
    if isinstance(e, Case):
        return "case"
    if isinstance(e, Between):
        return f"{expr_name(e.expr)} between"
    if isinstance(e, InList):
        return f"{expr_name(e.expr)} in"
    if isinstance(e, IsNull):
        return f"{expr_name(e.expr)} is null"
    if isinstance(e, Like):
        return f"{expr_name(e.expr)} like"
    return "expr"


def _is_agg_name(name: str) -> bool:
    if name in AGGREGATE_FUNCS:
        return True
    from .udf import aggregate_udf
    return aggregate_udf(name) is not None


def contains_aggregate(e) -> bool:
    if isinstance(e, FuncCall):
        if e.over is not None:
            return False  # window functions are row-level, not GROUP BY
        if _is_agg_name(e.name):
            return True
        return any(contains_aggregate(a) for a in e.args)
    if isinstance(e, BinaryOp):
        return contains_aggregate(e.left) or contains_aggregate(e.right)
    if isinstance(e, UnaryOp):
        return contains_aggregate(e.operand)
    if isinstance(e, Cast):
        return contains_aggregate(e.expr)
    if isinstance(e, Case):
        return any(contains_aggregate(c) or contains_aggregate(v)
                   for c, v in e.whens) or (
            e.else_ is not None and contains_aggregate(e.else_))
    if isinstance(e, (Between,)):
        return any(contains_aggregate(x) for x in (e.expr, e.low, e.high))
    if isinstance(e, InList):
        return contains_aggregate(e.expr)
    if isinstance(e, (IsNull, Like)):
        return contains_aggregate(e.expr)
    return False


def collect_aggregates(e, out: list) -> None:
    """Find aggregate FuncCall nodes (outermost) in an expression tree."""
    if isinstance(e, FuncCall):
        if e.over is not None:
            return  # window functions handled separately
        if _is_agg_name(e.name):
            out.append(e)
            return
        for a in e.args:
            collect_aggregates(a, out)
        return
    if isinstance(e, BinaryOp):
        collect_aggregates(e.left, out)
        collect_aggregates(e.right, out)
    elif isinstance(e, UnaryOp):
        collect_aggregates(e.operand, out)
    elif isinstance(e, Cast):
        collect_aggregates(e.expr, out)
    elif isinstance(e, Case):
        for c, v in e.whens:
            collect_aggregates(c, out)
            collect_aggregates(v, out)
        if e.else_ is not None:
            collect_aggregates(e.else_, out)
    elif isinstance(e, Between):
        for x in (e.expr, e.low, e.high):
            collect_aggregates(x, out)
    elif isinstance(e, (InList, IsNull, Like)):
        collect_aggregates(e.expr, out)


# ------------------------------------------------------------------ broadcast
def as_tensor(v: Value, env: Env) -> torch.Tensor:
    if isinstance(v, torch.Tensor):
        return v
    if isinstance(v, Column):
        if v.kind == "numeric":
            return v.data
        raise SqlError("binary column used in numeric context")
    if v is None:
        return torch.full((env.n_rows,), float("nan"), device=env.device)
    if isinstance(v, bool):
        return torch.full((env.n_rows,), v, dtype=torch.bool, device=env.device)
    if isinstance(v, int):
        return torch.full((env.n_rows,), v, dtype=torch.int64, device=env.device)
    if isinstance(v, float):
        return torch.full((env.n_rows,), v, dtype=torch.float64,
                          device=env.device)
    raise SqlError(f"cannot use {type(v).__name__} in numeric context")


def _binary_bytes(col: Column) -> list:
    return col.to_pylist()


def _bin_eq_literal(col: Column, lit: bytes, device) -> torch.Tensor:
    """Vectorized bytes == literal: length check + content gather compare."""
    lengths = col.offsets[1:] - col.offsets[:-1]
    L = len(lit)
    mask = lengths == L
    if L == 0 or not bool(mask.any()):
        return mask
    cand = torch.nonzero(mask).flatten()
    starts = col.offsets[:-1][cand]
    pos = starts.unsqueeze(1) + torch.arange(L, device=col.data.device)
    litt = torch.from_numpy(
        np.frombuffer(lit, dtype=np.uint8).copy()).to(col.data.device)
    eq = (col.data[pos] == litt).all(dim=1)
    out = torch.zeros_like(mask)
    out[cand] = eq
    return out


def _bin_eq_col(a: Column, b: Column) -> torch.Tensor:
    la = a.offsets[1:] - a.offsets[:-1]
    lb = b.offsets[1:] - b.offsets[:-1]
    mask = la == lb
    if not bool(mask.any()):
        return mask
    # fallback per-row compare for candidates (vectorize later if hot)
    av, bv = a.to_pylist(), b.to_pylist()
    res = [x == y for x, y in zip(av, bv)]
    return torch.tensor(res, dtype=torch.bool, device=a.data.device)


def eval_expr(e, env: Env) -> Value:
    if isinstance(e, Literal):
        return e.value
    if isinstance(e, ColumnRef):
        col = env.lookup(e)
        return col if col.kind == "binary" else col.data
    if isinstance(e, Star):
        raise SqlError("* only allowed in COUNT(*) or projection list")
    if isinstance(e, FuncCall):
        if e.over is not None or _is_agg_name(e.name):
            key = expr_name(e)
            if key in env.agg_results:
                return env.agg_results[key]
            if e.over is not None:
                raise SqlError(f"window fn {key} not computed")
            raise SqlError(f"aggregate {key} outside aggregate context")
        return _eval_func(e, env)
    if isinstance(e, UnaryOp):
        if e.op == "not":
            v = as_tensor(eval_expr(e.operand, env), env)
            return ~v.bool()
        v = as_tensor(eval_expr(e.operand, env), env)
        return -v
    if isinstance(e, BinaryOp):
        return _eval_binop(e, env)
    if isinstance(e, Cast):
        v = eval_expr(e.expr, env)
        return _eval_cast(v, e.to_type, env)
    if isinstance(e, Case):
        branch_vals = [eval_expr(v, env) for _, v in e.whens]
        else_v = eval_expr(e.else_, env) if e.else_ is not None else None
        if any(isinstance(v, str) or
               (isinstance(v, Column) and v.kind == "binary")
               for v in list(branch_vals) + [else_v]):
            # string-valued CASE: host row loop (sqlite/DataFusion parity)
            def _rows(v):
                if v is None:
                    return [None] * env.n_rows
                if isinstance(v, str):
                    return [v.encode()] * env.n_rows
                if isinstance(v, Column) and v.kind == "binary":
                    return v.to_pylist()
                raise SqlError("CASE mixes string and numeric branches")
            out_rows = _rows(else_v)
            for (cond, _), v in zip(reversed(list(e.whens)),
                                    reversed(branch_vals)):
                c = as_tensor(eval_expr(cond, env),
                              env).bool().cpu().tolist()
                vr = _rows(v)
                out_rows = [b if m else a
                            for a, b, m in zip(out_rows, vr, c)]
            out = Column.from_bytes([x if x is not None else b""
                                     for x in out_rows])
            if any(x is None for x in out_rows):
                out = Column(out.kind, out.data, out.offsets,
                             torch.tensor([x is not None for x in out_rows],
                                          dtype=torch.bool,
                                          device=env.device))
            return out
        result = None
        for cond, val in reversed(list(e.whens)):
            c = as_tensor(eval_expr(cond, env), env).bool()
            v = as_tensor(eval_expr(val, env), env)
            if result is None:
                if e.else_ is not None:
                    result = as_tensor(eval_expr(e.else_, env), env)
                else:
                    result = torch.full(
                        (env.n_rows,), float("nan"), device=env.device)
            result, v = _promote(result, v)
            result = torch.where(c, v, result)
        return result
    if isinstance(e, Between):
        ev = eval_expr(e.expr, env)
        if isinstance(ev, Column) and ev.kind == "binary":
            lo_v = eval_expr(e.low, env)
            hi_v = eval_expr(e.high, env)
            if not isinstance(lo_v, str) or not isinstance(hi_v, str):
                raise SqlError("string BETWEEN needs string bounds")
            lob, hib = lo_v.encode(), hi_v.encode()
            rows = ev.to_pylist()
            r = torch.tensor([x is not None and lob <= x <= hib
                              for x in rows], dtype=torch.bool,
                             device=env.device)
            return ~r if e.negated else r
        v = as_tensor(ev, env)
        lo = as_tensor(eval_expr(e.low, env), env)
        hi = as_tensor(eval_expr(e.high, env), env)
        r = (v >= lo) & (v <= hi)
        return ~r if e.negated else r
    if isinstance(e, InList):
        v = eval_expr(e.expr, env)
        if isinstance(v, Column) and v.kind == "binary":
            r = torch.zeros(env.n_rows, dtype=torch.bool, device=env.device)
            for item in e.items:
                lit = eval_expr(item, env)
                if isinstance(lit, str):
                    lit = lit.encode()
                r |= _bin_eq_literal(v, lit, env.device)
        else:
            t = as_tensor(v, env)
            r = torch.zeros_like(t, dtype=torch.bool)
            for item in e.items:
                r |= (t == as_tensor(eval_expr(item, env), env))
        return ~r if e.negated else r
    if isinstance(e, IsNull):
        # a ColumnRef's validity lives on the Column (numeric eval returns
        # the raw tensor) — consult it directly
        if isinstance(e.expr, ColumnRef):
            try:
                col = env.lookup(e.expr)
            except SqlError:
                col = None
            if isinstance(col, Column) and col.validity is not None:
                r = ~col.validity
                return ~r if e.negated else r
        v = eval_expr(e.expr, env)
        if isinstance(v, Column) and v.validity is not None:
            r = ~v.validity
        elif isinstance(v, torch.Tensor) and v.is_floating_point():
            r = torch.isnan(v)
        else:
            r = torch.zeros(env.n_rows, dtype=torch.bool, device=env.device)
        val = expr_validity(e.expr, env)  # strict + div-by-zero bits
        if val is not None:
            r = r | ~val
        return ~r if e.negated else r
    if isinstance(e, Like):
        v = eval_expr(e.expr, env)
        if not isinstance(v, Column) or v.kind != "binary":
            raise SqlError("LIKE requires a string column")
        pat = e.pattern
        if v.data.is_cuda and len(pat) <= 66 and "_" not in pat \
                and "%" not in pat.strip("%"):
            from .. import ops
            body = pat.strip("%").encode()
            if pat.startswith("%") and pat.endswith("%") and len(pat) > 1:
                mode = 0
            elif pat.endswith("%") and not pat.startswith("%"):
                mode = 1
            elif pat.startswith("%") and not pat.endswith("%"):
                mode = 2
            else:
                mode = 3
            # pybind converts Python bytes → std::string raw (no re-encode)
            r = ops.require_native().bytes_match(
                v.data.contiguous(), v.offsets.contiguous(), body, mode)
            return ~r if e.negated else r
        vals = v.to_pylist()
        if pat.startswith("%") and pat.endswith("%") and len(pat) > 1:
            needle = pat[1:-1].encode()
            res = [needle in x for x in vals]
        elif pat.endswith("%"):
            p = pat[:-1].encode()
            res = [x.startswith(p) for x in vals]
        elif pat.startswith("%"):
            p = pat[1:].encode()
            res = [x.endswith(p) for x in vals]
        else:
            res = [x == pat.encode() for x in vals]
        r = torch.tensor(res, dtype=torch.bool, device=env.device)
        return ~r if e.negated else r
    raise SqlError(f"cannot evaluate {type(e).__name__}")


def _promote(a: torch.Tensor, b: torch.Tensor):
    dt = torch.promote_types(a.dtype, b.dtype)
    return a.to(dt), b.to(dt)


def _eval_binop(e: BinaryOp, env: Env) -> Value:
    if e.op in ("and", "or"):
        l = as_tensor(eval_expr(e.left, env), env).bool()
        r = as_tensor(eval_expr(e.right, env), env).bool()
        return (l & r) if e.op == "and" else (l | r)
    lv = eval_expr(e.left, env)
    rv = eval_expr(e.right, env)
    # string comparison / concat paths
    l_bin = isinstance(lv, Column) and lv.kind == "binary"
    r_bin = isinstance(rv, Column) and rv.kind == "binary"
    if e.op == "||":
        def _strs(v, is_bin):
            if is_bin:
                return v.to_strlist()
            if isinstance(v, torch.Tensor):  # numeric column → per-row str
                return [str(int(x)) if float(x).is_integer() else str(x)
                        for x in v.cpu().tolist()]
            return [str(v)] * env.n_rows
        if l_bin or r_bin or isinstance(lv, (str, torch.Tensor))                 or isinstance(rv, (str, torch.Tensor)):
            ls = _strs(lv, l_bin)
            rs = _strs(rv, r_bin)
            return Column.from_strings([a + b for a, b in zip(ls, rs)])
        raise SqlError("|| requires strings")
    if l_bin or r_bin or isinstance(lv, str) or isinstance(rv, str):
        if e.op not in ("=", "!="):
            # ordered string compare: CPU fallback
            ls = lv.to_pylist() if l_bin else [
                (lv.encode() if isinstance(lv, str) else lv)] * env.n_rows
            rs = rv.to_pylist() if r_bin else [
                (rv.encode() if isinstance(rv, str) else rv)] * env.n_rows
            fn = {"<": lambda a, b: a < b, "<=": lambda a, b: a <= b,
                  ">": lambda a, b: a > b, ">=": lambda a, b: a >= b}[e.op]
            return torch.tensor([fn(a, b) for a, b in zip(ls, rs)],
                                dtype=torch.bool, device=env.device)
        if l_bin and isinstance(rv, str):
            r = _bin_eq_literal(lv, rv.encode(), env.device)
        elif r_bin and isinstance(lv, str):
            r = _bin_eq_literal(rv, lv.encode(), env.device)
        elif l_bin and r_bin:
            r = _bin_eq_col(lv, rv)
        else:
            raise SqlError("string compared with non-string")
        return ~r if e.op == "!=" else r
    l = as_tensor(lv, env)
    r = as_tensor(rv, env)
    if e.op in ("=", "!=", "<", "<=", ">", ">="):
        l, r = _promote(l, r)
        return {
            "=": l == r, "!=": l != r, "<": l < r,
            "<=": l <= r, ">": l > r, ">=": l >= r,
        }[e.op]
    l, r = _promote(l, r)
    if e.op == "+":
        return l + r
    if e.op == "-":
        return l - r
    if e.op == "*":
        return l * r
    if e.op == "/":
        # zero divisors are NULL (expr_validity masks them); substitute 1
        # so the kernel never faults
        rz = torch.where(r == 0, torch.ones_like(r), r)
        if l.dtype.is_floating_point or r.dtype.is_floating_point:
            return l / rz
        return torch.div(l, rz, rounding_mode="trunc")
    if e.op == "%":
        # SQL %: truncated remainder (sign of the dividend — sqlite/DataFusion)
        rz = torch.where(r == 0, torch.ones_like(r), r)
        return torch.fmod(l, rz)
    raise SqlError(f"unknown operator {e.op}")


_CAST_TYPES = {
    "int": torch.int64, "integer": torch.int64, "bigint": torch.int64,
    "smallint": torch.int16, "tinyint": torch.int8,
    "float": torch.float32, "real": torch.float32,
    "double": torch.float64, "bool": torch.bool, "boolean": torch.bool,
    "bf16": torch.bfloat16, "bfloat16": torch.bfloat16,
}


def _eval_cast(v: Value, to_type: str, env: Env) -> Value:
    ty = to_type.lower()
    if ty in ("varchar", "text", "string"):
        if isinstance(v, Column) and v.kind == "binary":
            return v
        if isinstance(v, str):
            return v
        if isinstance(v, bool):
            return "1" if v else "0"
        if isinstance(v, (int, float)):
            return str(v)
        t = as_tensor(v, env)
        vals = t.detach().cpu().tolist()
        return Column.from_strings([
            str(int(x)) if isinstance(x, float) and x.is_integer()
            and not t.dtype.is_floating_point else str(x) for x in vals
        ])
    if ty not in _CAST_TYPES:
        raise SqlError(f"unknown cast type {to_type!r}")
    dt = _CAST_TYPES[ty]
    if isinstance(v, str):  # constant: sqlite-style best-effort numeric
        try:
            f = float(v)
        except ValueError:
            f = 0.0
        return f if dt.is_floating_point else int(f)
    if isinstance(v, Column) and v.kind == "binary":
        vals = v.to_strlist()
        if dt.is_floating_point:
            return torch.tensor([float(x) for x in vals], dtype=dt,
                                device=env.device)
        return torch.tensor([int(float(x)) for x in vals], dtype=torch.int64,
                            device=env.device).to(dt)
    t = as_tensor(v, env)
    if not dt.is_floating_point and t.dtype.is_floating_point:
        t = t.trunc()
    return t.to(dt)


_UNARY_MATH = {
    "abs": torch.abs, "sqrt": torch.sqrt, "floor": torch.floor,
    "ceil": torch.ceil, "exp": torch.exp, "ln": torch.log,
    "log10": torch.log10, "log2": torch.log2, "sin": torch.sin,
    "cos": torch.cos, "tan": torch.tan, "sign": torch.sign,
}


def _fmt_num(x) -> str:
    if isinstance(x, bool):
        return "true" if x else "false"
    if isinstance(x, float) and x == int(x) and abs(x) < 1e15:
        return str(int(x))
    return str(x)


def _eval_func(e: FuncCall, env: Env) -> Value:
    name = e.name
    if name in _UNARY_MATH:
        v = as_tensor(eval_expr(e.args[0], env), env)
        return _UNARY_MATH[name](v.double() if not v.dtype.is_floating_point
                                 else v)
    if name == "round":
        v = as_tensor(eval_expr(e.args[0], env), env).double()
        digits = 0
        if len(e.args) > 1:
            digits = int(eval_expr(e.args[1], env))
        return torch.round(v, decimals=digits)
    if name in ("pow", "power"):
        b = as_tensor(eval_expr(e.args[0], env), env).double()
        p = as_tensor(eval_expr(e.args[1], env), env).double()
        return torch.pow(b, p)
    if name in ("substr", "substring"):
        v = eval_expr(e.args[0], env)
        def _si(x):
            return int(x.item()) if isinstance(x, torch.Tensor) else int(x)
        start = _si(eval_expr(e.args[1], env))
        ln = _si(eval_expr(e.args[2], env)) if len(e.args) > 2 else None
        def _sub(b):
            if b is None:
                return None
            # SQL/sqlite 1-based start; negative counts from the end
            if start > 0:
                i = start - 1
            elif start < 0:
                i = max(len(b) + start, 0)
            else:
                i = 0
            return b[i:] if ln is None else b[i:i + max(ln, 0)]
        if isinstance(v, Column) and v.kind == "binary":
            vals = [_sub(x) for x in v.to_pylist()]
            out = Column.from_bytes([x if x is not None else b""
                                     for x in vals])
            if any(x is None for x in vals):
                out = Column(out.kind, out.data, out.offsets,
                             torch.tensor([x is not None for x in vals],
                                          dtype=torch.bool,
                                          device=env.device))
            return out
        if isinstance(v, str):
            return _sub(v.encode()).decode("utf-8", "replace")
        raise SqlError("substr() requires a string")
    if name == "coalesce":
        args_v = [eval_expr(a, env) for a in e.args]
        if any(isinstance(v, Column) and v.kind == "binary"
               for v in args_v) or \
                all(isinstance(v, (str, type(None))) for v in args_v):
            # string coalesce: first non-NULL per row (host fallback)
            rows = None
            for v in args_v:
                if isinstance(v, Column) and v.kind == "binary":
                    cur = v.to_pylist()
                elif isinstance(v, str):
                    cur = [v.encode()] * env.n_rows
                elif v is None:
                    cur = [None] * env.n_rows
                else:
                    raise SqlError("coalesce: mixed string/numeric args")
                rows = cur if rows is None else [
                    a if a is not None else b for a, b in zip(rows, cur)]
            out = Column.from_bytes([x if x is not None else b""
                                     for x in rows])
            if any(x is None for x in rows):
                out = Column(out.kind, out.data, out.offsets,
                             torch.tensor([x is not None for x in rows],
                                          dtype=torch.bool,
                                          device=env.device))
            return out
        result = None
        missing = None  # rows still NULL so far
        for a in e.args:
            validity = None
            if isinstance(a, ColumnRef):
                try:
                    c = env.lookup(a)
                    if isinstance(c, Column):
                        validity = c.validity
                except SqlError:
                    pass
            v = as_tensor(eval_expr(a, env), env)
            miss_here = torch.isnan(v) if v.is_floating_point() \
                else torch.zeros(v.shape, dtype=torch.bool, device=v.device)
            if validity is not None:
                miss_here = miss_here | ~validity
            if result is None:
                result, missing = v, miss_here
            else:
                result, v = _promote(result, v)
                result = torch.where(missing, v, result)
                missing = missing & miss_here
        return result
    if name in ("to_int", "to_float", "to_string"):
        # VRL-style conversions (processor/expr_proc.py translate_vrl)
        v = eval_expr(e.args[0], env)
        if name == "to_string":
            if isinstance(v, Column) and v.kind == "binary":
                return v
            t = as_tensor(v, env)
            return Column.from_strings([_fmt_num(x) for x in t.tolist()])
        if isinstance(v, Column) and v.kind == "binary":
            vals = v.to_strlist()
            if name == "to_int":
                return torch.tensor([int(float(x)) if x else 0 for x in vals],
                                    dtype=torch.int64, device=env.device)
            return torch.tensor([float(x) if x else 0.0 for x in vals],
                                dtype=torch.float64, device=env.device)
        t = as_tensor(v, env)
        if name == "to_int":
            return t.trunc().to(torch.int64) if t.dtype.is_floating_point \
                else t.to(torch.int64)
        return t.to(torch.float64)
    if name in ("contains", "starts_with", "ends_with"):
        v = eval_expr(e.args[0], env)
        pat = eval_expr(e.args[1], env)
        if not (isinstance(v, Column) and v.kind == "binary") \
                or not isinstance(pat, str):
            raise SqlError(f"{name}(column, 'literal') expects a string "
                           "column and literal")
        mode = {"contains": 0, "starts_with": 1, "ends_with": 2}[name]
        if v.data.is_cuda and len(pat.encode()) <= 64:
            from .. import ops
            r = ops.require_native().bytes_match(
                v.data.contiguous(), v.offsets.contiguous(),
                pat.encode(), mode)
        else:
            f = {"contains": (lambda s, p: p in s),
                 "starts_with": bytes.startswith,
                 "ends_with": bytes.endswith}[name]
            pb = pat.encode()
            r = torch.tensor([bool(f(x, pb)) for x in v.to_pylist()],
                             dtype=torch.bool, device=env.device)
        if isinstance(v, Column) and v.validity is not None:
            r = r & v.validity
        return r
    if name in ("length", "char_length", "octet_length"):
        v = eval_expr(e.args[0], env)
        if isinstance(v, Column) and v.kind == "binary":
            return (v.offsets[1:] - v.offsets[:-1]).to(torch.int64)
        raise SqlError("length() requires a string column")
    if name in ("upper", "lower"):
        v = eval_expr(e.args[0], env)
        if isinstance(v, Column) and v.kind == "binary":
            vals = v.to_strlist()
            f = str.upper if name == "upper" else str.lower
            return Column.from_strings([f(x) for x in vals])
        if isinstance(v, str):
            return v.upper() if name == "upper" else v.lower()
        raise SqlError(f"{name}() requires a string")
    if name == "replace":
        v = eval_expr(e.args[0], env)
        old_s = eval_expr(e.args[1], env)
        new_s = eval_expr(e.args[2], env)
        if not isinstance(old_s, str) or not isinstance(new_s, str):
            raise SqlError("replace() needs constant string args")
        if isinstance(v, Column) and v.kind == "binary":
            vals = [None if x is None else
                    x.replace(old_s.encode(), new_s.encode())
                    for x in v.to_pylist()]
            out = Column.from_bytes([x if x is not None else b""
                                     for x in vals])
            if any(x is None for x in vals):
                out = Column(out.kind, out.data, out.offsets,
                             torch.tensor([x is not None for x in vals],
                                          dtype=torch.bool,
                                          device=env.device))
            return out
        if isinstance(v, str):
            return v.replace(old_s, new_s)
        raise SqlError("replace() requires a string")
    if name == "concat":
        parts = [eval_expr(a, env) for a in e.args]
        lists = []
        for p in parts:
            if isinstance(p, Column) and p.kind == "binary":
                lists.append(p.to_strlist())
            else:
                lists.append([str(p)] * env.n_rows)
        return Column.from_strings(["".join(t) for t in zip(*lists)])
    if name == "nullif":
        av = eval_expr(e.args[0], env)
        bv = eval_expr(e.args[1], env)
        if (isinstance(av, Column) and av.kind == "binary") \
                or isinstance(av, str):
            rows = av.to_pylist() if isinstance(av, Column) \
                else [av.encode()] * env.n_rows
            if isinstance(bv, str):
                cmp = [bv.encode()] * env.n_rows
            elif isinstance(bv, Column) and bv.kind == "binary":
                cmp = bv.to_pylist()
            else:
                raise SqlError("nullif: mixed string/numeric args")
            vals = [None if a2 == b2 else a2 for a2, b2 in zip(rows, cmp)]
            out = Column.from_bytes([x if x is not None else b""
                                     for x in vals])
            if any(x is None for x in vals):
                out = Column(out.kind, out.data, out.offsets,
                             torch.tensor([x is not None for x in vals],
                                          dtype=torch.bool,
                                          device=env.device))
            return out
        a = as_tensor(av, env).double()
        b = as_tensor(bv, env).double()
        return torch.where(a == b, torch.full_like(a, float("nan")), a)
    from .udf import scalar_udf
    udf = scalar_udf(name)
    if udf is not None:
        args = [eval_expr(a, env) for a in e.args]
        args = [as_tensor(a, env) if not isinstance(a, Column) else a
                for a in args]
        return udf(*args)
    raise SqlError(f"unknown function {name}()")


def collect_window_calls(e, out: list) -> None:
    """Find FuncCall nodes with an OVER clause."""
    if isinstance(e, FuncCall):
        if e.over is not None:
            out.append(e)
            return
        for a in e.args:
            collect_window_calls(a, out)
        return
    if isinstance(e, BinaryOp):
        collect_window_calls(e.left, out)
        collect_window_calls(e.right, out)
    elif isinstance(e, UnaryOp):
        collect_window_calls(e.operand, out)
    elif isinstance(e, Cast):
        collect_window_calls(e.expr, out)
    elif isinstance(e, Case):
        for c, v in e.whens:
            collect_window_calls(c, out)
            collect_window_calls(v, out)
        if e.else_ is not None:
            collect_window_calls(e.else_, out)


def expr_validity(e, env: Env):
    """Combined validity of the columns an expression touches (strict-NULL
    propagation for elementwise expressions): None = all valid. Expressions
    containing null-AWARE constructs (IS NULL, coalesce, CASE) handle
    validity themselves and return None here. Division/modulo by zero is
    NULL (sqlite/DataFusion), so those divisors add value-dependent bits."""
    cols: list = []
    divs: list = []
    if _collect_strict_refs(e, cols, divs) is False:
        return None
    v = None
    for ref in cols:
        try:
            c = env.lookup(ref)
        except SqlError:
            continue
        if isinstance(c, Column) and c.validity is not None:
            v = c.validity if v is None else (v & c.validity)
    for d in divs:
        dv = eval_expr(d, env)
        if isinstance(dv, torch.Tensor):
            nz = dv != 0
        elif isinstance(dv, (int, float)):
            if dv != 0:
                continue
            nz = torch.zeros(env.n_rows, dtype=torch.bool,
                             device=env.device)
        else:
            continue
        v = nz if v is None else (v & nz)
    return v


def _collect_strict_refs(e, out: list, divs: list = None):
    """Gather ColumnRefs (and /,% divisor exprs into *divs*); returns False
    if the expr contains a null-aware construct (the caller must not apply
    strict propagation)."""
    if divs is None:
        divs = []
    if isinstance(e, ColumnRef):
        out.append(e)
        return True
    if isinstance(e, (IsNull, Case)):
        return False
    if isinstance(e, FuncCall):
        if e.name in ("coalesce", "ifnull", "count"):
            return False
        return all(_collect_strict_refs(a, out, divs) is not False
                   for a in e.args)
    if isinstance(e, BinaryOp):
        if e.op in ("/", "%"):
            divs.append(e.right)
        return (_collect_strict_refs(e.left, out, divs) is not False
                and _collect_strict_refs(e.right, out, divs) is not False)
    if isinstance(e, UnaryOp):
        return _collect_strict_refs(e.operand, out, divs)
    if isinstance(e, Cast):
        return _collect_strict_refs(e.expr, out)
    if isinstance(e, (InList, Between, Like)):
        inner = getattr(e, "expr", None)
        if inner is not None:
            return _collect_strict_refs(inner, out)
    return True
