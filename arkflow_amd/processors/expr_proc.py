"""`vrl`-analog processor: columnar expression remapping.

The reference embeds Vector Remap Language for per-row transforms
(crates/arkflow-plugin/src/processor/vrl.rs). An interpreted row-wise DSL is
exactly what a GPU engine must avoid, so the MI355X-native equivalent is a
columnar expression processor: each assignment `col = <sql-expr>` is compiled
once and evaluated as device-wide tensor ops per batch — the expression subset
of VRL at kernel speed. (Arbitrary row-wise Python remains available via the
`python` processor.)

Config:
  assignments: {out_col: "expr", ...}   # evaluated left-to-right
  drop: [cols...]                       # columns to remove afterwards
  statement: ".x = .a + 1\ndel(.b)"     # VRL-style source, translated to the
                                        # two forms above (vrl.rs configs port
                                        # without rewriting)
"""
from __future__ import annotations

from typing import Dict, List

from ..batch import Column, MessageBatch
from ..errors import ConfigError
from ..registry import register
from ..spi import Processor
from ..sql.eval import Env, as_tensor, eval_expr
from ..sql.parser import parse_sql

import re

_PATH = re.compile(r"\.([A-Za-z_][A-Za-z0-9_]*)")
_DEL = re.compile(r"^del\(\s*\.([A-Za-z_][A-Za-z0-9_]*)\s*\)$")
_ASSIGN = re.compile(r"^\.([A-Za-z_][A-Za-z0-9_]*)\s*=\s*(.+)$", re.S)
_FN_MAP = {"upcase": "upper", "downcase": "lower", "to_int": "to_int",
           "to_float": "to_float", "to_string": "to_string"}


def _translate_vrl_expr(rhs: str) -> str:
    """VRL expression → SQL expression: `.field` → field, `??` → coalesce,
    `to_int(x)` → CAST, upcase/downcase → upper/lower, `!=`/`==` kept."""
    rhs = rhs.strip()
    # a ?? b (right-assoc, lowest precedence) → coalesce(a, b)
    if "??" in rhs:
        parts = [p.strip() for p in rhs.split("??")]
        rhs = "coalesce(" + ", ".join(parts) + ")"
    rhs = _PATH.sub(r"\1", rhs)
    rhs = re.sub(r"\bupcase\(", "upper(", rhs)
    rhs = re.sub(r"\bdowncase\(", "lower(", rhs)
    rhs = rhs.replace("==", "=")
    return rhs


def translate_vrl(source: str):
    """Translate a VRL-subset program into (assignments, drops).
    Supported: `.out = <expr>`, `del(.field)`, `#` comments, `;`/newline
    statement separators, flat field paths. Unsupported VRL (nested paths,
    control flow, error coalescing on fallible calls) raises ConfigError."""
    assignments, drops = [], []
    cleaned = "\n".join(line.split("#", 1)[0]
                        for line in source.splitlines())
    for stmt in re.split(r"[;\n]", cleaned):
        stmt = stmt.strip()
        if not stmt:
            continue
        m = _DEL.match(stmt)
        if m:
            drops.append(m.group(1))
            continue
        m = _ASSIGN.match(stmt)
        if m:
            assignments.append((m.group(1), _translate_vrl_expr(m.group(2))))
            continue
        raise ConfigError(
            f"vrl: unsupported statement {stmt!r} (supported subset: "
            "'.col = expr', 'del(.col)')")
    return assignments, drops


class ExprProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        assignments = list((config.get("assignments") or {}).items())
        drops = list(config.get("drop") or [])
        source = config.get("statement") or config.get("source")
        if source:
            a2, d2 = translate_vrl(source)
            assignments.extend(a2)
            drops.extend(d2)
        if not assignments and not drops:
            raise ConfigError("expr processor requires 'assignments', 'drop' "
                              "or a VRL 'statement'")
        self.assignments = [
            (name, parse_sql(f"SELECT {expr}").projections[0][0])
            for name, expr in assignments
        ]
        self.drop = drops

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows == 0:
            return []
        cols: Dict[str, Column] = dict(batch.columns)
        for name, expr in self.assignments:
            env = Env(cols, batch.num_rows, batch.device)
            v = eval_expr(expr, env)
            if isinstance(v, Column):
                cols[name] = v
            else:
                cols[name] = Column("numeric", as_tensor(v, env))
        for d in self.drop:
            cols.pop(d, None)
        return [MessageBatch(cols, batch.input_name)]


@register("processor", "expr",
          description="Columnar expression remap (VRL-analog): "
                      "assignments of SQL expressions to columns",
          example={"type": "expr",
                   "assignments": {"total": "price * quantity"}})
def _build_expr(config: dict, resource=None) -> ExprProcessor:
    return ExprProcessor(config, resource)


class VrlProcessor(Processor):
    """Real VRL programs, row-wise (reference processor/vrl.rs semantics:
    per-event remap with type-preserving conversion both directions).

    ``on_error``: what to do when an uncaught error / `fn!()` abort hits a
    row — ``keep`` (default: emit the original row unchanged), ``skip``
    (drop the row), ``fail`` (fail the batch → error_output path).
    """

    def __init__(self, config: dict, resource=None):
        from .vrl_lang import VrlProgram
        source = config.get("statement") or config.get("source") \
            or config.get("program")
        if not source:
            raise ConfigError("vrl processor requires 'source'")
        self.program = VrlProgram(source)
        self.on_error = config.get("on_error", "keep")
        if self.on_error not in ("keep", "skip", "fail"):
            raise ConfigError("vrl on_error must be keep|skip|fail")

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        from ..errors import ProcessError
        from .vrl_lang import VrlAbort, VrlError
        if batch.num_rows == 0:
            return []
        rows = batch.to_rows()
        out_rows = []
        for row in rows:
            # bytes → str for string columns (VRL values are JSON-like)
            ev = {k: (v.decode("utf-8", "replace")
                      if isinstance(v, (bytes, bytearray)) else v)
                  for k, v in row.items()}
            try:
                out_rows.append(self.program.remap(dict(ev)))
            except (VrlError, VrlAbort) as e:
                if self.on_error == "fail":
                    raise ProcessError(f"vrl: {e}") from e
                if self.on_error == "keep":
                    out_rows.append(ev)
        if not out_rows:
            return []
        return [_rows_to_batch(out_rows, batch.input_name, batch.device)]


def _rows_to_batch(rows: List[dict], input_name, device) -> MessageBatch:
    """Union of keys → typed columns; nested values JSON-encode; missing
    values carry a validity mask (type-preserving conversion back)."""
    import json as _json

    import torch

    names: List[str] = []
    for r in rows:
        for k in r:
            if k not in names:
                names.append(k)
    cols: Dict[str, Column] = {}
    for name in names:
        vals = [r.get(name) for r in rows]
        present = [v is not None for v in vals]
        non_null = [v for v in vals if v is not None]
        if non_null and all(isinstance(v, bool) for v in non_null):
            c = Column("numeric", torch.tensor(
                [bool(v) if v is not None else False for v in vals]))
        elif non_null and all(isinstance(v, int) and not isinstance(v, bool)
                              for v in non_null):
            c = Column("numeric", torch.tensor(
                [int(v) if v is not None else 0 for v in vals],
                dtype=torch.int64))
        elif non_null and all(isinstance(v, (int, float))
                              and not isinstance(v, bool)
                              for v in non_null):
            c = Column("numeric", torch.tensor(
                [float(v) if v is not None else 0.0 for v in vals],
                dtype=torch.float64))
        elif non_null and all(isinstance(v, str) for v in non_null):
            c = Column.from_strings([v if v is not None else ""
                                     for v in vals])
        else:  # mixed / nested → JSON encoding
            c = Column.from_strings(
                [_json.dumps(v, separators=(",", ":"))
                 if v is not None else "" for v in vals])
        if not all(present):
            c.validity = torch.tensor(present)
        cols[name] = c
    b = MessageBatch(cols, input_name)
    if device is not None and device.type != "cpu":
        b = b.to(device)
    return b


@register("processor", "vrl",
          description="VRL remap: flat '.col = expr' programs run columnar "
                      "(tensor ops); full VRL (nested paths, if/else, "
                      "fallible functions, ??) runs the row-wise interpreter",
          example={"type": "vrl", "source": '.v2 = .value * 2'})
def _build_vrl(config: dict, resource=None):
    source = config.get("statement") or config.get("source") \
        or config.get("program")
    if config.get("assignments") or config.get("drop"):
        return ExprProcessor(config, resource)
    if source:
        try:
            # flat assignment/del subset → columnar fast path (any parse
            # failure — unsupported statement shape, VRL-only functions —
            # falls through to the interpreter)
            probe = dict(config)
            probe["statement"] = source
            return ExprProcessor(probe, resource)
        except Exception:  # noqa: BLE001
            pass
    return VrlProcessor(config, resource)
