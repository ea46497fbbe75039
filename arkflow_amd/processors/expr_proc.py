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


# `vrl` alias so reference configs with a vrl processor map onto the columnar
# expression engine (VRL programs must be rewritten as assignments).
@register("processor", "vrl",
          description="Alias of `expr` — columnar VRL-equivalent remapping",
          example={"type": "vrl", "assignments": {"v2": "value * 2"}})
def _build_vrl(config: dict, resource=None) -> ExprProcessor:
    return ExprProcessor(config, resource)
