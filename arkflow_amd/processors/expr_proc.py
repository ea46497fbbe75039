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
"""
from __future__ import annotations

from typing import Dict, List

from ..batch import Column, MessageBatch
from ..errors import ConfigError
from ..registry import register
from ..spi import Processor
from ..sql.eval import Env, as_tensor, eval_expr
from ..sql.parser import parse_sql


class ExprProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        assignments = config.get("assignments") or {}
        if not assignments and not config.get("drop"):
            raise ConfigError("expr processor requires 'assignments' or 'drop'")
        self.assignments = [
            (name, parse_sql(f"SELECT {expr}").projections[0][0])
            for name, expr in assignments.items()
        ]
        self.drop = list(config.get("drop") or [])

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
