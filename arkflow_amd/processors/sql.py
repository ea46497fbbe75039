"""`sql` processor: run a SQL statement per batch against table `flow`.

Mirrors reference crates/arkflow-plugin/src/processor/sql.rs: the statement is
pre-parsed and reused (:189), DDL/DML rejected (:189-192), table name
configurable, temporary-table joins keyed by an evaluated expression
(:148-183). Execution is the engine's own columnar planner on HIP kernels
(GPU) / torch (CPU) instead of DataFusion.
"""
from __future__ import annotations

from typing import List

import torch

from ..batch import MessageBatch
from ..registry import register
from ..spi import Processor
from ..sql.engine import DEFAULT_TABLE, SqlExecutor
from ..sql.eval import Env, as_tensor, eval_expr
from ..sql.parser import parse_sql


class SqlProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.query = config.get("query")
        if not self.query:
            from ..errors import ConfigError
            raise ConfigError("sql processor requires 'query'")
        self.table_name = config.get("table_name", DEFAULT_TABLE)
        self.executor = SqlExecutor(self.query)
        self.resource = resource
        # temporaries: [{name, key}] — key is a SQL expression evaluated over
        # the batch; its values are passed to Temporary.get
        self.temporary_specs = config.get("temporaries") or []
        self._key_exprs = {
            t["name"]: parse_sql(f"SELECT {t['key']}").projections[0][0]
            for t in self.temporary_specs if "key" in t
        }
        # native one-call fast path for `SELECT * FROM flow WHERE col OP lit`
        self._fast_filter = self._detect_fast_filter()

    def _detect_fast_filter(self):
        from ..sql.parser import BinaryOp, ColumnRef, Literal, Star
        s = self.executor.select
        if (s.joins or s.group_by or s.having or s.order_by or s.distinct
                or s.limit is not None or s.offset or s.union_all
                or self.temporary_specs or s.where is None):
            return None
        if len(s.projections) != 1 or not isinstance(s.projections[0][0],
                                                     Star):
            return None
        w = s.where
        if (isinstance(w, BinaryOp) and w.op in ("<", "<=", ">", ">=", "=",
                                                 "!=")
                and isinstance(w.left, ColumnRef)
                and isinstance(w.right, Literal)
                and isinstance(w.right.value, (int, float))):
            opi = {"<": 0, "<=": 1, ">": 2, ">=": 3, "=": 4, "!=": 5}[w.op]
            return (w.left.name, opi, float(w.right.value))
        return None

    def _try_fast_filter(self, batch: MessageBatch):
        """One C++ call: fused compare+compact+multi-gather (all columns
        numeric, device-resident). Returns None when not applicable."""
        if self._fast_filter is None or batch.device.type != "cuda":
            return None
        col_name, opi, scalar = self._fast_filter
        if col_name not in batch.columns:
            return None
        names, tensors = [], []
        for n, c in batch.columns.items():
            if c.kind != "numeric" or c.validity is not None:
                return None
            names.append(n)
            tensors.append(c.data)
        fcol = batch.column(col_name).data
        if fcol.dtype not in (torch.float32, torch.int64, torch.int32):
            return None
        from .. import ops
        nat = ops.require_native()
        outs, total = nat.fused_filter_gather(
            tensors, names.index(col_name), opi, scalar)
        from ..batch import Column
        return MessageBatch(
            {n: Column("numeric", t) for n, t in zip(names, outs)},
            batch.input_name)

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows == 0:
            return []  # ProcessResult::None (sql.rs:208-210)
        fast = self._try_fast_filter(batch)
        if fast is not None:
            return [fast] if fast.num_rows else []
        tables = {self.table_name: batch}
        if self.table_name != DEFAULT_TABLE:
            tables.setdefault(DEFAULT_TABLE, batch)
        for spec in self.temporary_specs:
            name = spec["name"]
            temp = (self.resource.temporaries.get(name)
                    if self.resource else None)
            if temp is None:
                from ..errors import ProcessError
                raise ProcessError(f"unknown temporary table {name!r}")
            keys = None
            if name in self._key_exprs:
                env = Env(batch.columns, batch.num_rows, batch.device)
                v = eval_expr(self._key_exprs[name], env)
                from ..batch import Column
                if isinstance(v, Column):
                    keys = v.to_pylist()
                else:
                    keys = as_tensor(v, env).detach().cpu().tolist()
            t_batch = await temp.get(keys)
            if t_batch is not None:
                tables[name] = t_batch
        result = self.executor.execute(tables)
        if result.num_rows == 0 and not self.executor.is_aggregate:
            return []
        result.input_name = batch.input_name
        return [result]


@register("processor", "sql",
          description="SQL per batch over table 'flow' (filter/project/"
                      "aggregate/join; HIP kernels on GPU)",
          example={"type": "sql",
                   "query": "SELECT * FROM flow WHERE value >= 10"})
def _build_sql(config: dict, resource=None) -> SqlProcessor:
    return SqlProcessor(config, resource)
