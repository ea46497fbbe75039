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
        # native one-call fused filter→group→agg for
        # `SELECT key, count(*)/sum/min/max(col)… WHERE col OP lit GROUP BY key`
        self._fast_agg = self._detect_fast_agg()

    def _detect_fast_filter(self):
        from ..sql.parser import BinaryOp, ColumnRef, Literal, Star
        s = self.executor.select
        if (s.joins or s.group_by or s.having or s.order_by or s.distinct
                or s.limit is not None or s.offset or s.set_ops
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

    def _detect_fast_agg(self):
        """`SELECT key, AGG…` with a simple scalar WHERE and a single int
        GROUP BY key → ONE native call (fused_filter_agg: filter + gather +
        LDS hash group + segment reductions, two host syncs). The Python-op
        chain for this shape is host-dispatch-bound at batch 8192
        (BASELINE config 2)."""
        from ..sql.parser import BinaryOp, ColumnRef, FuncCall, Literal
        s = self.executor.select
        if (s.joins or s.having or s.order_by or s.distinct
                or s.limit is not None or s.offset or s.set_ops
                or self.temporary_specs):
            return None
        if len(s.group_by) != 1 or not isinstance(s.group_by[0], ColumnRef):
            return None
        key = s.group_by[0].name
        # WHERE col OP literal (optional: no WHERE → keep all rows)
        filt = None
        if s.where is not None:
            w = s.where
            if not (isinstance(w, BinaryOp)
                    and w.op in ("<", "<=", ">", ">=", "=", "!=")
                    and isinstance(w.left, ColumnRef)
                    and isinstance(w.right, Literal)
                    and isinstance(w.right.value, (int, float))):
                return None
            opi = {"<": 0, "<=": 1, ">": 2, ">=": 3, "=": 4, "!=": 5}[w.op]
            filt = (w.left.name, opi, float(w.right.value))
        # projections: the key + count(*)/sum/min/max/avg(plain column)
        plan = []  # (kind, payload, alias)
        for e, alias in s.projections:
            if isinstance(e, ColumnRef) and e.name == key:
                plan.append(("key", None, alias or key))
                continue
            if isinstance(e, FuncCall) and not e.over and not e.distinct:
                fn = e.name
                if fn == "count" and (not e.args or
                                      e.args[0].__class__.__name__ == "Star"):
                    plan.append(("count", None, alias or "count"))
                    continue
                if fn in ("sum", "min", "max", "avg") and len(e.args) == 1 \
                        and isinstance(e.args[0], ColumnRef):
                    plan.append((fn, e.args[0].name, alias or fn))
                    continue
            return None
        if not any(k in ("count", "sum", "min", "max", "avg")
                   for k, _, _ in plan):
            return None
        return (key, filt, plan)

    def _try_fast_agg(self, batch: MessageBatch):
        if self._fast_agg is None or batch.device.type != "cuda":
            return None
        key, filt, plan = self._fast_agg
        kc = batch.columns.get(key)
        if kc is None or kc.kind != "numeric" or kc.validity is not None \
                or kc.data.dtype != torch.int64:
            return None
        if filt is None:
            # no WHERE: synthesize an always-true filter on the first f32 col
            fname, opi, scalar = None, 5, float("nan")  # x != nan ⇒ all
        else:
            fname, opi, scalar = filt
        val_names = sorted({p for k, p, _ in plan
                            if k in ("sum", "min", "max", "avg")})
        tensors = {}
        for n in ([fname] if fname else []) + val_names:
            c = batch.columns.get(n)
            if c is None or c.kind != "numeric" or c.validity is not None \
                    or c.data.dtype != torch.float32:
                return None
            tensors[n] = c.data
        if fname is None:
            if not val_names:
                return None
            fname = val_names[0]
            tensors.setdefault(fname, batch.column(fname).data)
        ops_code = {"sum": 0, "avg": 0, "min": 1, "max": 2}
        val_list = []
        for n in val_names:
            # one reduction per (col, op) pair used by the plan
            for k, p, _ in plan:
                if p == n and k in ("sum", "min", "max", "avg"):
                    val_list.append((n, "sum" if k == "avg" else k))
        val_list = list(dict.fromkeys(val_list))
        from .. import ops as _ops
        nat = _ops.require_native()
        uniq, counts, reduced = nat.fused_filter_agg(
            kc.data, tensors[fname], opi, scalar,
            [tensors[n] for n, _ in val_list],
            [ops_code[o] for _, o in val_list])
        rmap = {pair: t for pair, t in zip(val_list, reduced)}
        from ..batch import Column
        cols = {}
        for k, p, alias in plan:
            if k == "key":
                cols[alias] = Column("numeric", uniq)
            elif k == "count":
                cols[alias] = Column("numeric", counts.to(torch.int64))
            elif k == "avg":
                cols[alias] = Column(
                    "numeric", rmap[(p, "sum")].double() / counts.double())
            else:
                cols[alias] = Column("numeric", rmap[(p, k)])
        return MessageBatch(cols, batch.input_name)

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
        fast = self._try_fast_agg(batch)
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
