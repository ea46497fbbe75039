"""SQL planner + columnar executor.

Replaces DataFusion's logical/physical plan for the engine's SQL subset
(reference processor/sql.rs:107-146 execute_query). Execution order:
FROM/JOIN → WHERE → GROUP BY + aggregates → HAVING → SELECT projection →
DISTINCT → ORDER BY → LIMIT. All row-level work is tensor ops routed through
:mod:`arkflow_amd.ops` (HIP kernels on GPU, torch on CPU).
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import torch

from .. import ops
from ..batch import Column, MessageBatch
from .eval import (
    Env,
    as_tensor,
    collect_aggregates,
    collect_window_calls,
    contains_aggregate,
    eval_expr,
    expr_name,
)
from .parser import (
    BinaryOp,
    ColumnRef,
    FuncCall,
    Literal,
    ExistsSubquery,
    ScalarSubquery,
    Select,
    SqlError,
    Star,
    parse_sql,
)

DEFAULT_TABLE = "flow"  # reference processor/sql.rs registers the batch as `flow`


class SqlExecutor:
    """Pre-parsed, reusable statement (reference pre-parses once, sql.rs:189)."""

    def __init__(self, sql: str):
        self.sql = sql
        self.select: Select = parse_sql(sql)
        self._aggs: List[FuncCall] = []
        for e, _ in self.select.projections:
            collect_aggregates(e, self._aggs)
        if self.select.having is not None:
            collect_aggregates(self.select.having, self._aggs)
        # dedup by expression text
        seen = {}
        for a in self._aggs:
            seen.setdefault(expr_name(a), a)
        self._aggs = list(seen.values())
        self.is_aggregate = bool(self._aggs or self.select.group_by)
        self._windows: List[FuncCall] = []
        for e, _ in self.select.projections:
            collect_window_calls(e, self._windows)
        wseen = {}
        for w_ in self._windows:
            wseen.setdefault(expr_name(w_), w_)
        self._windows = list(wseen.values())
        self._setops = [(op, SqlExecutor._from_select(sub))
                        for op, sub in self.select.set_ops]

    @staticmethod
    def _from_select(select: Select) -> "SqlExecutor":
        ex = SqlExecutor.__new__(SqlExecutor)
        ex.sql = ""
        ex.select = select
        ex._aggs = []
        for e, _ in select.projections:
            collect_aggregates(e, ex._aggs)
        if select.having is not None:
            collect_aggregates(select.having, ex._aggs)
        seen = {}
        for a in ex._aggs:
            seen.setdefault(expr_name(a), a)
        ex._aggs = list(seen.values())
        ex.is_aggregate = bool(ex._aggs or select.group_by)
        ex._windows = []
        for e, _ in select.projections:
            collect_window_calls(e, ex._windows)
        ex._setops = [(op, SqlExecutor._from_select(sub))
                      for op, sub in select.set_ops]
        return ex

    # -------------------------------------------------- subquery rewriting
    def _rewrite_subqueries(self, e, tables):
        """Per-execute rewrite of UNCORRELATED subqueries in WHERE / JOIN ON
        into literals: scalar subqueries become their single value; IN
        (SELECT ...) becomes a literal item list. Statements are pre-parsed
        and re-executed per batch, so the AST itself is never mutated —
        rewritten copies are built each execute."""
        import dataclasses

        from .parser import InList as _IL
        if e is None:
            return None
        if isinstance(e, ExistsSubquery):
            sub = SqlExecutor._from_select(e.select).execute(tables)
            hit = sub.num_rows > 0
            return Literal(not hit if e.negated else hit)
        if isinstance(e, ScalarSubquery):
            sub = SqlExecutor._from_select(e.select).execute(tables)
            cols = list(sub.columns.values())
            if len(cols) != 1:
                raise SqlError("scalar subquery must return one column")
            rows = cols[0].to_pylist()
            if len(rows) == 0:
                return Literal(None)
            if len(rows) > 1:
                raise SqlError("scalar subquery returned more than one row")
            v = rows[0]
            if isinstance(v, (bytes, bytearray)):
                v = v.decode("utf-8", "replace")
            return Literal(v)
        if isinstance(e, _IL) and e.subquery is not None:
            sub = SqlExecutor._from_select(e.subquery).execute(tables)
            cols = list(sub.columns.values())
            if len(cols) != 1:
                raise SqlError("IN subquery must return one column")
            items = []
            for v in cols[0].to_pylist():
                if v is None:
                    continue  # NULL members never equal anything
                if isinstance(v, (bytes, bytearray)):
                    v = v.decode("utf-8", "replace")
                items.append(Literal(v))
            return _IL(self._rewrite_subqueries(e.expr, tables), items,
                       e.negated)
        if dataclasses.is_dataclass(e) and not isinstance(e, Select):
            changed = False
            kwargs = {}
            for f in dataclasses.fields(e):
                v = getattr(e, f.name)
                if f.name == "whens" and isinstance(v, list):
                    nv = [(self._rewrite_subqueries(c, tables),
                           self._rewrite_subqueries(r, tables))
                          for c, r in v]
                    changed = True
                elif isinstance(v, list) and f.name in ("args", "items"):
                    nv = [self._rewrite_subqueries(x, tables) for x in v]
                    changed |= any(a is not b for a, b in zip(nv, v))
                elif dataclasses.is_dataclass(v) \
                        and not isinstance(v, Select):
                    nv = self._rewrite_subqueries(v, tables)
                    changed |= nv is not v
                else:
                    nv = v
                kwargs[f.name] = nv
            return dataclasses.replace(e, **kwargs) if changed else e
        return e

    @staticmethod
    def _contains_subquery(e) -> bool:
        import dataclasses
        if e is None or not dataclasses.is_dataclass(e):
            return False
        if isinstance(e, (ScalarSubquery, ExistsSubquery)):
            return True
        from .parser import InList as _IL
        if isinstance(e, _IL) and e.subquery is not None:
            return True
        if isinstance(e, Select):
            return False
        for f in dataclasses.fields(e):
            v = getattr(e, f.name)
            if isinstance(v, list):
                for x in v:
                    x2 = x[0] if isinstance(x, tuple) else x
                    if SqlExecutor._contains_subquery(x2):
                        return True
            elif SqlExecutor._contains_subquery(v):
                return True
        return False

    # ------------------------------------------------------------------- run
    def execute(self, tables: Dict[str, MessageBatch]) -> MessageBatch:
        sel = self.select
        from_spec = sel.from_table or DEFAULT_TABLE
        if isinstance(from_spec, Select):
            # derived table: execute the subquery, use it as the base
            base = SqlExecutor._from_select(from_spec).execute(tables)
            from_name = sel.from_alias or "subquery"
        else:
            from_name = from_spec
            if from_name not in tables:
                raise SqlError(
                    f"unknown table {from_name!r}; have {sorted(tables)}")
            base = tables[from_name]
        device = base.device

        columns: Dict[str, Column] = dict(base.columns)
        alias = sel.from_alias or from_name
        for k, v in base.columns.items():
            columns[f"{alias}.{k}"] = v
        order: List[str] = list(base.columns.keys())

        # ------------------------------------------------------------- joins
        for j in sel.joins:
            if self._contains_subquery(j.on):
                import dataclasses as _dc
                j = _dc.replace(
                    j, on=self._rewrite_subqueries(j.on, tables))
            if isinstance(j.table, Select):
                if not j.alias:
                    raise SqlError("a joined subquery requires an alias")
                right = SqlExecutor._from_select(j.table).execute(tables)
            else:
                right = tables.get(j.table)
                if right is None:
                    raise SqlError(f"unknown join table {j.table!r}")
            columns, order = self._apply_join(
                columns, order, alias, j, right, device)

        n_rows = len(columns[order[0]]) if order else 0
        env = Env(columns, n_rows, device)

        # ------------------------------------------------------------- where
        if sel.where is not None:
            where_e = self._rewrite_subqueries(sel.where, tables) \
                if self._contains_subquery(sel.where) else sel.where
            fused = self._try_fused_filter(where_e, columns, env)
            if fused is not None:
                columns, n_rows = fused
            else:
                idx = self._filter_indices(where_e, env)
                columns = {k: c.take(idx) for k, c in columns.items()}
                n_rows = int(idx.shape[0])
            env = Env(columns, n_rows, device)

        # ----------------------------------------------------------- windows
        if self._windows and not self.is_aggregate:
            wres = {expr_name(w_): self._compute_window(w_, env)
                    for w_ in self._windows}
            env = Env(env.columns, env.n_rows, device, wres)

        # --------------------------------------------------------- aggregate
        if self.is_aggregate:
            env, order = self._aggregate(env, order)

        # ------------------------------------------------------------ having
        if sel.having is not None:
            mask = as_tensor(eval_expr(sel.having, env), env).bool()
            idx = ops.mask_to_indices(mask).long()
            cols = {k: c.take(idx) for k, c in env.columns.items()}
            aggs = {k: v[idx] for k, v in env.agg_results.items()}
            env = Env(cols, int(idx.shape[0]), device, aggs)

        # -------------------------------------------------------- projection
        out_cols: Dict[str, Column] = {}
        for e, alias_name in sel.projections:
            if isinstance(e, Star):
                for name in order:
                    if e.table and not name.startswith(f"{e.table}."):
                        continue
                    out_name = name.split(".", 1)[1] if e.table else name
                    if out_name not in out_cols:
                        out_cols[out_name] = env.columns[name]
                continue
            v = eval_expr(e, env)
            name = alias_name or expr_name(e)
            col = _to_column(v, env)
            if col.validity is None and not self.is_aggregate:
                from .eval import expr_validity
                vv = expr_validity(e, env)
                if vv is not None and len(vv) == len(col):
                    col = Column(col.kind, col.data, col.offsets, vv)
            elif col.validity is None and self.is_aggregate \
                    and isinstance(e, ColumnRef):
                # group-key projection: the NULL group's key must stay NULL
                # (aggregate RESULTS never inherit input validity)
                try:
                    src_col = env.lookup(e)
                except SqlError:
                    src_col = None
                if isinstance(src_col, Column) \
                        and src_col.validity is not None \
                        and len(src_col.validity) == len(col):
                    col = Column(col.kind, col.data, col.offsets,
                                 src_col.validity)
            out_cols[name] = col
        result = MessageBatch(out_cols, input_name=None)

        # ---------------------------------------------------------- distinct
        if sel.distinct:
            gid, _, g = _encode_keys(
                [c if c.kind == "binary" else c.data
                 for c in result.columns.values()], device)
            rep = _first_index_per_group(gid, g)
            result = result.take(rep)

        # ---------------------------------------------------------- order by
        # (compound selects: ORDER BY/LIMIT bind to the whole compound and
        # run after the set ops below)
        if sel.order_by and not self._setops:
            cur_cols = dict(env.columns)  # permuted alongside result
            final_env = Env(
                {**cur_cols, **result.columns},
                result.num_rows, device, env.agg_results)
            # multi-key sort: stable sorts applied last-key-first
            from .parser import Literal as _Lit
            out_names = list(result.columns)
            for key_spec in reversed(sel.order_by):
                e, asc = key_spec[0], key_spec[1]
                nulls_first = key_spec[2] if len(key_spec) > 2 else None
                if isinstance(e, _Lit) and isinstance(e.value, int) \
                        and not isinstance(e.value, bool) \
                        and 1 <= e.value <= len(out_names):
                    # positional ORDER BY n → nth projection (sqlite/standard)
                    from .parser import ColumnRef as _CRef
                    e = _CRef(out_names[e.value - 1], None)
                v = eval_expr(e, final_env)
                key = v.data if isinstance(v, Column) and v.kind == "numeric" \
                    else v
                if isinstance(key, Column):
                    # binary sort key: CPU fallback
                    import numpy as np
                    vals = key.to_pylist()
                    # NULLs (None) get a stable placeholder here; the
                    # validity stable-sort below puts them in their
                    # NULLS FIRST/LAST position
                    idx = torch.tensor(
                        sorted(range(len(vals)),
                               key=lambda i: vals[i]
                               if vals[i] is not None else b"",
                               reverse=not asc),
                        dtype=torch.int64, device=device)
                else:
                    idx = ops.sort_indices(key, ascending=asc)
                from .eval import expr_validity
                kv = expr_validity(e, final_env)
                if kv is not None and bool((~kv).any()):
                    # default: NULL sorts as smallest (sqlite/standard);
                    # explicit NULLS FIRST/LAST overrides placement
                    if nulls_first is None:
                        null_asc = asc
                    else:
                        null_asc = nulls_first  # FIRST ⇔ invalid(0) first
                    idx = idx[ops.sort_indices(
                        kv[idx].to(torch.int32), ascending=null_asc)]
                result = result.take(idx)
                # take from the ALREADY-permuted columns: re-taking from the
                # original env would drop earlier keys' permutations (bug
                # caught by 3-key ORDER BY differential)
                cur_cols = {k: c.take(idx) for k, c in cur_cols.items()}
                final_env = Env(
                    {**cur_cols, **result.columns},
                    result.num_rows, device, env.agg_results)

        # ---------------------------------------------------- offset / limit
        if not self._setops:
            if sel.offset:
                result = result.slice(min(sel.offset, result.num_rows),
                                      max(result.num_rows - sel.offset, 0))
            if sel.limit is not None and result.num_rows > sel.limit:
                result = result.slice(0, sel.limit)
            return result

        # ------------------------------------------- set ops (compound)
        # UNION [ALL] / INTERSECT / EXCEPT left-to-right, then the
        # compound-level ORDER BY / OFFSET / LIMIT
        for op, ex in self._setops:
            result = self._apply_setop(result, op, ex.execute(tables),
                                       device)
        result = self._compound_order_limit(result, device)
        return result

    # ------------------------------------------------------------- set ops
    @staticmethod
    def _apply_setop(result: MessageBatch, op: str, other: MessageBatch,
                     device) -> MessageBatch:
        from ..batch import concat_batches
        if other.column_names != result.column_names and \
                len(other.column_names) == len(result.column_names):
            # positional semantics: rename the right side to our names
            other = MessageBatch(
                dict(zip(result.column_names, other.columns.values())),
                other.input_name)
        if op == "union_all":
            return concat_batches([result, other])
        both = concat_batches([result, other])
        gid, _, g = _encode_keys(
            [c if c.kind == "binary" else c.data
             for c in both.columns.values()], device)
        nl = result.num_rows
        if op == "union":
            rep = _first_index_per_group(gid, g)
            rep = rep[rep < both.num_rows].sort().values
            return both.take(rep)
        gl = gid[:nl]
        first = _first_index_per_group(gl, g)
        ar = torch.arange(nl, dtype=torch.int64, device=gl.device)
        rep_rows = (first[gl] == ar).nonzero(as_tuple=True)[0]
        right_has = torch.zeros(g, dtype=torch.bool, device=gl.device)
        if both.num_rows > nl:
            right_has[gid[nl:]] = True
        keep = right_has[gl[rep_rows]]
        if op == "except":
            keep = ~keep
        return result.take(rep_rows[keep])

    def _compound_order_limit(self, result: MessageBatch, device
                              ) -> MessageBatch:
        """Compound-level ORDER BY / OFFSET / LIMIT: keys may reference the
        output columns (names or positions) only — sqlite/standard."""
        sel = self.select
        if sel.order_by:
            out_names = list(result.columns)
            from .parser import Literal as _Lit
            for key_spec in reversed(sel.order_by):
                e, asc = key_spec[0], key_spec[1]
                nulls_first = key_spec[2] if len(key_spec) > 2 else None
                if isinstance(e, _Lit) and isinstance(e.value, int) \
                        and not isinstance(e.value, bool) \
                        and 1 <= e.value <= len(out_names):
                    e = ColumnRef(out_names[e.value - 1], None)
                env = Env(dict(result.columns), result.num_rows, device)
                v = eval_expr(e, env)
                if isinstance(v, Column) and v.kind == "binary":
                    vals = v.to_pylist()
                    idx = torch.tensor(
                        sorted(range(len(vals)),
                               key=lambda i: vals[i]
                               if vals[i] is not None else b"",
                               reverse=not asc),
                        dtype=torch.int64, device=device)
                else:
                    idx = ops.sort_indices(as_tensor(v, env), ascending=asc)
                from .eval import expr_validity
                kv = expr_validity(e, env)
                if isinstance(v, Column) and v.validity is not None:
                    kv = v.validity if kv is None else (kv & v.validity)
                if kv is not None and bool((~kv).any()):
                    null_asc = asc if nulls_first is None else nulls_first
                    idx = idx[ops.sort_indices(
                        kv[idx].to(torch.int32), ascending=null_asc)]
                result = result.take(idx)
        if sel.offset:
            result = result.slice(min(sel.offset, result.num_rows),
                                  max(result.num_rows - sel.offset, 0))
        if sel.limit is not None and result.num_rows > sel.limit:
            result = result.slice(0, sel.limit)
        return result

    # --------------------------------------------------------------- windows
    def _running_window_agg(self, w_: FuncCall, name: str,
                            vals: torch.Tensor, gid: torch.Tensor,
                            env: Env) -> torch.Tensor:
        """RANGE UNBOUNDED PRECEDING..CURRENT ROW over the partition order:
        sort by (partition, ORDER BY keys), scan, and give equal peers the
        frame end of their tie group."""
        device = env.device
        n = env.n_rows
        perm = torch.arange(n, dtype=torch.int64, device=device)
        sort_keys = []
        for e, asc in reversed(w_.over.order_by):
            v = eval_expr(e, env)
            key = v.data if isinstance(v, Column) and v.kind == "numeric" \
                else as_tensor(v, env)
            sort_keys.append((key, asc))
            perm = perm[ops.sort_indices(key[perm], ascending=asc)]
        perm = perm[ops.sort_indices(gid[perm], ascending=True)]
        sg = gid[perm]
        sv = vals[perm].double()
        ar = torch.arange(n, dtype=torch.int64, device=device)
        part_start = torch.zeros(n, dtype=torch.bool, device=device)
        part_start[0] = True
        part_start[1:] = sg[1:] != sg[:-1]
        group_start = torch.cummax(ar * part_start, 0).values
        tie_change = part_start.clone()
        for key, asc in sort_keys:
            sk = key[perm]
            tie_change[1:] |= sk[1:] != sk[:-1]
        tie_id = torch.cumsum(tie_change.long(), 0) - 1
        n_ties = int(tie_id[-1].item()) + 1 if n else 0
        last = torch.zeros(max(n_ties, 1), dtype=torch.int64, device=device)
        last.scatter_reduce_(0, tie_id, ar, "amax", include_self=False)
        frame_end = last[tie_id]  # last peer of my tie group
        if name in ("sum", "avg", "count"):
            cs = torch.cumsum(sv, 0)
            base = cs[group_start] - sv[group_start]  # sum before partition
            run_sum = cs[frame_end] - base
            run_cnt = (frame_end - group_start + 1).double()
            if name == "count":
                out_sorted = run_cnt
            elif name == "sum":
                out_sorted = run_sum
            else:
                out_sorted = run_sum / run_cnt
        else:  # min / max: segmented scan by doubling within partitions
            y = sv.clone()
            dist = ar - group_start
            shift = 1
            while shift < n:
                shifted = torch.empty_like(y)
                shifted[shift:] = y[:-shift]
                shifted[:shift] = y[:shift]  # unused (masked)
                mask = dist >= shift
                y = torch.where(mask,
                                torch.maximum(y, shifted) if name == "max"
                                else torch.minimum(y, shifted), y)
                shift <<= 1
            out_sorted = y[frame_end]
        out = torch.empty(n, dtype=torch.float64, device=device)
        out[perm] = out_sorted
        return out

    def _compute_window(self, w_: FuncCall, env: Env) -> torch.Tensor:
        """Window functions over PARTITION BY (full-partition frame):
        row_number/rank/dense_rank and partition-wide aggregates."""
        device = env.device
        n = env.n_rows
        if w_.over.partition_by:
            keys = [eval_expr(p, env) for p in w_.over.partition_by]
            keys = [k if isinstance(k, Column) else as_tensor(k, env)
                    for k in keys]
            single = keys[0]
            if len(keys) == 1 and isinstance(single, torch.Tensor) \
                    and single.dtype in (torch.int64, torch.int32):
                gid, _, g = ops.hash_group(single)
                gid = gid.long()
            else:
                gid, _, g = _encode_keys(keys, device)
        else:
            gid = torch.zeros(n, dtype=torch.int64, device=device)
            g = 1
        name = w_.name
        if name in ("sum", "count", "avg", "min", "max"):
            if name == "count" and (not w_.args or isinstance(w_.args[0],
                                                              Star)):
                vals = torch.ones(n, device=device)
            else:
                vals = as_tensor(eval_expr(w_.args[0], env), env)
            if w_.over.order_by:
                # running aggregate: the standard's default frame with
                # ORDER BY is RANGE UNBOUNDED PRECEDING..CURRENT ROW —
                # equal peers share the frame end (sqlite/DataFusion)
                return self._running_window_agg(w_, name, vals, gid, env)
            per_group = ops.segment_reduce(
                vals.float() if name != "count" else vals, gid.to(torch.int32),
                g, "mean" if name == "avg" else name)
            return per_group[gid]
        from .udf import window_udf
        wudf = window_udf(name)
        # rank family / window UDFs: stable sort by order keys (last-first),
        # then by gid
        perm = torch.arange(n, dtype=torch.int64, device=device)
        sort_keys = []
        for e, asc in reversed(w_.over.order_by):
            v = eval_expr(e, env)
            key = v.data if isinstance(v, Column) and v.kind == "numeric" \
                else as_tensor(v, env)
            sort_keys.append((key, asc))
            idx = ops.sort_indices(key[perm], ascending=asc)
            perm = perm[idx]
        idx = ops.sort_indices(gid[perm], ascending=True)
        perm = perm[idx]
        sorted_gid = gid[perm]
        ar = torch.arange(n, dtype=torch.int64, device=device)
        part_start = torch.zeros(n, dtype=torch.bool, device=device)
        part_start[0] = True
        part_start[1:] = sorted_gid[1:] != sorted_gid[:-1]
        group_start = torch.cummax(ar * part_start, 0).values
        if name in ("lag", "lead", "first_value"):
            # offset/navigation family over the sorted partition order;
            # out-of-frame rows are NULL unless a default arg is given
            if not w_.args or isinstance(w_.args[0], Star):
                raise SqlError(f"{name}() requires a value argument")
            v0 = eval_expr(w_.args[0], env)
            vals = v0.data if isinstance(v0, Column) and v0.kind == "numeric" \
                else as_tensor(v0, env)
            vs = vals[perm]
            if name == "first_value":
                out_sorted_v = vs[group_start]
                valid_sorted = None
            else:
                k = 1
                if len(w_.args) > 1 and isinstance(w_.args[1], Literal):
                    k = int(w_.args[1].value)
                default = None
                if len(w_.args) > 2:  # any scalar expr (e.g. -1)
                    dv = eval_expr(w_.args[2], env)
                    if isinstance(dv, Column):
                        dv = dv.data
                    if isinstance(dv, torch.Tensor):
                        default = float(dv.reshape(-1)[0].item())
                    else:
                        default = float(dv)
                counts = torch.bincount(sorted_gid, minlength=max(g, 1))
                group_end = group_start + counts[sorted_gid] - 1
                if name == "lag":
                    src = ar - k
                    valid_sorted = src >= group_start
                else:
                    src = ar + k
                    valid_sorted = src <= group_end
                out_sorted_v = vs[src.clamp(0, n - 1)]
                if default is not None:
                    out_sorted_v = torch.where(
                        valid_sorted, out_sorted_v,
                        torch.full_like(out_sorted_v, float(default)))
                    valid_sorted = None
            out_v = torch.empty_like(out_sorted_v)
            out_v[perm] = out_sorted_v
            if valid_sorted is None:
                return out_v
            out_valid = torch.empty(n, dtype=torch.bool, device=device)
            out_valid[perm] = valid_sorted
            col = Column("numeric", out_v)
            col.validity = out_valid
            return col
        out_sorted = ar - group_start + 1  # row_number
        if name in ("rank", "dense_rank"):
            tie_change = part_start.clone()
            for key, asc in sort_keys:
                sk = key[perm]
                tie_change[1:] |= sk[1:] != sk[:-1]
            if name == "rank":
                rank_start = torch.cummax(ar * tie_change, 0).values
                out_sorted = rank_start - group_start + 1
            else:
                cum = torch.cumsum(tie_change.long(), 0)
                base = cum[group_start]
                out_sorted = cum - base + 1
        elif wudf is not None:
            vals = None
            if w_.args and not isinstance(w_.args[0], Star):
                vals = as_tensor(eval_expr(w_.args[0], env), env)
            res = wudf(vals, gid, g, perm)
            return res if res.dtype != torch.bool else res.long()
        elif name != "row_number":
            raise SqlError(f"unsupported window function {name}()")
        out = torch.empty(n, dtype=torch.int64, device=device)
        out[perm] = out_sorted
        return out

    # --------------------------------------------------------------- filters
    def _try_fused_filter(self, pred, columns, env: Env):
        """`col OP literal` over all-numeric device columns → ONE native call
        (fused compare+compact+multi-gather, csrc bindings)."""
        if not (isinstance(pred, BinaryOp) and pred.op in (
                "<", "<=", ">", ">=", "=", "!=")):
            return None
        l, r = pred.left, pred.right
        if not (isinstance(l, ColumnRef) and isinstance(r, Literal)
                and isinstance(r.value, (int, float))):
            return None
        try:
            fcol = env.lookup(l)
        except SqlError:
            return None
        if fcol.kind != "numeric" or not fcol.data.is_cuda \
                or fcol.data.dtype not in (torch.float32, torch.int64,
                                           torch.int32):
            return None
        names, tensors = [], []
        for n, c in columns.items():
            if c.kind != "numeric" or c.validity is not None:
                return None
            names.append(n)
            tensors.append(c.data)
        fidx = None
        for i, t in enumerate(tensors):
            if t.data_ptr() == fcol.data.data_ptr():
                fidx = i
                break
        if fidx is None:
            return None
        from ..ops import native_available, require_native
        if not native_available():
            return None
        opi = {"<": 0, "<=": 1, ">": 2, ">=": 3, "=": 4, "!=": 5}[pred.op]
        outs, total = require_native().fused_filter_gather(
            tensors, fidx, opi, float(r.value))
        return ({n: Column("numeric", t) for n, t in zip(names, outs)},
                int(total))

    def _filter_indices(self, pred, env: Env) -> torch.Tensor:
        """WHERE → row indices. Fast path: `col OP numeric-literal` fuses
        compare+compact in one HIP kernel (csrc/filter.hip)."""
        if isinstance(pred, BinaryOp) and pred.op in (
                "<", "<=", ">", ">=", "=", "!="):
            l, r = pred.left, pred.right
            if isinstance(l, ColumnRef) and isinstance(r, Literal) \
                    and isinstance(r.value, (int, float)):
                col = env.lookup(l)
                if col.kind == "numeric" and col.validity is None \
                        and col.data.dtype in (
                        torch.float32, torch.int64, torch.int32):
                    return ops.filter_cmp_scalar(
                        col.data, pred.op, float(r.value)).long()
        mask = as_tensor(eval_expr(pred, env), env).bool()
        from .eval import expr_validity
        v = expr_validity(pred, env)
        if v is not None:
            mask = mask & v  # NULL predicate → row dropped (SQL three-valued)
        return ops.mask_to_indices(mask).long()

    # ------------------------------------------------------------------ join
    def _apply_join(self, columns, order, left_alias, j, right: MessageBatch,
                    device):
        r_alias = j.alias or j.table
        conjuncts = _split_and(j.on)

        def side_of(ref: ColumnRef):
            if ref.table == r_alias or (
                    ref.table is None and ref.name in right.columns
                    and ref.name not in columns):
                return "right"
            return "left"

        def _refs(e, acc):
            if isinstance(e, ColumnRef):
                acc.append(e)
            for attr in ("left", "right", "expr", "low", "high"):
                sub = getattr(e, attr, None)
                if sub is not None and not isinstance(sub, str):
                    _refs(sub, acc)
            for sub in getattr(e, "args", []) or []:
                _refs(sub, acc)
            return acc

        def expr_side(e):
            sides = {side_of(r) for r in _refs(e, [])}
            if not sides:
                return "none"  # constant: cannot drive the hash join
            if len(sides) > 1:
                return "mixed"
            return sides.pop()

        # equality conjuncts whose sides each reference exactly one table
        # become (oriented) hash keys — arbitrary EXPRESSIONS allowed
        # (DataFusion supports e.g. ON f.k + 1 = d.k); everything else is
        # a residual condition
        key_pairs = []
        residual = []
        for c in conjuncts:
            if isinstance(c, BinaryOp) and c.op == "=":
                ls, rs = expr_side(c.left), expr_side(c.right)
                if ls == "left" and rs == "right":
                    key_pairs.append((c.left, c.right))
                    continue
                if ls == "right" and rs == "left":
                    key_pairs.append((c.right, c.left))
                    continue
            residual.append(c)
        if not key_pairs:
            raise SqlError("JOIN requires at least one equality condition "
                           "relating the two tables")

        l_keys, r_keys = [], []
        n_left = len(columns[order[0]])
        l_env = Env(columns, n_left, device)
        r_cols_q = {f"{r_alias}.{k}": v for k, v in right.columns.items()}
        r_env = Env({**right.columns, **r_cols_q}, right.num_rows, device)
        for a, b in key_pairs:
            l_keys.append(as_tensor(eval_expr(a, l_env), l_env))
            r_keys.append(as_tensor(eval_expr(b, r_env), r_env))
        lk, _, _ = (_encode_keys(l_keys, device) if len(l_keys) > 1
                    else (l_keys[0], None, 0))
        rk = r_keys[0] if len(r_keys) == 1 else None
        if len(l_keys) > 1:
            # co-encode both sides so hashes align
            lk, rk = _co_encode(l_keys, r_keys, device)
        lk = lk.to(torch.int64) if lk.dtype not in (
            torch.int64, torch.float32) else lk
        rk = rk.to(lk.dtype) if rk.dtype != lk.dtype else rk
        # NULL join keys never match (SQL): drop invalid-key rows from each
        # side before the hash join, remapping indices back afterwards. For
        # LEFT joins, invalid-left rows re-enter unmatched (NULL right side).
        from .eval import expr_validity
        l_val = r_val = None
        for a, b in key_pairs:
            va = expr_validity(a, l_env)
            vb = expr_validity(b, r_env)
            if va is not None:
                l_val = va if l_val is None else (l_val & va)
            if vb is not None:
                r_val = vb if r_val is None else (r_val & vb)
        left_map = right_map = None
        null_left = None
        if l_val is not None and bool((~l_val).any()):
            left_map = l_val.nonzero(as_tuple=True)[0]
            null_left = (~l_val).nonzero(as_tuple=True)[0]
            lk = lk[left_map]
        if r_val is not None and bool((~r_val).any()):
            right_map = r_val.nonzero(as_tuple=True)[0]
            rk = rk[right_map]
        if j.kind == "left":
            l_idx, r_idx = ops.join_left(lk, rk)
        else:
            l_idx, r_idx = ops.join_inner(lk, rk)
        if left_map is not None:
            l_idx = left_map[l_idx.long()]
            if j.kind == "left" and null_left.numel():
                l_idx = torch.cat([l_idx, null_left])
                r_idx = torch.cat([r_idx, torch.full(
                    (null_left.numel(),), -1, dtype=r_idx.dtype,
                    device=r_idx.device)])
        if right_map is not None:
            matched = r_idx >= 0
            remapped = r_idx.clone().long()
            remapped[matched] = right_map[r_idx[matched].long()]
            r_idx = remapped

        def build(li, ri):
            cols: Dict[str, Column] = {}
            ordr: List[str] = []
            for name in order:
                c = columns[name].take(li)
                cols[name] = c
                cols[f"{left_alias}.{name.split('.', 1)[-1]}"] = c
                ordr.append(name)
            safe_r = ri.clamp(min=0)
            nulls = ri < 0
            for name, col in right.columns.items():
                taken = col.take(safe_r)
                if bool(nulls.any()):
                    taken = Column(taken.kind, taken.data, taken.offsets,
                                   ~nulls)
                qual = f"{r_alias}.{name}"
                cols[qual] = taken
                if name not in cols:  # unqualified only when unambiguous
                    cols[name] = taken
                    ordr.append(name)
                else:
                    ordr.append(qual)
            return cols, ordr

        new_cols, new_order = build(l_idx, r_idx)
        if residual:
            n = int(l_idx.shape[0])
            env = Env(new_cols, n, device)
            mask = torch.ones(n, dtype=torch.bool, device=device)
            for c in residual:
                mask &= as_tensor(eval_expr(c, env), env).bool()
            if j.kind == "left":
                # LEFT JOIN semantics: a residual ON conjunct qualifies the
                # MATCH, it does not filter rows — matches that fail it
                # null-extend. Null-extended rows pass through untouched; a
                # left row whose every match fails re-enters once with a
                # NULL right side. (Found by the sqlite differential suite:
                # anti-join idiom ... ON f.k = d.k AND d.k < 3 WHERE d.k
                # IS NULL returned nothing.)
                null_rows = r_idx < 0
                matched_keep = mask & ~null_rows
                li_long = l_idx.long()
                had = torch.zeros(n_left, dtype=torch.bool, device=device)
                had[li_long[~null_rows]] = True
                kept_any = torch.zeros(n_left, dtype=torch.bool,
                                       device=device)
                kept_any[li_long[matched_keep]] = True
                demoted = (had & ~kept_any).nonzero(as_tuple=True)[0]
                keep = matched_keep | null_rows
                l2 = torch.cat([l_idx[keep],
                                demoted.to(l_idx.dtype)])
                r2 = torch.cat([r_idx[keep],
                                torch.full((demoted.numel(),), -1,
                                           dtype=r_idx.dtype,
                                           device=r_idx.device)])
                new_cols, new_order = build(l2, r2)
            else:
                idx = ops.mask_to_indices(mask).long()
                new_cols = {k: c.take(idx) for k, c in new_cols.items()}
        return new_cols, new_order

    # ------------------------------------------------------------- aggregate
    def _aggregate(self, env: Env, order: List[str]
                   ) -> Tuple[Env, List[str]]:
        sel = self.select
        device = env.device
        n = env.n_rows
        if sel.group_by:
            # GROUP BY may name a projection alias (sqlite/DataFusion
            # semantics): substitute the aliased expression
            from .parser import ColumnRef as _CR
            aliases = {alias: pe for pe, alias in sel.projections
                       if alias is not None}
            group_exprs = [aliases[e.name]
                           if isinstance(e, _CR) and e.table is None
                           and e.name in aliases and e.name not in env.columns
                           else e
                           for e in sel.group_by]
            key_vals = [eval_expr(e, env) for e in group_exprs]
            keys = []
            for ge, v in zip(group_exprs, key_vals):
                k = v if isinstance(v, Column) else as_tensor(v, env)
                from .eval import expr_validity
                validity = expr_validity(ge, env)
                if validity is not None and bool((~validity).any()):
                    # NULL keys form ONE group: normalize the value under
                    # the null and add the validity bit to the key material
                    kt = k.data if isinstance(k, Column) else k
                    if kt.dtype != torch.bool and kt.dim() == 1 \
                            and not isinstance(k, Column):
                        k = torch.where(validity, kt,
                                        torch.zeros_like(kt))
                    keys.append(k)
                    keys.append(validity)
                    continue
                keys.append(k)
            single = keys[0]
            if len(keys) == 1 and isinstance(single, torch.Tensor) \
                    and single.dtype in (torch.int64, torch.int32):
                gid, _, g = ops.hash_group(single)
                gid = gid.long()
            else:
                gid, _, g = _encode_keys(keys, device)
        else:
            gid = torch.zeros(n, dtype=torch.int64, device=device)
            g = 1 if n > 0 else 1  # global aggregate yields one row even if empty

        agg_results: Dict[str, torch.Tensor] = {}
        for a in self._aggs:
            agg_results[expr_name(a)] = self._compute_agg(a, env, gid, g)

        # representative row per group for key/non-agg column access
        if n > 0:
            rep = _first_index_per_group(gid, g)
            cols = {k: c.take(rep) for k, c in env.columns.items()}
        else:
            cols = {k: c.slice(0, 0) for k, c in env.columns.items()}
            if not sel.group_by:
                # 1-row global aggregate over empty input: null keys
                cols = {k: _null_like(c, 1) for k, c in env.columns.items()}
        return Env(cols, g, device, agg_results), order

    def _compute_agg(self, a: FuncCall, env: Env, gid: torch.Tensor, g: int
                     ) -> torch.Tensor:
        name = a.name
        device = env.device
        from ..sql.udf import aggregate_udf
        udf = aggregate_udf(name)
        if udf is not None:
            vals = as_tensor(eval_expr(a.args[0], env), env) if a.args \
                else torch.ones(env.n_rows, device=device)
            return udf(vals, gid, g)
        if env.n_rows == 0:
            if name == "count":
                return torch.zeros(g, dtype=torch.int64, device=device)
            return torch.full((g,), float("nan"), device=device)
        if name == "count" and (not a.args or isinstance(a.args[0], Star)):
            ones = torch.ones(env.n_rows, device=device)
            if a.filter is not None:
                fm = as_tensor(eval_expr(a.filter, env), env).bool()
                ones = torch.where(fm, ones, torch.zeros_like(ones))
                return ops.segment_reduce(ones, gid, g, "sum"
                                          ).to(torch.int64)
            return ops.segment_reduce(ones, gid, g, "count")
        arg = eval_expr(a.args[0], env)
        if name == "count":
            if a.distinct:
                key = arg if isinstance(arg, Column) and arg.kind == "binary" \
                    else (arg.data if isinstance(arg, Column) else arg)
                pair_gid, _, _ = _encode_keys([gid, key], device)
                # distinct (gid, val) pairs per gid
                uniq_pair = _first_index_per_group(
                    pair_gid, int(pair_gid.max().item()) + 1)
                sub_gid = gid[uniq_pair]
                return ops.segment_reduce(
                    torch.ones(sub_gid.shape[0], device=device),
                    sub_gid, g, "count")
            from .eval import expr_validity
            validity = expr_validity(a.args[0], env)
            if isinstance(arg, Column) and arg.validity is not None:
                validity = arg.validity if validity is None \
                    else (validity & arg.validity)
            if validity is not None:
                return ops.segment_reduce(
                    validity.to(torch.float32), gid, g, "sum"
                ).to(torch.int64)
            return ops.segment_reduce(
                torch.ones(env.n_rows, device=device), gid, g, "count")
        if isinstance(arg, Column) and arg.kind == "binary" \
                and name in ("min", "max"):
            # lexicographic min/max over strings: host fallback (DataFusion
            # supports string MIN/MAX; group counts are small post-hash)
            pv = arg.to_pylist()
            gl = gid.detach().to("cpu").tolist()
            best = [None] * g
            for v, gi in zip(pv, gl):
                if v is None:
                    continue
                b = best[gi]
                if b is None or (v < b if name == "min" else v > b):
                    best[gi] = v
            out = Column.from_bytes([b if b is not None else b""
                                     for b in best])
            if any(b is None for b in best):
                out = Column(out.kind, out.data, out.offsets,
                             torch.tensor([b is not None for b in best],
                                          dtype=torch.bool, device=device))
            return out
        vals = as_tensor(arg, env)
        from .eval import expr_validity
        validity = expr_validity(a.args[0], env) if a.args else None
        if a.filter is not None:
            fm = as_tensor(eval_expr(a.filter, env), env).bool()
            validity = fm if validity is None else (validity & fm)
        if vals.dtype.is_floating_point and bool(torch.isnan(vals).any()):
            # NaN is the in-band NULL for value-level nulls (e.g. CASE
            # without ELSE); aggregates must skip it like any NULL
            nn = ~torch.isnan(vals)
            validity = nn if validity is None else (validity & nn)
        if validity is not None and bool((~validity).any()):
            # SQL aggregates ignore NULL inputs
            vf = vals.float()
            if name in ("sum", "avg"):
                s = ops.segment_reduce(
                    torch.where(validity, vf, torch.zeros_like(vf)),
                    gid, g, "sum")
                cnt = ops.segment_reduce(validity.float(), gid, g, "sum")
                if name == "sum":
                    # SQL: SUM over zero qualifying rows is NULL, not 0
                    if bool((cnt == 0).any()):
                        return torch.where(cnt == 0,
                                           torch.full_like(s, float("nan")),
                                           s)
                    return s.to(torch.int64) \
                        if not vals.dtype.is_floating_point else s
                return s / cnt
            fill = float("inf") if name == "min" else float("-inf")
            out = ops.segment_reduce(
                torch.where(validity, vf, torch.full_like(vf, fill)),
                gid, g, name)
            out = torch.where(torch.isinf(out),
                              torch.full_like(out, float("nan")), out)
            return out.to(torch.int64) if not vals.dtype.is_floating_point \
                and not torch.isnan(out).any() else out
        if name == "avg":
            return ops.segment_reduce(vals, gid, g, "mean")
        out = ops.segment_reduce(vals, gid, g, name)  # sum|min|max
        if not vals.dtype.is_floating_point and name in ("sum", "min", "max"):
            out = out.to(torch.int64)
        elif vals.dtype == torch.float32 and out.dtype == torch.float64:
            out = out  # keep f64 accumulations for numeric fidelity
        return out


# -------------------------------------------------------------------- helpers
def _split_and(e) -> list:
    if isinstance(e, BinaryOp) and e.op == "and":
        return _split_and(e.left) + _split_and(e.right)
    return [e]


def _to_column(v, env: Env) -> Column:
    if isinstance(v, Column):
        return v
    if isinstance(v, torch.Tensor):
        return Column("numeric", v)
    if isinstance(v, str):
        return Column.from_strings([v] * env.n_rows)
    # scalar broadcast
    return Column("numeric", as_tensor(v, env))


def _encode_keys(keys: list, device) -> Tuple[torch.Tensor, None, int]:
    """Encode 1..k key columns (tensors or binary Columns) into group ids."""
    if any(isinstance(k, Column) and k.kind == "binary" for k in keys):
        if all((k.data.is_cuda if isinstance(k, Column) else k.is_cuda)
               for k in keys):
            # device path: FNV-1a row hashes for binary keys, then the same
            # numeric encoding (csrc bytes_hash kernel)
            from ..ops import native_available, require_native
            if native_available():
                nat = require_native()
                ts = [nat.bytes_hash(k.data, k.offsets)
                      if isinstance(k, Column) and k.kind == "binary"
                      else (k.data if isinstance(k, Column) else k)
                      for k in keys]
                return _encode_keys(ts, device)
        # dictionary-encode binary keys host-side (CPU / fallback)
        lists = []
        for k in keys:
            if isinstance(k, Column):
                lists.append(k.to_pylist())
            else:
                lists.append(k.detach().cpu().tolist())
        seen: dict = {}
        gids = []
        for row in zip(*lists):
            gids.append(seen.setdefault(row, len(seen)))
        return (torch.tensor(gids, dtype=torch.int64, device=device), None,
                len(seen))
    ts = []
    all_int = all(not k.dtype.is_floating_point for k in keys)
    for k in keys:
        ts.append(k.to(torch.int64) if all_int else k.to(torch.float64))
    stacked = torch.stack(ts, dim=1)
    uniq, inverse = torch.unique(stacked, dim=0, return_inverse=True)
    return inverse.to(torch.int64), None, int(uniq.shape[0])


def _co_encode(l_keys, r_keys, device):
    n_l = l_keys[0].shape[0]
    combined = [torch.cat([l.to(torch.float64), r.to(torch.float64)])
                for l, r in zip(l_keys, r_keys)]
    gid, _, _ = _encode_keys(combined, device)
    return gid[:n_l], gid[n_l:]


def _first_index_per_group(gid: torch.Tensor, g: int) -> torch.Tensor:
    n = gid.shape[0]
    if gid.is_cuda and n < (1 << 24):
        # int64 scatter_reduce on GPU is CAS-loop atomics — use the native
        # flipped-f32 segment min (exact for indices < 2^24)
        idx_f = torch.arange(n, dtype=torch.float32, device=gid.device)
        return ops.segment_reduce(idx_f, gid.to(torch.int32), g,
                                  "min").to(torch.int64)
    idx = torch.arange(n, dtype=torch.int64, device=gid.device)
    first = torch.full((g,), n, dtype=torch.int64, device=gid.device)
    first.scatter_reduce_(0, gid.long(), idx, reduce="amin", include_self=True)
    return first


def _null_like(c: Column, n: int) -> Column:
    import torch as _t
    validity = _t.zeros(n, dtype=_t.bool)
    if c.kind == "numeric":
        return Column("numeric",
                      _t.zeros(n, dtype=c.data.dtype, device=c.data.device),
                      validity=validity.to(c.data.device))
    off = _t.zeros(n + 1, dtype=_t.int64, device=c.data.device)
    return Column("binary", _t.empty(0, dtype=_t.uint8, device=c.data.device),
                  off, validity.to(c.data.device))
