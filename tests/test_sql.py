"""SQL engine tests: parser + executor semantics (CPU; the same executor runs
HIP kernels on GPU — see test_gpu_kernels.py for device parity)."""
import math

import pytest
import torch

from arkflow_amd.batch import MessageBatch
from arkflow_amd.sql.engine import SqlExecutor
from arkflow_amd.sql.parser import SqlError, parse_sql


def q(sql, **tables):
    return SqlExecutor(sql).execute(tables)


@pytest.fixture
def flow():
    return MessageBatch.from_dict({
        "id": [1, 2, 3, 4, 5, 6],
        "value": [10.0, 20.0, 5.0, 40.0, 15.0, 20.0],
        "grp": [1, 1, 2, 2, 3, 3],
        "name": ["alpha", "beta", "gamma", "alpha", "delta", "beta"],
    })


def test_select_star_where(flow):
    r = q("SELECT * FROM flow WHERE value >= 15", flow=flow)
    assert r.column("id").to_pylist() == [2, 4, 5, 6]
    assert r.column_names == ["id", "value", "grp", "name"]


def test_projection_arithmetic_alias(flow):
    r = q("SELECT id, value * 2 + 1 AS v2 FROM flow WHERE id <= 2", flow=flow)
    assert r.column("v2").to_pylist() == [21.0, 41.0]


def test_where_and_or_not_between_in(flow):
    r = q("SELECT id FROM flow WHERE (value > 10 AND grp = 1) OR id = 5",
          flow=flow)
    assert r.column("id").to_pylist() == [2, 5]
    r = q("SELECT id FROM flow WHERE value BETWEEN 10 AND 20", flow=flow)
    assert r.column("id").to_pylist() == [1, 2, 5, 6]
    r = q("SELECT id FROM flow WHERE id IN (1, 3, 6)", flow=flow)
    assert r.column("id").to_pylist() == [1, 3, 6]
    r = q("SELECT id FROM flow WHERE NOT value < 15", flow=flow)
    assert r.column("id").to_pylist() == [2, 4, 5, 6]


def test_string_predicates(flow):
    r = q("SELECT id FROM flow WHERE name = 'alpha'", flow=flow)
    assert r.column("id").to_pylist() == [1, 4]
    r = q("SELECT id FROM flow WHERE name LIKE '%et%'", flow=flow)
    assert r.column("id").to_pylist() == [2, 6]
    r = q("SELECT id, length(name) AS l FROM flow WHERE name != 'beta'",
          flow=flow)
    assert r.column("l").to_pylist() == [5, 5, 5, 5]


def test_global_aggregates(flow):
    r = q("SELECT count(*) AS c, sum(value) AS s, avg(value) AS a, "
          "min(value) AS lo, max(value) AS hi FROM flow", flow=flow)
    assert r.num_rows == 1
    assert r.column("c").to_pylist() == [6]
    assert r.column("s").to_pylist() == [110.0]
    assert abs(r.column("a").to_pylist()[0] - 110.0 / 6) < 1e-9
    assert r.column("lo").to_pylist() == [5.0]
    assert r.column("hi").to_pylist() == [40.0]


def test_group_by(flow):
    r = q("SELECT grp, sum(value) AS s, count(*) AS c FROM flow "
          "GROUP BY grp ORDER BY grp", flow=flow)
    assert r.column("grp").to_pylist() == [1, 2, 3]
    assert r.column("s").to_pylist() == [30.0, 45.0, 35.0]
    assert r.column("c").to_pylist() == [2, 2, 2]


def test_group_by_string_key(flow):
    r = q("SELECT name, count(*) AS c FROM flow GROUP BY name "
          "ORDER BY c DESC, name", flow=flow)
    names = r.column("name").to_strlist()
    assert names[:2] == ["alpha", "beta"]
    assert r.column("c").to_pylist() == [2, 2, 1, 1]


def test_having(flow):
    r = q("SELECT grp, sum(value) AS s FROM flow GROUP BY grp "
          "HAVING sum(value) > 30 ORDER BY grp", flow=flow)
    assert r.column("grp").to_pylist() == [2, 3]


def test_count_distinct(flow):
    r = q("SELECT count(DISTINCT grp) AS g FROM flow", flow=flow)
    assert r.column("g").to_pylist() == [3]


def test_order_limit(flow):
    r = q("SELECT id, value FROM flow ORDER BY value DESC, id ASC LIMIT 3",
          flow=flow)
    assert r.column("id").to_pylist() == [4, 2, 6]


def test_distinct(flow):
    r = q("SELECT DISTINCT grp FROM flow ORDER BY grp", flow=flow)
    assert r.column("grp").to_pylist() == [1, 2, 3]


def test_case_when(flow):
    r = q("SELECT id, CASE WHEN value >= 20 THEN 1 ELSE 0 END AS big "
          "FROM flow ORDER BY id", flow=flow)
    assert r.column("big").to_pylist() == [0.0, 1.0, 0.0, 1.0, 0.0, 1.0]


def test_cast_and_functions(flow):
    r = q("SELECT cast(value AS int) AS vi, sqrt(value) AS sq FROM flow "
          "WHERE id = 4", flow=flow)
    assert r.column("vi").to_pylist() == [40]
    assert abs(r.column("sq").to_pylist()[0] - math.sqrt(40)) < 1e-9


def test_join_inner(flow):
    dims = MessageBatch.from_dict({
        "grp": [1, 2],
        "label": ["one", "two"],
    })
    r = q("SELECT f.id, d.label FROM flow f JOIN dims d ON f.grp = d.grp "
          "ORDER BY f.id", flow=flow, dims=dims)
    assert r.column("id").to_pylist() == [1, 2, 3, 4]
    assert r.column("label").to_strlist() == ["one", "one", "two", "two"]


def test_join_left(flow):
    dims = MessageBatch.from_dict({"grp": [1], "label": ["one"]})
    r = q("SELECT f.id, d.label FROM flow f LEFT JOIN dims d ON f.grp = d.grp "
          "ORDER BY f.id", flow=flow, dims=dims)
    assert r.num_rows == 6
    labels = r.column("label").to_pylist()
    assert labels[0] == b"one" and labels[2] is None


def test_empty_aggregate_returns_row():
    empty = MessageBatch.from_dict({"v": []})
    r = q("SELECT count(*) AS c FROM flow", flow=empty)
    assert r.column("c").to_pylist() == [0]


def test_ddl_rejected():
    with pytest.raises(SqlError):
        parse_sql("DROP TABLE flow")
    with pytest.raises(SqlError):
        parse_sql("INSERT INTO flow VALUES (1)")


def test_unknown_column_errors(flow):
    with pytest.raises(SqlError):
        q("SELECT nope FROM flow", flow=flow)


def test_meta_columns_queryable():
    """Reference lib.rs metadata tests: __meta_* columns are SQL-visible."""
    b = MessageBatch.from_dict({
        "v": [1, 2, 3],
        "__meta_offset": [100, 101, 102],
        "__meta_source": ["k1", "k1", "k2"],
    })
    r = q("SELECT v FROM flow WHERE __meta_offset > 100 "
          "AND __meta_source = 'k1'", flow=b)
    assert r.column("v").to_pylist() == [2]


def test_sql_processor_component(run):
    import asyncio
    from arkflow_amd.processors.sql import SqlProcessor
    p = SqlProcessor({"query": "SELECT id FROM flow WHERE id > 1"})
    b = MessageBatch.from_dict({"id": [1, 2, 3]})
    out = run(p.process(b))
    assert out[0].column("id").to_pylist() == [2, 3]
    # empty in → ProcessResult::None
    assert run(p.process(MessageBatch.from_dict({"id": []}))) == []


def test_scalar_and_aggregate_udf(flow):
    import torch
    from arkflow_amd.sql.udf import register_aggregate_udf, register_scalar_udf
    register_scalar_udf("clip10", lambda t: torch.clamp(
        t if isinstance(t, torch.Tensor) else t.data, max=10.0))
    register_aggregate_udf(
        "sumsq", lambda vals, gid, g: torch.zeros(
            g, dtype=torch.float64).scatter_add_(
                0, gid.long(), vals.double() ** 2))
    r = q("SELECT id, clip10(value) AS c FROM flow ORDER BY id LIMIT 2",
          flow=flow)
    assert r.column("c").to_pylist() == [10.0, 10.0]
    r = q("SELECT grp, sumsq(value) AS s FROM flow GROUP BY grp ORDER BY grp",
          flow=flow)
    assert r.column("s").to_pylist()[0] == 10.0 ** 2 + 20.0 ** 2


def test_offset_and_union_all(flow):
    r = q("SELECT id FROM flow ORDER BY id LIMIT 2 OFFSET 3", flow=flow)
    assert r.column("id").to_pylist() == [4, 5]
    other = MessageBatch.from_dict({"x": [100, 200]})
    r = q("SELECT id FROM flow WHERE id <= 2 UNION ALL SELECT x FROM other",
          flow=flow, other=other)
    assert r.column("id").to_pylist() == [1, 2, 100, 200]


def test_window_functions(flow):
    r = q("SELECT id, row_number() OVER (PARTITION BY grp ORDER BY value) "
          "AS rn FROM flow ORDER BY id", flow=flow)
    # grp1: ids 1(v10),2(v20) → rn 1,2; grp2: 3(v5),4(v40) → 1,2;
    # grp3: 5(v15),6(v20) → 1,2
    assert r.column("rn").to_pylist() == [1, 2, 1, 2, 1, 2]
    r = q("SELECT id, sum(value) OVER (PARTITION BY grp) AS s FROM flow "
          "ORDER BY id", flow=flow)
    assert r.column("s").to_pylist() == [30.0, 30.0, 45.0, 45.0, 35.0, 35.0]
    r = q("SELECT id, rank() OVER (ORDER BY value) AS rk FROM flow "
          "ORDER BY id", flow=flow)
    # values: 10,20,5,40,15,20 → ranks 2,4,1,6,3,4
    assert r.column("rk").to_pylist() == [2, 4, 1, 6, 3, 4]
    r = q("SELECT id, dense_rank() OVER (ORDER BY value) AS dr FROM flow "
          "ORDER BY id", flow=flow)
    assert r.column("dr").to_pylist() == [2, 4, 1, 5, 3, 4]


def test_window_udf():
    """Custom window UDF (reference udf/window_udf.rs): per-partition
    normalized value via the registered callable."""
    import torch
    from arkflow_amd.sql.udf import register_window_udf

    def frac_of_partition(vals, gid, g, perm):
        from arkflow_amd import ops
        totals = ops.segment_reduce(vals.float(), gid.to(torch.int32), g,
                                    "sum")
        return vals.float() / totals[gid]

    register_window_udf("frac_of_partition", frac_of_partition)
    flow = MessageBatch.from_dict({"k": [1, 1, 2], "v": [1.0, 3.0, 10.0]})
    r = q("SELECT v, frac_of_partition(v) OVER (PARTITION BY k) AS f "
          "FROM flow ORDER BY v", flow=flow)
    assert r.column("f").to_pylist() == [0.25, 0.75, 1.0]


def test_null_propagation_strict():
    """Strict SQL NULL semantics on validity columns: NULL predicates drop
    rows, arithmetic over NULL yields NULL, IS NULL / coalesce see through."""
    import torch
    from arkflow_amd.batch import Column, MessageBatch
    c = Column.from_numeric(torch.tensor([1.0, 2.0, 3.0, 4.0]))
    c.validity = torch.tensor([True, False, True, True])
    flow = MessageBatch({"v": c,
                         "w": Column.from_numeric(
                             torch.tensor([10.0, 20.0, 30.0, 40.0]))})
    r = q("SELECT w FROM flow WHERE v > 0", flow=flow)
    assert r.column("w").to_pylist() == [10.0, 30.0, 40.0]  # NULL row dropped
    r = q("SELECT v + w AS s FROM flow", flow=flow)
    assert r.column("s").to_pylist() == [11.0, None, 33.0, 44.0]
    r = q("SELECT coalesce(v, 0 - 1) AS cv, w FROM flow WHERE v IS NULL",
          flow=flow)
    assert r.column("w").to_pylist() == [20.0]
    r = q("SELECT w FROM flow WHERE v IS NOT NULL AND v >= 3", flow=flow)
    assert r.column("w").to_pylist() == [30.0, 40.0]


def test_aggregates_skip_nulls():
    import torch
    from arkflow_amd.batch import Column, MessageBatch
    c = Column.from_numeric(torch.tensor([1.0, 100.0, 3.0, 100.0]))
    c.validity = torch.tensor([True, False, True, False])
    flow = MessageBatch({"v": c, "k": Column.from_numeric(
        torch.tensor([1, 1, 2, 2]))})
    r = q("SELECT k, sum(v) AS s, count(v) AS c, avg(v) AS m, max(v) AS hi "
          "FROM flow GROUP BY k ORDER BY k", flow=flow)
    assert r.column("s").to_pylist() == [1.0, 3.0]
    assert r.column("c").to_pylist() == [1, 1]
    assert r.column("m").to_pylist() == [1.0, 3.0]
    assert r.column("hi").to_pylist() == [1.0, 3.0]


def test_null_group_by_keys_one_group():
    import torch
    from arkflow_amd.batch import Column, MessageBatch
    c = Column.from_numeric(torch.tensor([1, 7, 1, 9], dtype=torch.int64))
    c.validity = torch.tensor([True, False, True, False])
    flow = MessageBatch({"k": c, "v": Column.from_numeric(
        torch.tensor([10., 20., 30., 40.]))})
    r = q("SELECT k, count(*) AS n, sum(v) AS s FROM flow GROUP BY k "
          "ORDER BY n, s", flow=flow)
    # two groups: k=1 (rows 0,2) and NULL (rows 1,3 despite differing 7/9)
    assert r.column("n").to_pylist() == [2, 2]
    assert sorted(r.column("s").to_pylist()) == [40.0, 60.0]


def test_order_by_null_placement():
    import torch
    from arkflow_amd.batch import Column, MessageBatch
    c = Column.from_numeric(torch.tensor([5., 1., 3.]))
    c.validity = torch.tensor([True, False, True])
    flow = MessageBatch({"v": c, "w": Column.from_numeric(
        torch.tensor([1, 2, 3]))})
    r = q("SELECT w FROM flow ORDER BY v", flow=flow)
    assert r.column("w").to_pylist() == [2, 3, 1]  # NULL smallest → first
    r = q("SELECT w FROM flow ORDER BY v DESC", flow=flow)
    assert r.column("w").to_pylist() == [1, 3, 2]  # DESC → NULL last


def test_null_join_keys_never_match():
    import torch
    from arkflow_amd.batch import Column, MessageBatch
    lk = Column.from_numeric(torch.tensor([1, 2, 3], dtype=torch.int64))
    lk.validity = torch.tensor([True, False, True])
    flow = MessageBatch({"k": lk, "a": Column.from_numeric(
        torch.tensor([10, 20, 30]))})
    dims = MessageBatch.from_dict({"k": [1, 2], "label": ["one", "two"]})
    r = q("SELECT a FROM flow f JOIN dims d ON f.k = d.k ORDER BY a",
          flow=flow, dims=dims)
    assert r.column("a").to_pylist() == [10]  # NULL k=2 must not match
    r = q("SELECT a, d.label AS lbl FROM flow f LEFT JOIN dims d "
          "ON f.k = d.k ORDER BY a", flow=flow, dims=dims)
    assert r.column("a").to_pylist() == [10, 20, 30]
    assert r.column("lbl").to_pylist()[1] is None  # unmatched, not 'two'


def test_string_predicate_functions():
    flow = MessageBatch.from_dict({"s": ["apple", "grape", "applesauce"],
                                   "v": [1, 2, 3]})
    r = q("SELECT v FROM flow WHERE starts_with(s, 'app')", flow=flow)
    assert r.column("v").to_pylist() == [1, 3]
    r = q("SELECT v FROM flow WHERE contains(s, 'rap')", flow=flow)
    assert r.column("v").to_pylist() == [2]
    r = q("SELECT v FROM flow WHERE ends_with(s, 'sauce')", flow=flow)
    assert r.column("v").to_pylist() == [3]


def test_order_by_nulls_first_last():
    """Explicit NULLS FIRST/LAST placement (default stays NULL-smallest)."""
    import torch

    from arkflow_amd.batch import MessageBatch
    from arkflow_amd.sql.engine import SqlExecutor

    b = MessageBatch.from_dict({"v": [3.0, 1.0, 2.0, 9.0]})
    b.columns["v"].validity = torch.tensor([True, True, False, True])

    def vals(q):
        out = SqlExecutor(q).execute({"flow": b})
        col = out.column("v")
        vv = col.to_pylist()
        if col.validity is not None:
            return [None if not ok else x
                    for x, ok in zip(vv, col.validity.tolist())]
        return vv

    assert vals("SELECT v FROM flow ORDER BY v NULLS FIRST") == \
        [None, 1.0, 3.0, 9.0]
    assert vals("SELECT v FROM flow ORDER BY v NULLS LAST") == \
        [1.0, 3.0, 9.0, None]
    assert vals("SELECT v FROM flow ORDER BY v DESC NULLS FIRST") == \
        [None, 9.0, 3.0, 1.0]
    assert vals("SELECT v FROM flow ORDER BY v DESC NULLS LAST") == \
        [9.0, 3.0, 1.0, None]
    # default: NULL smallest (ASC → first, DESC → last)
    assert vals("SELECT v FROM flow ORDER BY v") == [None, 1.0, 3.0, 9.0]
    assert vals("SELECT v FROM flow ORDER BY v DESC") == \
        [9.0, 3.0, 1.0, None]


def test_window_lag_lead_first_value():
    from arkflow_amd.batch import MessageBatch
    from arkflow_amd.sql.engine import SqlExecutor

    b = MessageBatch.from_dict({
        "k": [1, 1, 1, 2, 2],
        "v": [10.0, 20.0, 30.0, 5.0, 7.0],
        "t": [1, 2, 3, 1, 2],
    })

    def run(q):
        out = SqlExecutor(q).execute({"flow": b})
        return out

    out = run("SELECT t, k, lag(v) OVER (PARTITION BY k ORDER BY t) AS p "
              "FROM flow ORDER BY k, t")
    col = out.column("p")
    assert col.validity.tolist() == [False, True, True, False, True]
    vals = col.to_pylist()
    assert vals[1:3] == [10.0, 20.0] and vals[4] == 5.0

    out = run("SELECT t, k, lead(v, 1, -1) OVER (PARTITION BY k ORDER BY t) "
              "AS nx FROM flow ORDER BY k, t")
    assert out.column("nx").to_pylist() == [20.0, 30.0, -1.0, 7.0, -1.0]
    assert out.column("nx").validity is None

    out = run("SELECT t, k, first_value(v) OVER (PARTITION BY k ORDER BY t) "
              "AS fv FROM flow ORDER BY k, t")
    assert out.column("fv").to_pylist() == [10.0, 10.0, 10.0, 5.0, 5.0]

    # lag by 2
    out = run("SELECT t, k, lag(v, 2) OVER (PARTITION BY k ORDER BY t) AS p2 "
              "FROM flow ORDER BY k, t")
    assert out.column("p2").validity.tolist() == [False, False, True,
                                                  False, False]
