"""User-defined SQL functions.

Mirrors reference crates/arkflow-plugin/src/udf/{scalar,aggregate,window}_udf.rs:
globally registered UDFs become available to every SQL statement. Scalar UDFs
receive torch tensors (device-resident on GPU → the UDF body is tensor
kernels); aggregate UDFs receive (values, group_ids, num_groups).
"""
from __future__ import annotations

from typing import Callable, Dict

_SCALAR_UDFS: Dict[str, Callable] = {}
_AGGREGATE_UDFS: Dict[str, Callable] = {}
_WINDOW_UDFS: Dict[str, Callable] = {}


def register_scalar_udf(name: str, fn: Callable) -> None:
    """fn(*tensors) -> tensor; name is matched case-insensitively."""
    _SCALAR_UDFS[name.lower()] = fn


def register_aggregate_udf(name: str, fn: Callable) -> None:
    """fn(values, group_ids, num_groups) -> tensor[num_groups]."""
    _AGGREGATE_UDFS[name.lower()] = fn


def scalar_udf(name: str):
    return _SCALAR_UDFS.get(name.lower())


def aggregate_udf(name: str):
    return _AGGREGATE_UDFS.get(name.lower())


def register_window_udf(name: str, fn: Callable) -> None:
    """fn(values, group_ids, num_groups, order_perm) -> tensor[n] per-row.
    values is the evaluated first argument (or None for no-arg functions),
    group_ids the partition id per row, order_perm the row permutation that
    sorts by (partition, ORDER BY keys)."""
    _WINDOW_UDFS[name.lower()] = fn


def window_udf(name: str):
    return _WINDOW_UDFS.get(name.lower())
