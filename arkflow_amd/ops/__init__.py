"""Physical op dispatch: hand-written HIP kernels on GPU, torch on CPU.

Every hot physical operator of the SQL/inference path routes through here.
On a CUDA(=ROCm) device the native gfx950 extension (arkflow_amd._native,
built from csrc/) is REQUIRED — if it is missing we raise
GpuExtensionMissing instead of silently falling back to eager PyTorch, so a
GPU test can never pass on a non-native path. On CPU, torch reference
implementations keep CI green and serve as the numerics oracle.
"""
from __future__ import annotations
from typing import Optional, Tuple

import torch

from ..errors import GpuExtensionMissing

_native = None
_native_err: Optional[str] = None


def _load_native():
    global _native, _native_err
    if _native is not None or _native_err is not None:
        return _native
    try:
        from .. import _native as mod  # built in-tree by setup.py build_ext
        _native = mod
    except ImportError as e:
        _native_err = str(e)
    return _native


def native_available() -> bool:
    return _load_native() is not None


def require_native():
    mod = _load_native()
    if mod is None:
        raise GpuExtensionMissing(
            f"arkflow_amd._native HIP extension not built ({_native_err}); "
            "run `python setup.py build_ext --inplace` (gfx950)"
        )
    return mod


def _use_native(t: torch.Tensor) -> bool:
    if t.is_cuda:
        require_native()  # fail loudly on GPU without the extension
        return True
    return False


# --------------------------------------------------------------------- filter
def mask_to_indices(mask: torch.Tensor) -> torch.Tensor:
    """Order-preserving stream compaction: bool[n] → int32 indices of set rows.

    GPU: two-pass block-count + scan + scatter HIP kernel (csrc/filter.hip),
    replacing the reference's DataFusion FilterExec (processor/sql.rs:107).
    """
    if _use_native(mask):
        return require_native().mask_to_indices(mask)
    return torch.nonzero(mask, as_tuple=False).flatten().to(torch.int32)


def filter_cmp_scalar(col: torch.Tensor, op: str, scalar: float) -> torch.Tensor:
    """Fused compare+compact for the common `WHERE col OP literal` shape:
    returns indices directly without materializing the mask."""
    if _use_native(col):
        opi = {"<": 0, "<=": 1, ">": 2, ">=": 3, "=": 4, "!=": 5}[op]
        return require_native().filter_cmp_scalar(col, opi, float(scalar))
    mask = {
        "<": col < scalar, "<=": col <= scalar, ">": col > scalar,
        ">=": col >= scalar, "=": col == scalar, "!=": col != scalar,
    }[op]
    return torch.nonzero(mask, as_tuple=False).flatten().to(torch.int32)


def gather(col: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
    """Row gather. GPU: coalesced gather kernel handling 1/2/4/8-byte elems."""
    if _use_native(col):
        return require_native().gather(col, indices)
    return col[indices.long()]


# ------------------------------------------------------------------ aggregate
def hash_group(keys: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, int]:
    """Group rows by an integer key column.

    Returns (group_ids int32[n], unique_keys[g], num_groups). GPU: LDS-tiled
    open-addressing hash build (csrc/hash_agg.hip) — replaces DataFusion's
    hash-aggregate physical operator.
    """
    if _use_native(keys):
        gid, uniq = require_native().hash_group_i64(keys.to(torch.int64))
        return gid, uniq, int(uniq.shape[0])
    uniq, inverse = torch.unique(keys, return_inverse=True)
    return inverse.to(torch.int32), uniq, int(uniq.shape[0])


def segment_reduce(values: torch.Tensor, group_ids: torch.Tensor,
                   num_groups: int, op: str) -> torch.Tensor:
    """Per-group reduction: op in sum|min|max|count|mean."""
    gid = group_ids.long()
    if op == "count":
        if values.is_cuda:
            # int64 scatter_add on GPU is a CAS loop — catastrophic under
            # high key contention; the native f32 LDS-buffered sum is exact
            # for counts < 2^24 per group and orders of magnitude faster.
            ones = torch.ones(values.shape[0], dtype=torch.float32,
                              device=values.device)
            return segment_reduce(ones, group_ids, num_groups,
                                  "sum").to(torch.int64)
        out = torch.zeros(num_groups, dtype=torch.int64, device=values.device)
        out.scatter_add_(0, gid, torch.ones_like(gid))
        return out
    v = values.to(torch.float64) if values.dtype in (
        torch.float16, torch.bfloat16) else values
    if _use_native(values) and op in ("sum", "min", "max") \
            and values.dtype == torch.float32:
        opi = {"sum": 0, "min": 1, "max": 2}[op]
        return require_native().segment_reduce_f32(
            values, group_ids, num_groups, opi)
    if op == "sum":
        out = torch.zeros(num_groups, dtype=v.dtype, device=v.device)
        out.scatter_add_(0, gid, v)
        return out
    if op == "mean":
        s = segment_reduce(values, group_ids, num_groups, "sum")
        c = segment_reduce(values, group_ids, num_groups, "count")
        return s.double() / c.double()
    if op in ("min", "max"):
        init = float("inf") if op == "min" else float("-inf")
        out = torch.full((num_groups,), init, dtype=torch.float64,
                         device=v.device)
        out.scatter_reduce_(0, gid, v.double(), reduce="amin" if op == "min"
                            else "amax", include_self=True)
        return out
    raise ValueError(f"unknown segment op {op!r}")


# ----------------------------------------------------------------------- sort
def sort_indices(key: torch.Tensor, ascending: bool = True) -> torch.Tensor:
    """Stable argsort. GPU: native LSD radix sort (csrc/radix_sort.hip) for
    f32/i64/i32 keys; torch elsewhere."""
    if _use_native(key) and key.dtype in (torch.float32, torch.int64,
                                          torch.int32):
        return require_native().radix_argsort(
            key.contiguous(), not ascending).long()
    return torch.argsort(key, stable=True, descending=not ascending)


# ----------------------------------------------------------------------- join
def join_inner(left_keys: torch.Tensor, right_keys: torch.Tensor
               ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Inner equi-join: returns (left_idx, right_idx) pairs.

    GPU: build/probe hash-join kernel (csrc/hash_join.hip). CPU fallback:
    sort-merge via searchsorted (handles duplicate keys on both sides).
    """
    if _use_native(left_keys):
        return require_native().join_inner_i64(
            left_keys.to(torch.int64), right_keys.to(torch.int64))
    return _join_sort_merge(left_keys, right_keys)


def _join_sort_merge(lk: torch.Tensor, rk: torch.Tensor):
    r_sorted, r_order = torch.sort(rk, stable=True)
    lo = torch.searchsorted(r_sorted, lk, side="left")
    hi = torch.searchsorted(r_sorted, lk, side="right")
    counts = (hi - lo).clamp(min=0)
    total = int(counts.sum().item())
    if total == 0:
        e = torch.empty(0, dtype=torch.int64, device=lk.device)
        return e, e.clone()
    l_idx = torch.repeat_interleave(
        torch.arange(lk.shape[0], device=lk.device), counts)
    offs = torch.repeat_interleave(lo, counts)
    within = torch.arange(total, device=lk.device) - torch.repeat_interleave(
        torch.cumsum(counts, 0) - counts, counts)
    r_idx = r_order[offs + within]
    return l_idx, r_idx


def join_left(left_keys: torch.Tensor, right_keys: torch.Tensor
              ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Left join: right_idx is -1 for unmatched left rows."""
    l_idx, r_idx = join_inner(left_keys, right_keys)
    matched = torch.zeros(left_keys.shape[0], dtype=torch.bool,
                          device=left_keys.device)
    matched[l_idx] = True
    un = torch.nonzero(~matched).flatten()
    l_all = torch.cat([l_idx, un])
    r_all = torch.cat([r_idx, torch.full_like(un, -1)])
    order = torch.argsort(l_all, stable=True)
    return l_all[order], r_all[order]
