"""`python` processor: call a user Python function per batch.

Mirrors reference crates/arkflow-plugin/src/processor/python.rs:47-98 (PyO3 +
pyarrow zero-copy, run under spawn_blocking). Here the engine itself is
Python, so the function is imported (module:function) or compiled from
inline `code`, receives the batch (as a pyarrow Table when convert="pyarrow",
else as a MessageBatch) and returns a batch / table / dict / list thereof.
CPU-bound user code runs in a thread executor to keep the stream loop live.
"""
from __future__ import annotations

import asyncio
import importlib
from typing import List

from ..batch import Column, MessageBatch
from ..errors import ConfigError, ProcessError
from ..registry import register
from ..spi import Processor


def _to_pyarrow(batch: MessageBatch):
    import pyarrow as pa
    import torch
    arrays = {}
    for name, col in batch.columns.items():
        if col.kind == "numeric":
            t = col.data.detach().cpu()
            if t.dtype.is_floating_point and t.dtype not in (
                    torch.float32, torch.float64):
                t = t.float()
            arrays[name] = pa.array(t.numpy())
        else:
            arrays[name] = pa.array(
                [v.decode("utf-8", "replace") if v is not None else None
                 for v in col.to_pylist()])
    return pa.table(arrays)


def _from_any(obj, input_name=None) -> MessageBatch:
    import pyarrow as pa
    if isinstance(obj, MessageBatch):
        return obj
    if isinstance(obj, pa.Table):
        cols = {}
        for name in obj.column_names:
            arr = obj.column(name).combine_chunks()
            if pa.types.is_string(arr.type) or pa.types.is_binary(arr.type):
                cols[name] = Column.from_strings(arr.to_pylist())
            else:
                import numpy as np
                cols[name] = Column.from_numeric(
                    np.asarray(arr.to_numpy(zero_copy_only=False)))
        return MessageBatch(cols, input_name)
    if isinstance(obj, dict):
        return MessageBatch.from_dict(obj, input_name)
    raise ProcessError(f"python processor returned {type(obj).__name__}")


class PythonProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.convert = config.get("convert", "batch")  # batch | pyarrow
        if config.get("code"):
            ns: dict = {}
            exec(config["code"], ns)  # noqa: S102 — user-supplied, like VRL
            fn_name = config.get("function", "process")
            self.fn = ns.get(fn_name)
            if self.fn is None:
                raise ConfigError(f"code does not define {fn_name}()")
        elif config.get("module"):
            mod = importlib.import_module(config["module"])
            self.fn = getattr(mod, config.get("function", "process"))
        else:
            raise ConfigError("python processor requires 'code' or 'module'")

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows == 0:
            return []
        arg = _to_pyarrow(batch) if self.convert == "pyarrow" else batch
        loop = asyncio.get_running_loop()
        result = await loop.run_in_executor(None, self.fn, arg)
        if result is None:
            return []
        if isinstance(result, (list, tuple)):
            return [_from_any(r, batch.input_name) for r in result]
        return [_from_any(result, batch.input_name)]


@register("processor", "python",
          description="Run a user Python function per batch "
                      "(module:function or inline code; pyarrow convert)",
          example={"type": "python", "code":
                   "def process(batch):\n    return batch"})
def _build_python(config: dict, resource=None) -> PythonProcessor:
    return PythonProcessor(config, resource)
