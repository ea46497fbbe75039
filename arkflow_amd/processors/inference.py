"""`inference` processor: native streaming ML inference on device-resident
batches — the component the reference only promises through its python
processor (reference README.md:16-18 "ML inference"; processor/python.rs is
the escape hatch). Models: `mlp_anomaly` (per-row score) and `bert_base`
(per-sequence logits, rows chunked into seq_len-token sequences).
"""
from __future__ import annotations

from typing import List

import torch

from ..batch import Column, MessageBatch
from ..registry import register
from ..spi import Processor


class InferenceProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.model_name = config.get("model", "mlp_anomaly")
        self.device = torch.device(config["device"]) if config.get("device") \
            else getattr(resource, "device", torch.device("cpu"))
        self.columns = config.get("columns")  # None = all numeric non-meta
        self.output_column = config.get("output_column", "score")
        self.seed = int(config.get("seed", 1234))
        if self.model_name in ("mlp", "mlp_anomaly"):
            from ..models.mlp import MlpAnomalyDetector
            self.in_features = int(config.get("in_features", 0))
            hidden = config.get("hidden", [256, 256])
            self._mlp_hidden = [int(h) for h in hidden]
            self.model = None  # lazy: in_features can come from first batch
            if self.in_features:
                self.model = MlpAnomalyDetector(
                    self.in_features, self._mlp_hidden, self.device, self.seed)
        elif self.model_name in ("bert", "bert_base"):
            from ..models.bert import BertConfig, BertEncoder
            cfg = BertConfig(
                layers=int(config.get("layers", 12)),
                hidden=int(config.get("hidden_size", 768)),
                heads=int(config.get("heads", 12)),
                ff=int(config.get("ff", 3072)),
                seq_len=int(config.get("seq_len", 128)),
                num_labels=int(config.get("num_labels", 2)),
            )
            self.token_column = config.get("token_column", "token")
            self.model = BertEncoder(cfg, self.device, self.seed)
            self.seq_len = cfg.seq_len
        else:
            from ..errors import ConfigError
            raise ConfigError(f"unknown inference model {self.model_name!r}")

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if batch.num_rows == 0:
            return []
        if self.model_name in ("mlp", "mlp_anomaly"):
            return [self._process_mlp(batch)]
        return [self._process_bert(batch)]

    # --------------------------------------------------------------- mlp path
    def _feature_columns(self, batch: MessageBatch) -> List[str]:
        if self.columns:
            return self.columns
        return [
            n for n, c in batch.columns.items()
            if c.kind == "numeric" and not n.startswith("__meta_")
            and c.data.dtype.is_floating_point
        ]

    def _process_mlp(self, batch: MessageBatch) -> MessageBatch:
        names = self._feature_columns(batch)
        feats = torch.stack(
            [batch.column(n).data.to(self.device, torch.float32)
             for n in names], dim=1)
        if self.model is None:
            from ..models.mlp import MlpAnomalyDetector
            self.model = MlpAnomalyDetector(
                feats.shape[1], self._mlp_hidden, self.device, self.seed)
        scores = self.model.forward(feats)
        return batch.with_columns({self.output_column: Column("numeric",
                                                              scores)})

    # -------------------------------------------------------------- bert path
    def _process_bert(self, batch: MessageBatch) -> MessageBatch:
        tokens = batch.column(self.token_column).data.to(
            self.device, torch.int64)
        n_seq = tokens.shape[0] // self.seq_len
        if n_seq == 0:
            # short batch: pad one sequence with zeros
            pad = self.seq_len - tokens.shape[0]
            tokens = torch.nn.functional.pad(tokens, (0, pad))
            n_seq = 1
        ids = tokens[: n_seq * self.seq_len].reshape(n_seq, self.seq_len)
        logits = self.model.forward(ids)  # [n_seq, num_labels]
        cols = {
            f"logit_{i}": Column("numeric", logits[:, i].contiguous())
            for i in range(logits.shape[1])
        }
        cols[self.output_column] = Column(
            "numeric", torch.argmax(logits, dim=1).to(torch.int64))
        out = MessageBatch(cols, input_name=batch.input_name)
        return out


@register("processor", "inference",
          description="Native ML inference (mlp_anomaly per-row score; "
                      "bert_base per-sequence logits) — bf16 MFMA kernels",
          example={"type": "inference", "model": "mlp_anomaly",
                   "columns": ["f0", "f1"], "output_column": "score"})
def _build_inference(config: dict, resource=None) -> InferenceProcessor:
    return InferenceProcessor(config, resource)
