"""`repartition` processor: RCCL all-to-all key repartitioning.

Shards keyed state across the node's GPUs (BASELINE config 4: session-window
+ join_buffer repartition via RCCL all-to-all over xGMI). Every rank's stream
must process batches at the same cadence — this is a collective. Single-rank
runs pass batches through unchanged, so configs are rank-count agnostic.
"""
from __future__ import annotations

from typing import List

from ..batch import MessageBatch
from ..errors import ConfigError
from ..parallel import dist as pdist
from ..registry import register
from ..spi import Processor


class RepartitionProcessor(Processor):
    def __init__(self, config: dict, resource=None):
        self.key = config.get("key")
        if not self.key:
            raise ConfigError("repartition requires 'key'")

    async def process(self, batch: MessageBatch) -> List[MessageBatch]:
        if pdist.world_size() <= 1:
            return [batch] if batch.num_rows else []
        out = pdist.repartition_by_key(batch, self.key)
        return [out]


@register("processor", "repartition",
          description="Hash-repartition rows across GPUs (RCCL all-to-all "
                      "over xGMI) so equal keys colocate",
          example={"type": "repartition", "key": "session_id"})
def _build_repartition(config: dict, resource=None) -> RepartitionProcessor:
    return RepartitionProcessor(config, resource)
