"""crawler_amd.parallel — intra-node distribution.

The reference's Dapr pub/sub fan-out (orchestrator -> worker topics over
Redis, distributed/pubsub.go) is replaced MI355X-natively by:

- a control plane over torch.distributed.TCPStore: atomic-counter work
  queues (StoreQueue) carrying WorkItem/WorkResult/Status records between
  the orchestrator rank and worker ranks — small, latency-tolerant,
  GPU-agnostic;
- a data plane over RCCL/xGMI process groups: the seen-channel set is
  merged by all-gathering newly-claimed hashes each round (see bench.py
  and ops/gpu.SeenSet) and bulk JSONL stays on-GPU until the host spill.
"""
from .messages import (  # noqa: F401
    ControlMessage,
    StatusMessage,
    WorkItem,
    WorkResult,
    new_trace_id,
)
from .queue import StoreQueue, Heartbeats  # noqa: F401
