"""crawler_amd — an MI355X-native distributed crawl/ingest engine.

A from-scratch rebuild of the capabilities of
researchaccelerator-hub/distributed-crawler (a Go/Dapr/TDLib Telegram+YouTube
crawler) for a single 8x MI355X node:

- the per-post hot path (message parse, t.me link extraction, username
  filtering, JSONL encode, channel dedup) runs as hand-written CDNA4 HIP
  kernels over packed record batches (``crawler_amd.ops``);
- the crawl frontier / random-walk state / seen-channel set live in device
  memory and are merged across GPUs with RCCL collectives over xGMI
  (``crawler_amd.parallel``);
- live platform APIs are replaced by a deterministic synthetic feed engine
  (``crawler_amd.feed``) with TDLib-shaped semantics (cache/server latency
  classes, FLOOD_WAIT and 400 injection, entity-annotated text).

Behavioral contracts preserved from the reference:
CLI flag surface (reference main.go:751-812), the unified JSONL Post schema
(reference model/data.go:9-149), progress.json checkpointing, and the
sampling-method semantics (channel / snowball / random-walk / random).
"""

__version__ = "0.1.0"
