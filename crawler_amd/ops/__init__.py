"""crawler_amd.ops — the per-post hot path.

Two implementations of every op:

- ``golden``: pure-Python reference semantics (link extraction, UTF-16
  offsets, username filter, message->Post assembly, JSONL encode). Used by
  the CPU execution path (BASELINE config #1) and as the oracle for kernel
  numerics tests.
- ``gpu``: hand-written CDNA4 HIP kernels over packed record batches
  (``crawler_amd.ops.csrc``), loaded via :func:`load_hip_ext`. On a GPU box
  the HIP extension is REQUIRED — ops raise if it cannot be loaded rather
  than silently falling back to eager Python.
"""
from .golden import (  # noqa: F401
    DiscoveredLink,
    Entity,
    FormattedText,
    SynthMessage,
    build_telegram_link_and_message_id,
    extract_channel_links,
    extract_links_with_source,
    filter_username,
    parse_message,
    utf16_offset_to_bytes,
)
