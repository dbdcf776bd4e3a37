"""Golden batch pipeline: packed batch -> JSONL lines + outlinks (CPU oracle).

This is the per-batch equivalent of the reference's per-message hot loop
(crawl/runner.go:1110-1550 processAllMessagesWithProcessor +
telegramhelper.ParseMessage): decode each packed record, assemble the Post,
emit its JSONL line and its discovered outlinks. The HIP parse+encode kernel
(ops/csrc) must produce byte-identical JSONL and the same outlink sets.
"""
from __future__ import annotations

import datetime as _dt
from typing import List, Optional, Tuple

from . import batch as B
from . import golden as G


def encode_batch(
    batch: B.MessageBatch,
    now: Optional[_dt.datetime] = None,
    skip_media: bool = True,
    min_post_date: Optional[_dt.datetime] = None,
) -> Tuple[List[bytes], List[List[Tuple[str, str]]]]:
    """Returns (jsonl_lines, per-message outlink (name, source) lists)."""
    now = now or _dt.datetime.now(_dt.timezone.utc)
    lines: List[bytes] = []
    links: List[List[Tuple[str, str]]] = []
    channels = [B.channel_row(batch, c) for c in range(batch.n_channels)]
    for i in range(batch.n):
        msg = B.unpack_message(batch, i)
        ch = channels[int(batch.meta["channel_idx"][i])]
        comments = B.unpack_comments(batch, i)
        post = G.parse_message(
            msg,
            channel_username=ch.username,
            chat_title=ch.title,
            member_count=ch.member_count,
            post_count=ch.post_count,
            total_views=ch.total_views,
            comments=comments,
            min_post_date=min_post_date,
            skip_media=skip_media,
            now=now,
        )
        if post is None:
            lines.append(b"")
            links.append([])
            continue
        lines.append(post.to_jsonl().encode("utf-8"))
        links.append(
            [(l.name, l.source_type) for l in G.extract_links_with_source(msg)]
        )
    return lines, links
