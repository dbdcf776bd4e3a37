"""Random-walk sampling: outlink validation, walkback, edges, 400 recovery.

Parity map (reference crawl/runner.go):
- pick_walkback_channel: pickWalkbackChannel (runner.go:118-139,
  maxWalkbackAttempts=10 -> WalkbackExhausted)
- standard-mode outlink loop (IsDiscovered / cached-chat-id fast path,
  SearchPublicChat validation with FLOOD_WAIT skip/retire and 400 ->
  invalid cache): runner.go:1310-1383
- walkback decision + forward edge + skipped edges + page_buffer append:
  runner.go:1459-1541 (walkback when no new channels OR
  rand(1..100) <= walkback_rate; forward edges share the chain's
  sequence_id, walkback starts a fresh chain on the NEXT page)
- tandem pending-edge batching: runner.go:1252-1306, 1413-1456
- handle_400_replacement: Handle400Replacement (runner.go:152-284)
"""
from __future__ import annotations

import random
import uuid
from typing import Dict, Optional

from ..feed.client import TelegramAPIError
from ..ops.golden import filter_username
from . import errors as E
from .state import EdgeRecord, Page, RandomWalkStore

MAX_WALKBACK_ATTEMPTS = 10  # runner.go:118


def pick_walkback_channel(sm, source_url: str, exclude: Optional[set],
                          rng=None) -> str:
    exclude = exclude or set()
    rng = rng or random.Random()
    for attempt in range(MAX_WALKBACK_ATTEMPTS):
        url = sm.get_random_discovered_channel(rng)
        if url is None:
            raise E.WalkbackExhausted(source_url)
        if url == source_url or url in exclude:
            continue
        return url
    raise E.WalkbackExhausted(source_url)


def validate_outlinks(links, source_channel: str, sm, rw: RandomWalkStore,
                      client, cfg) -> Dict[str, bool]:
    """Standard-mode outlink loop (runner.go:1310-1383): returns the
    newChannels set. Side effects: discovered-set/seed-cache updates,
    invalid-channel cache inserts on 400."""
    new_channels: Dict[str, bool] = {}
    for link in links:
        name = link.name
        if name == source_channel:
            continue
        ok, _reason = filter_username(name)
        if not ok:
            continue
        if rw.is_invalid_channel(name):
            continue
        cid, cached = sm.get_cached_chat_id(name)
        if sm.is_discovered_channel(name) or cached:
            new_channels[name] = True
            continue
        # validate via SearchPublicChat, FLOOD_WAIT-aware
        try:
            info = client.search_public_chat(name)
        except TelegramAPIError as err:
            msg = str(err)
            secs, is_flood = E.parse_flood_wait_secs(msg)
            if is_flood:
                if secs >= E.FLOOD_WAIT_RETIRE_THRESHOLD_SECS:
                    raise E.FloodWaitRetire(msg)
                continue  # short ban: skip this outlink (runner.go:1338)
            if E.is_tdlib_400(msg):
                rw.mark_invalid_channel(name)
                continue
            continue
        sm.add_discovered_channel(name)
        sm.upsert_seed_channel_chat_id(name, info.chat_id)
        rw.upsert_seed_channel(name, info.chat_id)
        new_channels[name] = True
    return new_channels


def walk_tail(owner: Page, new_channels: Dict[str, bool], sm,
              rw: RandomWalkStore, cfg, rng=None) -> Page:
    """Walkback decision + edge records + page_buffer (runner.go:1459-1541).

    Returns the next page placed in the buffer."""
    rng = rng or random.Random()
    new_channels = dict(new_channels)
    page = Page(
        id=str(uuid.uuid4()), parent_id=owner.id, depth=owner.depth + 1,
        status="unfetched",
    )
    edge = EdgeRecord(source_channel=owner.url, skipped=False)
    walkback = not new_channels
    rnd = None
    if not walkback:
        rnd = rng.randint(1, 100)
    if walkback or cfg.walkback_rate >= rnd:
        edge.walkback = True
        url = pick_walkback_channel(sm, owner.url, set(new_channels), rng)
        page.url = url
        edge.sequence_id = owner.sequence_id
        page.sequence_id = str(uuid.uuid4())  # fresh chain after walkback
    else:
        edge.walkback = False
        names = sorted(new_channels)
        page.url = names[rng.randrange(len(names))]
        del new_channels[page.url]
        edge.sequence_id = owner.sequence_id
        page.sequence_id = owner.sequence_id
    edge.destination_channel = page.url
    edges = [edge]
    for name in new_channels:
        edges.append(EdgeRecord(
            destination_channel=name, source_channel=owner.url,
            skipped=True, walkback=False, sequence_id=owner.sequence_id,
        ))
    rw.add_page(page)
    rw.save_edge_records(edges)
    return page


def walk_tail_fast(owner: Page, names_sorted, sm, rw: RandomWalkStore,
                   cfg, rng, now) -> Page:
    """walk_tail with identical decisions/rng consumption, built for
    the batched GPU hop: takes a pre-SORTED name list, stamps one
    timestamp, builds skipped edges positionally and appends them with
    one bulk call (the per-record path costs ~0.25s per 512-walker hop,
    VERDICT r01 item 5)."""
    page = Page(
        id=str(uuid.uuid4()), parent_id=owner.id, depth=owner.depth + 1,
        status="unfetched",
    )
    src = owner.url
    seq = owner.sequence_id
    walkback = not names_sorted
    rnd = None
    if not walkback:
        rnd = rng.randint(1, 100)
    if walkback or cfg.walkback_rate >= rnd:
        url = pick_walkback_channel(sm, src, set(names_sorted), rng)
        page.url = url
        page.sequence_id = str(uuid.uuid4())  # fresh chain after walkback
        skipped = names_sorted
        edge = EdgeRecord(url, now, src, True, False, seq, "")
    else:
        pick = rng.randrange(len(names_sorted))
        page.url = names_sorted[pick]
        page.sequence_id = seq
        skipped = (names_sorted[:pick] + names_sorted[pick + 1:]
                   if len(names_sorted) > 1 else [])
        edge = EdgeRecord(page.url, now, src, False, False, seq, "")
    # EdgeRecord(destination, discovery_time, source, walkback, skipped,
    # sequence_id, crawl_id); skipped edges land as ONE O(1) block,
    # expanded lazily by the store
    rw.add_page(page)
    rw.save_edge_records_fast([edge])
    rw.save_skipped_edges_block(src, seq, skipped, now)
    return page


def tandem_tail(owner: Page, links, sm, rw: RandomWalkStore, cfg,
                rng=None) -> Optional[str]:
    """Tandem mode (runner.go:1252-1306, 1413-1456): stream pending edges,
    close the batch; forced walkback when nothing was found. Returns the
    batch id (or None after a forced walkback)."""
    rng = rng or random.Random()
    batch_id = None
    for link in links:
        name = link.name
        if name == owner.url:
            continue
        ok, _ = filter_username(name)
        if not ok:
            continue
        if rw.is_invalid_channel(name):
            continue
        if batch_id is None:
            batch_id = rw.open_batch(
                cfg.crawl_id, owner.url, owner.id, owner.depth,
                owner.sequence_id,
            )
        rw.insert_pending_edge(
            batch_id, cfg.crawl_id, name, owner.url, owner.sequence_id,
            link.source_type,
        )
    if batch_id is not None:
        rw.close_batch(batch_id)
        return batch_id
    # forced walkback (runner.go:1427-1456)
    url = pick_walkback_channel(sm, owner.url, None, rng)
    page = Page(
        id=str(uuid.uuid4()), parent_id=owner.id, depth=owner.depth + 1,
        url=url, sequence_id=str(uuid.uuid4()), status="unfetched",
    )
    rw.add_page(page)
    rw.save_edge_records([EdgeRecord(
        destination_channel=url, source_channel=owner.url, walkback=True,
        skipped=False, sequence_id=owner.sequence_id,
    )])
    return None


def handle_400_replacement(sm, rw: RandomWalkStore, p: Page, cfg,
                           rng=None) -> None:
    """Handle400Replacement (runner.go:152-284)."""
    rng = rng or random.Random()
    channel = p.url
    seq = p.sequence_id
    rw.mark_invalid_channel(channel)
    rw.mark_seed_channel_invalid(channel)
    edge = rw.get_edge_record(seq, channel)
    rw.delete_edge_record(seq, channel)

    def walkback_from(source: str):
        url = pick_walkback_channel(sm, source, {p.url}, rng)
        repl = Page(
            id=str(uuid.uuid4()), parent_id=p.parent_id,
            depth=p.depth, url=url, sequence_id=str(uuid.uuid4()),
            status="unfetched",
        )
        rw.add_page(repl)
        rw.save_edge_records([EdgeRecord(
            destination_channel=url, source_channel=source, walkback=True,
            skipped=False, sequence_id=seq,
        )])

    if edge is None:
        if sm.is_seed_channel(channel):
            # seed replacement (handle400SeedReplacement,
            # runner.go:263-284): a random VALID seed channel becomes
            # the new page — fresh sequence chain, NO edge record
            seed_url = rw.get_random_seed_channel(rng)
            if seed_url is None:
                raise E.WalkbackExhausted(channel)
            rw.add_page(Page(
                id=str(uuid.uuid4()), parent_id=p.parent_id,
                depth=p.depth, url=seed_url,
                sequence_id=str(uuid.uuid4()), status="unfetched",
            ))
            return
        walkback_from(channel)
        return
    if edge.walkback:
        walkback_from(edge.source_channel)
        return
    skipped = rw.get_random_skipped_edge(
        {p.url}, rng, sequence_id=seq, source_channel=edge.source_channel
    )
    if skipped is None:
        walkback_from(edge.source_channel)
        return
    rw.promote_edge(seq, skipped.destination_channel)
    repl = Page(
        id=str(uuid.uuid4()), parent_id=p.parent_id, depth=p.depth,
        url=skipped.destination_channel, sequence_id=seq,
        status="unfetched",
    )
    rw.add_page(repl)
