"""YouTube crawl runner: sampling dispatch + post conversion + storage.

Parity: crawler/youtube/youtube_crawler.go FetchMessages sampling dispatch
(:245-443) + RunRandomYoutubeSample loop (dapr/standalone.go:1175-1243)
with 10-worker post conversion (youtube_crawler.go:354-427).
"""
from __future__ import annotations

import concurrent.futures as cf
import random
from typing import List

from ..engine.state import LocalStateManager
from .client import QuotaExceeded, SyntheticYouTubeClient
from .convert import convert_video_to_post
from .synth import SyntheticYouTubeIndex


def run_youtube(cfg, urls: List[str], client=None, sm=None,
                limit: int = 0) -> dict:
    """Dispatch by cfg.sampling_method: channel | random | snowball."""
    client = client or SyntheticYouTubeClient(
        SyntheticYouTubeIndex(seed=getattr(cfg, "synthetic_seed", 99)),
        min_channel_videos=cfg.min_channel_videos,
        rng=random.Random(getattr(cfg, "synthetic_seed", 99)),
    )
    sm = sm or LocalStateManager(cfg)
    limit = limit or (cfg.max_posts if cfg.max_posts > 0 else 100)

    if cfg.sampling_method == "random":
        videos = client.get_random_videos(limit)
    elif cfg.sampling_method == "snowball":
        seeds = [u for u in urls]
        # seed "urls" may be channel indices or UC ids; resolve indices
        resolved = []
        for s in seeds:
            if s.startswith("UC"):
                resolved.append(s)
            else:
                resolved.append(client.index.channel_id_of(int(s)))
        videos = client.get_snowball_videos(resolved, limit,
                                            max_depth=max(cfg.max_depth, 1))
    else:  # channel
        videos = []
        for s in urls:
            cid = s if s.startswith("UC") else client.index.channel_id_of(
                int(s)
            )
            client.get_channel_info(cid)
            videos.extend(client.get_channel_videos(cid, limit))

    # date window (FetchYoutubeChannelInfoAndVideos passes the
    # CalculateDateFilters range into the crawl job,
    # dapr/standalone.go:1092-1147): strict precedence, not combined
    from ..config import calculate_date_filters

    from_t, to_t = calculate_date_filters(cfg)
    if from_t is not None:
        videos = [v for v in videos
                  if v.published_at is not None
                  and from_t <= v.published_at <= to_t]

    # 10-worker post conversion pool (youtube_crawler.go:354)
    def conv(v):
        ch = None
        try:
            ch = client.get_channel_info(v.channel_id)
        except QuotaExceeded:
            pass
        # panic containment per video (youtube panic_test.go analog):
        # one malformed video never kills the conversion pool
        try:
            return convert_video_to_post(v, ch, crawl_label=cfg.crawl_label)
        except Exception:
            return None

    with cf.ThreadPoolExecutor(max_workers=10) as ex:
        posts = [p for p in ex.map(conv, videos) if p is not None]
    for p in posts:
        sm.store_post(p.channel_id, p)
    sm.save_state()
    sm.close()
    return {
        "videos": len(videos),
        "posts": len(posts),
        "quota_used": client.quota_used,
        **client.stats,
    }
