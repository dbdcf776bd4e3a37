"""YouTube platform crawler plugin.

Parity: crawler/youtube/youtube_crawler.go (sampling dispatch,
conversion pool) behind the registry interface of crawler/crawler.go.
"""
from __future__ import annotations

import sys

from ..registry import CrawlContext, PlatformCrawler


class YouTubeCrawler(PlatformCrawler):
    def platform_type(self) -> str:
        return "youtube"

    def run(self, ctx: CrawlContext) -> dict:
        from ..youtube.runner import run_youtube

        stats = run_youtube(ctx.cfg, ctx.urls)
        print(f"youtube crawl complete: {stats}", file=sys.stderr)
        return stats


def register(registry) -> None:
    registry.register("youtube", YouTubeCrawler)
