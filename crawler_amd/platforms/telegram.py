"""Telegram platform crawler plugin.

Owns the standalone / random-walk / tandem / GPU execution modes for the
synthetic Telegram feed (the reference's crawler/telegram/ plugin +
standalone/dapr runner bodies, behind the crawler/crawler.go registry
interface). The CLI resolves this via the registry — platform dispatch
is a lookup, not an if/else ladder.
"""
from __future__ import annotations

import os
import sys

from ..registry import CrawlContext, PlatformCrawler


class TelegramCrawler(PlatformCrawler):
    def __init__(self):
        self.pool = None
        self.chunker = None

    def platform_type(self) -> str:
        return "telegram"

    def initialize(self, ctx: CrawlContext) -> None:
        from ..feed.client import ConnectionPool

        cfg, args = ctx.cfg, ctx.args
        db_urls = [u for u in args.tdlib_database_urls.split(",") if u]
        pool_size = len(db_urls) if db_urls else args.pool_size
        self.pool = ConnectionPool(
            ctx.feed, pool_size, cfg.rate_limit,
            posts_per_channel=args.synthetic_posts,
            disable_rate_limits=cfg.disable_rate_limits,
        )

    def run(self, ctx: CrawlContext) -> dict:
        from ..engine import LocalStateManager, RandomWalkStore
        from ..engine.runner import RandomWalkRunner, StandaloneRunner

        cfg, args, urls = ctx.cfg, ctx.args, ctx.urls
        feed = ctx.feed
        if self.pool is None:
            self.initialize(ctx)
        sm = LocalStateManager(cfg)

        if cfg.combine_files:
            # CombineFiles mode (dapr/standalone.go:253-271): posts go
            # through the temp->watch combiner; combined files land
            # under the crawl's combined/ dir (the upload-binding mock)
            import shutil

            from ..engine.chunker import Chunker

            combined_dir = os.path.join(cfg.storage_root, cfg.crawl_id,
                                        "combined")
            os.makedirs(combined_dir, exist_ok=True)
            self.chunker = Chunker(
                cfg.combine_temp_dir, cfg.combine_watch_dir,
                cfg.combine_write_dir,
                upload=lambda p: shutil.copy(p, combined_dir),
                trigger_bytes=cfg.combine_trigger_size * 1024 * 1024,
                hard_cap_bytes=cfg.combine_hard_cap * 1024 * 1024,
            )
            self.chunker.verify_cleanup()  # crash recovery at startup
            self.chunker.start()
            sm.attach_chunker(self.chunker)

        if cfg.validate_only:
            from ..engine.validator import run_validation_loop

            rw = RandomWalkStore()
            run_validation_loop(cfg, sm, rw)
            return {"mode": "validate-only"}

        if args.gpu and cfg.sampling_method in ("channel", "snowball"):
            # MI355X execution mode: whole layers through the HIP kernels
            from ..engine.gpu_runner import GpuCrawlEngine

            engine = GpuCrawlEngine(
                cfg, sm, feed, posts_per_channel=args.synthetic_posts)
            stats = engine.run(urls)
            print(f"gpu crawl complete: {stats}", file=sys.stderr)
            return stats

        if cfg.sampling_method == "random-walk" and args.gpu:
            # MI355X execution mode: batched walker hops through the
            # HIP pipeline, pipelined two-pool form (the production
            # default; engine/gpu_randomwalk.py)
            from ..engine.gpu_randomwalk import GpuRandomWalk

            rw = RandomWalkStore()
            eng = GpuRandomWalk(
                cfg, sm, rw, feed,
                posts_per_hop=args.synthetic_posts,
                walkers=max(cfg.concurrency, 16),
            )
            if cfg.seed_size and not urls:
                urls = [feed.username_of(i)
                        for i in range(cfg.seed_size)]
            eng.seed(urls)
            stats = eng.run(max_pages=cfg.max_pages, pipelined=True)
            print(f"gpu random-walk complete: {stats}", file=sys.stderr)
            return stats

        if cfg.sampling_method == "random-walk":
            rw = RandomWalkStore()
            runner = RandomWalkRunner(cfg, sm, rw, self.pool)
            if cfg.seed_size and not urls:
                urls = [feed.username_of(i)
                        for i in range(cfg.seed_size)]
            runner.seed(urls)
            vthread = stop_evt = None
            if cfg.tandem_crawl:
                # The reference runs validator PODS against the shared
                # Postgres (dapr/standalone.go:276-314); single-node
                # analog: an in-process validator thread over the same
                # store (identical claim semantics).
                import threading

                from ..engine.validator import TandemValidator
                from ..feed.client import TokenBucket
                from ..feed.tme import MockTMe

                rl = None
                if not cfg.disable_rate_limits:
                    rl = TokenBucket(cfg.validator_request_rate * 60,
                                     100)
                v = TandemValidator(
                    cfg, sm, rw,
                    fetcher=MockTMe(universe=args.synthetic_universe),
                    rate_limiter=rl,
                )
                v.recover()
                stop_evt = threading.Event()

                def vloop():
                    while not stop_evt.is_set():
                        n = v.pump_edges()
                        b = v.pump_walkback()
                        if not n and not b:
                            stop_evt.wait(0.01)

                vthread = threading.Thread(target=vloop, daemon=True,
                                           name="tandem-validator")
                vthread.start()
            try:
                stats = runner.run(
                    max_seconds=cfg.max_crawl_duration_s or None
                )
            finally:
                if vthread is not None:
                    stop_evt.set()
                    vthread.join(timeout=10)
        else:
            runner = StandaloneRunner(cfg, sm, self.pool)
            stats = runner.run(urls)
        print(f"crawl complete: {stats}", file=sys.stderr)
        return stats

    def close(self) -> None:
        if self.chunker is not None:
            self.chunker.stop()
            self.chunker = None


def register(registry) -> None:
    registry.register("telegram", TelegramCrawler)
