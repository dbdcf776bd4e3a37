"""Sampling-method validation matrix (reference
common/sampling_validation.go:19-66 + main_test.go coverage)."""
import pytest

from crawler_amd.config import validate_sampling_method


def ok(**kw):
    validate_sampling_method(**kw)


def bad(match, **kw):
    with pytest.raises(ValueError, match=match):
        validate_sampling_method(**kw)


def test_platform_method_support_matrix():
    ok(platform="telegram", sampling_method="channel", url_list=["a"])
    ok(platform="telegram", sampling_method="snowball", url_list=["a"])
    ok(platform="telegram", sampling_method="random-walk", seed_size=5)
    ok(platform="youtube", sampling_method="channel", url_list=["a"])
    ok(platform="youtube", sampling_method="random")
    ok(platform="youtube", sampling_method="snowball", url_list=["a"])
    bad("not supported", platform="telegram", sampling_method="random")
    bad("not supported", platform="youtube",
        sampling_method="random-walk")
    bad("unsupported platform", platform="tiktok",
        sampling_method="channel")


def test_random_walk_xor_seed_sources():
    # neither
    bad("not both or neither", platform="telegram",
        sampling_method="random-walk")
    # both
    bad("not both or neither", platform="telegram",
        sampling_method="random-walk", url_list=["a"], seed_size=5)
    # url-file counts as a source
    ok(platform="telegram", sampling_method="random-walk",
       url_file="/tmp/urls.txt")
    ok(platform="telegram", sampling_method="random-walk",
       url_file_url="file:///tmp/urls.txt")


def test_random_walk_crawl_id_length():
    ok(platform="telegram", sampling_method="random-walk", seed_size=1,
       crawl_id="x" * 32)
    bad("32 characters", platform="telegram",
        sampling_method="random-walk", seed_size=1, crawl_id="x" * 33)


def test_urls_required_unless_dapr_job():
    bad("requires URLs", platform="telegram", sampling_method="channel")
    bad("requires URLs", platform="telegram", sampling_method="snowball")
    ok(platform="telegram", sampling_method="channel", mode="dapr-job")
    # random never needs URLs
    ok(platform="youtube", sampling_method="random", mode="")


def test_generate_crawl_id_format():
    """GenerateCrawlID 'YYYYMMDDHHMMSS' (common/utils.go:103-111)."""
    import datetime as dt

    from crawler_amd.config import generate_crawl_id

    t = dt.datetime(2026, 3, 4, 5, 6, 7, tzinfo=dt.timezone.utc)
    assert generate_crawl_id(t) == "20260304050607"
    # default: now-based, 14 digits, parseable back
    cid = generate_crawl_id()
    assert len(cid) == 14 and cid.isdigit()
    dt.datetime.strptime(cid, "%Y%m%d%H%M%S")


def test_parse_time_ago_units():
    """parseTimeAgo units (main.go parseTimeAgo): Nd / Nw / Nm / Ny and
    hour forms resolve to a past timestamp."""
    import datetime as dt

    from crawler_amd.config import parse_time_ago

    now = dt.datetime(2026, 6, 15, 12, 0, 0, tzinfo=dt.timezone.utc)
    assert parse_time_ago("30d", now=now) == now - dt.timedelta(days=30)
    assert parse_time_ago("2w", now=now) == now - dt.timedelta(weeks=2)
    assert parse_time_ago("6h", now=now) == now - dt.timedelta(hours=6)
    m = parse_time_ago("3m", now=now)
    assert (now - m).days in (89, 90, 91, 92)        # calendar months
    y = parse_time_ago("1y", now=now)
    assert (now - y).days in (365, 366)
    import pytest as _pt

    with _pt.raises(ValueError):
        parse_time_ago("5x", now=now)
