"""FLOOD_WAIT/400 parsing matrix (reference runner_flood_wait_test /
runner_400_test coverage) + hypothesis property tests of the golden ops."""
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from crawler_amd.engine.errors import (
    FLOOD_WAIT_RETIRE_THRESHOLD_SECS,
    is_tdlib_400,
    parse_flood_wait_secs,
)
from crawler_amd.models.post import go_json_escape
from crawler_amd.ops.golden import utf16_offset_to_bytes


# ---------- FLOOD_WAIT parse matrix (crawl/runner.go:55-97) ----------

@pytest.mark.parametrize("msg,secs,is_fw", [
    ("[429] FLOOD_WAIT_72560", 72560, True),
    ("FLOOD_WAIT_30", 30, True),
    ("429 Too Many Requests: retry after 120", 120, True),
    ("retry after 7", 7, True),
    ("FLOOD_WAIT_", 0, True),          # unparseable secs -> short ban
    ("retry after ", 0, True),
    ("some other error", 0, False),
    ("", 0, False),
])
def test_parse_flood_wait_matrix(msg, secs, is_fw):
    assert parse_flood_wait_secs(msg) == (secs, is_fw)


def test_retire_threshold_constant():
    assert FLOOD_WAIT_RETIRE_THRESHOLD_SECS == 300  # crawl/runner.go:49


@pytest.mark.parametrize("msg,is400", [
    ("[400] CHANNEL_INVALID", True),
    ("400 USERNAME_NOT_OCCUPIED: xyz", True),
    ("400 USERNAME_INVALID", True),
    ("no messages found in the chat", True),
    ("[429] FLOOD_WAIT_10", False),
    ("500 internal", False),
])
def test_is_tdlib_400_matrix(msg, is400):
    assert is_tdlib_400(msg) == is400


# ---------- property tests ----------

@settings(max_examples=200, deadline=None)
@given(st.text(max_size=200))
def test_escape_roundtrips_via_json(s):
    """Escaped output must parse back to the original string as JSON."""
    import json

    encoded = '"' + go_json_escape(s) + '"'
    assert json.loads(encoded) == s


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=120), st.integers(0, 140), st.integers(0, 60))
def test_utf16_offsets_match_utf16_encoding(s, off, ln):
    """Byte range returned for (off, len) must equal slicing the UTF-16
    code-unit sequence, whenever the offsets land on rune boundaries."""
    units = []
    byte_pos = []
    b = s.encode("utf-8")
    pos = 0
    for ch in s:
        n_units = 2 if ord(ch) >= 0x10000 else 1
        for _ in range(n_units):
            units.append(ch)
        byte_pos.append(pos)
        pos += len(ch.encode("utf-8"))
    total_units = len(units)
    start, end = utf16_offset_to_bytes(b, off, ln)
    # build expected via the same scan rules
    if off > total_units or (off == total_units):
        # offset never reached inside the loop -> (0, 0)
        assert (start, end) == (0, 0) or off == 0 and len(b) == 0
        return
    # only check boundary-aligned offsets (TDLib guarantees these)
    cum = 0
    boundaries = {}
    i = 0
    for ch in s:
        boundaries[cum] = i
        cum += 2 if ord(ch) >= 0x10000 else 1
        i += len(ch.encode("utf-8"))
    boundaries[cum] = len(b)
    if off in boundaries:
        assert start == boundaries[off]
        if off + ln in boundaries:
            assert end == boundaries[off + ln]
        else:
            # end fell inside a surrogate or past the end -> clamps to len
            assert end == len(b) or end in boundaries.values()


@settings(max_examples=100, deadline=None)
@given(st.text(
    alphabet=st.sampled_from("abct.me/ @_0фг🚀\n\"<&"), max_size=150
))
def test_link_extraction_agrees_with_regex(s):
    """Golden plaintext extraction == direct regex scan."""
    from crawler_amd.ops.golden import (
        CHANNEL_LINK_RE,
        FormattedText,
        SynthMessage,
        channel_name_from_match,
        extract_channel_links,
    )

    expected = []
    for m in CHANNEL_LINK_RE.finditer(s):
        name = channel_name_from_match(m)
        if name and name not in expected:
            expected.append(name)
    msg = SynthMessage(content_type="messageText",
                       text=FormattedText(text=s))
    assert extract_channel_links(msg) == expected


def test_connection_error_recreates_session(tmp_path):
    """Transport failure destroys + recreates the pooled session
    (HandleConnectionError, connection_pool.go:346-413): the pool stays
    at full strength with a FRESH conn id, nothing is retired."""
    from crawler_amd.config import CrawlerConfig
    from crawler_amd.engine import LocalStateManager, Page
    from crawler_amd.engine import errors as E
    from crawler_amd.engine.pipeline import run_for_channel_with_pool
    from crawler_amd.feed import FeedConfig, SyntheticFeed
    from crawler_amd.feed.client import ConnectionPool, FaultConfig

    cfg = CrawlerConfig(crawl_id="cr1", storage_root=str(tmp_path),
                        min_users=1, disable_rate_limits=True)
    feed = SyntheticFeed(FeedConfig(seed=3, universe=100,
                                    posts_per_channel=10))
    pool = ConnectionPool(
        feed, 2, cfg.rate_limit, posts_per_channel=10,
        disable_rate_limits=True,
        faults=FaultConfig(conn_reset_permille=1000),  # always reset
    )
    before = pool.stats()
    ids_before = set(pool.available.keys())
    sm = LocalStateManager(cfg)
    with pytest.raises(E.ConnectionDropped):
        run_for_channel_with_pool(pool, Page(id="p", url="c0000000001"),
                                  sm, cfg)
    after = pool.stats()
    assert after["available"] == before["available"]  # recreated, not lost
    assert after.get("retired", 0) == before.get("retired", 0)
    assert set(pool.available.keys()) != ids_before   # fresh session dir
    sm.close()


def test_connection_error_contained_by_runner(tmp_path):
    """A transport failure marks the page error and the crawl proceeds
    with the recreated session."""
    from crawler_amd.config import CrawlerConfig
    from crawler_amd.engine import LocalStateManager
    from crawler_amd.engine.runner import StandaloneRunner
    from crawler_amd.feed import FeedConfig, SyntheticFeed
    from crawler_amd.feed.client import ConnectionPool, FaultConfig

    cfg = CrawlerConfig(crawl_id="cr2", storage_root=str(tmp_path),
                        min_users=1, sampling_method="channel",
                        disable_rate_limits=True)
    feed = SyntheticFeed(FeedConfig(seed=3, universe=100,
                                    posts_per_channel=10))
    # ~30% of API calls drop the connection
    pool = ConnectionPool(
        feed, 2, cfg.rate_limit, posts_per_channel=10,
        disable_rate_limits=True,
        faults=FaultConfig(conn_reset_permille=300),
    )
    runner = StandaloneRunner(cfg, LocalStateManager(cfg), pool)
    stats = runner.run(["c%010d" % i for i in range(1, 9)])
    assert stats["pages"] + stats["errors"] == 8
    assert pool.stats()["available"] + pool.stats()["in_use"] == 2
