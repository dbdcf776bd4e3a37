"""Golden vectors transcribed LITERALLY from the reference's own tests.

Round-1 verdict item 7: all byte-exactness in this repo was
GPU <-> Python-oracle <-> C++-reference, every leg builder-authored.
These expected values are copied from the reference test files (cited
per block), so the oracle is pinned to the reference's published
behavior, not to itself.

Sources:
- /root/reference/telegramhelper/channel_links_test.go:65-330
- /root/reference/telegramhelper/username_filter_test.go:5-68
- /root/reference/telegramhelper/channelvalidator_test.go:15-85
- /root/reference/telegramhelper/tdutils.go:1005-1031 (msg-id >> 20)
- /root/reference/main_test.go:9-205 (sampling validation matrix)
"""
import pytest

from crawler_amd.engine.htmlvalidator import parse_channel_html
from crawler_amd.ops.golden import (Entity, FormattedText, SynthMessage,
                                    build_telegram_link_and_message_id,
                                    extract_channel_links,
                                    filter_username)


def _msg(content_type, text=None, entities=None):
    ft = FormattedText(text=text or "", entities=entities or [])
    if content_type == "messageText":
        return SynthMessage(content_type=content_type, text=ft)
    return SynthMessage(content_type=content_type, caption=ft)


# --- channel_links_test.go: plain-text regex scan ---

LINK_VECTORS = [
    # (test name, content_type, text, entities, expected sorted names)
    ("PlainTextTmeLink", "messageText",
     "Check out https://t.me/channelname for news", [], ["channelname"]),
    ("PlainTextTmeLinkNoScheme", "messageText",
     "Visit t.me/somechan today", [], ["somechan"]),
    ("PlainTextMultipleLinks", "messageText",
     "t.me/chanone and t.me/chantwo", [], ["chanone", "chantwo"]),
    ("TextEntityTypeTextUrl", "messageText", "click here",
     [Entity("text_url", 0, 10, url="https://t.me/linkedchan")],
     ["linkedchan"]),
    ("TextEntityTypeTextUrl_NonTme", "messageText", "link",
     [Entity("text_url", 0, 4, url="https://example.com/page")], []),
    # UTF-16 offset regressions (channel_links_test.go:115-175)
    ("Mention_ASCII", "messageText", "Hello @testchan!",
     [Entity("mention", 6, 9)], ["testchan"]),
    ("Mention_CyrillicPrefix", "messageText", "Привет @testchan",
     [Entity("mention", 7, 9)], ["testchan"]),
    ("Mention_EmojiPrefix", "messageText", "😀 @testchan",
     [Entity("mention", 3, 9)], ["testchan"]),
    ("Mention_ArabicPrefix", "messageText", "مرحبا @testchan",
     [Entity("mention", 6, 9)], ["testchan"]),
    ("TextEntityTypeUrl_TmeLink", "messageText",
     "See https://t.me/urlchan for details",
     [Entity("url", 4, 20)], ["urlchan"]),
    # media captions (channel_links_test.go:193-230)
    ("PhotoCaption", "messagePhoto", "t.me/photochan", [], ["photochan"]),
    ("VideoCaption", "messageVideo", "t.me/videochan", [], ["videochan"]),
    ("DocumentCaption", "messageDocument", "t.me/docchan", [],
     ["docchan"]),
    ("AnimationCaption", "messageAnimation", "t.me/animchan", [],
     ["animchan"]),
    ("AudioCaption", "messageAudio", "t.me/audiochan", [], ["audiochan"]),
    ("VoiceNoteCaption", "messageVoiceNote", "t.me/voicechan", [],
     ["voicechan"]),
    # reserved paths (channel_links_test.go:233-255)
    ("ReservedPath_Joinchat", "messageText",
     "https://t.me/joinchat/abc123", [], []),
    ("ReservedPath_Share", "messageText",
     "https://t.me/share/url?url=x", [], []),
    ("ReservedPath_Proxy", "messageText",
     "https://t.me/proxy?server=x", [], []),
    # case normalization (channel_links_test.go:281-285)
    ("CaseNormalization", "messageText", "t.me/MixedCase", [],
     ["mixedcase"]),
    # unsupported content type returns empty (channel_links_test.go:289)
    ("UnknownContentType", "messageSticker", "t.me/stickchan", [], []),
    # short names filtered by the {4,32} regex (channel_links_test.go:300)
    ("TooShortName", "messageText", "t.me/abc", [], []),
    # mention at end of string (channel_links_test.go:310)
    ("Mention_AtEndOfString", "messageText", "@endchan",
     [Entity("mention", 0, 8)], ["endchan"]),
]


@pytest.mark.parametrize(
    "name,ctype,text,entities,want",
    LINK_VECTORS, ids=[v[0] for v in LINK_VECTORS])
def test_channel_links_reference_vectors(name, ctype, text, entities,
                                         want):
    got = sorted(extract_channel_links(_msg(ctype, text, entities)))
    assert got == sorted(want)


def test_channel_links_deduplication_reference_vector():
    """channel_links_test.go:259-277: same channel as plain URL and as
    a TextUrl entity yields one entry."""
    msg = _msg("messageText", "Check t.me/samechan",
               [Entity("text_url", 6, 13, url="https://t.me/samechan")])
    got = extract_channel_links(msg)
    assert got == ["samechan"]


# --- username_filter_test.go:5-68 literal matrix ---

USERNAME_VECTORS = [
    ("valid simple", "testchannel", True, ""),
    ("valid with underscore", "test_channel", True, ""),
    ("valid with numbers", "channel123", True, ""),
    ("valid min length", "abcde", True, ""),
    ("valid 32 chars", "abcdefghijklmnopqrstuvwxyz123456", True, ""),
    ("too short 4 chars", "abcd", False, "too_short"),
    ("too short 1 char", "a", False, "too_short"),
    ("too short empty", "", False, "too_short"),
    ("too long 33 chars", "abcdefghijklmnopqrstuvwxyz1234567", False,
     "too_long"),
    ("starts with number", "1channel", False, "invalid_start_char"),
    ("starts with underscore", "_channel", False, "invalid_start_char"),
    ("starts with non-ASCII letter", "échannel", False,
     "invalid_start_char"),
    ("ends with underscore", "channel_", False, "ends_with_underscore"),
    ("contains space", "test channel", False, "invalid_char"),
    ("contains dash", "test-channel", False, "invalid_char"),
    ("contains dot", "test.channel", False, "invalid_char"),
    ("contains unicode", "téstchannel", False, "invalid_char"),
    ("ends with _bot", "some_bot", False, "bot_suffix"),
    ("ends with Bot", "SomeBot", False, "bot_suffix"),
    ("ends with BOT", "SomeBOT", False, "bot_suffix"),
    ("ends with _Bot", "Test_Bot", False, "bot_suffix"),
    ("looks like path", "usr/local", False, "invalid_char"),
    ("contains tilde", "home~user", False, "invalid_char"),
    ("contains dot path", "file.name", False, "invalid_char"),
]


@pytest.mark.parametrize(
    "name,username,want_valid,want_reason",
    USERNAME_VECTORS, ids=[v[0].replace(" ", "_")
                           for v in USERNAME_VECTORS])
def test_filter_username_reference_vectors(name, username, want_valid,
                                           want_reason):
    valid, reason = filter_username(username)
    assert valid == want_valid, (username, reason)
    if not want_valid:
        assert reason == want_reason, (username, reason, want_reason)


# --- channelvalidator_test.go:15-85 fixture classification ---

HTML_VECTORS = [
    ("valid-channel.html", "valid", ""),
    ("not-a-supergroup.html", "not_channel", "not_supergroup"),
    ("invalid-channel.html", "invalid", "not_found"),
    ("username-not-occupied.html", "invalid", "not_found"),
]


@pytest.mark.parametrize("fixture,status,reason", HTML_VECTORS,
                         ids=[v[0] for v in HTML_VECTORS])
def test_parse_channel_html_reference_vectors(fixture, status, reason):
    import os

    path = os.path.join(os.path.dirname(__file__), "..", "fixtures",
                        "telegram-html", fixture)
    with open(path, "rb") as f:
        body = f.read()
    res = parse_channel_html(body)
    assert res.status == status
    assert res.reason == reason


def test_parse_channel_html_unrecognised_title_errors():
    """channelvalidator_test.go:71-77."""
    with pytest.raises(Exception):
        r = parse_channel_html(
            b"<html><head><title>Something Unexpected</title></head>"
            b"</html>")
        if r is not None:  # impls may return error-status instead
            assert r.status not in ("valid", "not_channel", "invalid")


# --- tdutils.go:1005-1031 BuildTelegramLinkAndMessageID semantics ---

def test_build_link_msg_id_shift20():
    """public id = internal id / 1048576 (i.e. >> 20)."""
    msg = SynthMessage(msg_id=5 << 20)
    link, pid = build_telegram_link_and_message_id("chan", msg)
    assert pid == 5
    assert link == "https://t.me/chan/5"
    # media album appends ?single (tdutils.go:1022-1025)
    msg2 = SynthMessage(msg_id=7 << 20, media_album_id=99)
    link2, pid2 = build_telegram_link_and_message_id("chan", msg2)
    assert link2 == "https://t.me/chan/7?single"
    # private channel (no username) -> empty link
    link3, pid3 = build_telegram_link_and_message_id("", msg)
    assert link3 == "" and pid3 == 5


# --- main_test.go:9-205 TestValidateSamplingMethod literal matrix ---

SAMPLING_VECTORS = [
    # (platform, method, kwargs, expect_error_substring or None)
    ("telegram", "channel", dict(url_list=["https://t.me/test"]), None),
    ("telegram", "snowball",
     dict(url_list=["https://t.me/test1", "https://t.me/test2"]), None),
    ("telegram", "random", {}, "not supported for platform 'telegram'"),
    ("youtube", "channel",
     dict(url_list=["https://youtube.com/c/test"]), None),
    ("youtube", "random", {}, None),
    ("youtube", "snowball",
     dict(url_list=["https://youtube.com/c/seed"]), None),
    ("youtube", "channel", {}, "channel sampling requires URLs"),
    ("telegram", "snowball", {}, "snowball sampling requires URLs"),
    ("youtube", "channel", dict(url_file="/path/to/urls.txt"), None),
    ("unsupported", "channel",
     dict(url_list=["https://example.com"]), "unsupported platform"),
    ("youtube", "invalid",
     dict(url_list=["https://youtube.com/c/test"]),
     "not supported for platform 'youtube'"),
    ("telegram", "random-walk",
     dict(seed_size=100, crawl_id="my-crawl"), None),
    ("telegram", "random-walk",
     dict(url_list=["chan1", "chan2"], crawl_id="my-crawl"), None),
    ("telegram", "random-walk",
     dict(url_file_url="https://example.com/seeds.txt",
          crawl_id="my-crawl"), None),
    ("telegram", "random-walk", dict(crawl_id="my-crawl"),
     "must provide either seed urls or seed size"),
    ("telegram", "random-walk",
     dict(url_list=["chan1"], seed_size=100, crawl_id="my-crawl"),
     "must provide either seed urls or seed size"),
    ("telegram", "random-walk",
     dict(seed_size=100,
          crawl_id="this-crawl-id-is-way-too-long-and-exceeds-32-chars"),
     "crawl IDs cannot exceed 32 characters"),
    # dapr-job mode relaxations (TestValidateSamplingMethodDaprJobMode)
    ("telegram", "channel", dict(mode="dapr-job"), None),
    ("youtube", "snowball", dict(mode="dapr-job"), None),
    ("telegram", "channel", dict(mode="standalone"),
     "channel sampling requires URLs"),
    ("telegram", "channel", {}, "channel sampling requires URLs"),
]


@pytest.mark.parametrize("platform,method,kwargs,err", SAMPLING_VECTORS)
def test_sampling_validation_reference_vectors(platform, method, kwargs,
                                               err):
    from crawler_amd.config import validate_sampling_method

    if err is None:
        validate_sampling_method(platform, method, **kwargs)
    else:
        with pytest.raises(ValueError) as e:
            validate_sampling_method(platform, method, **kwargs)
        assert err in str(e.value), (err, str(e.value))


# --- dapr/calculate_date_filters_test.go:12-112 literal cases ---

def test_calculate_date_filters_reference_vectors():
    import datetime as dt

    from crawler_amd.config import CrawlerConfig, calculate_date_filters

    UTC = dt.timezone.utc
    d = lambda *a: dt.datetime(*a, tzinfo=UTC)
    now = d(2026, 6, 1)

    # DateBetween takes precedence over everything
    cfg = CrawlerConfig(crawl_id="c", date_between_min=d(2024, 1, 1),
                        date_between_max=d(2024, 12, 31),
                        post_recency=d(2023, 6, 1),
                        min_post_date=d(2022, 1, 1))
    assert calculate_date_filters(cfg, now) == (d(2024, 1, 1),
                                                d(2024, 12, 31))
    # only DateBetweenMin set -> branch NOT taken; PostRecency used
    cfg = CrawlerConfig(crawl_id="c", date_between_min=d(2024, 1, 1),
                        post_recency=d(2023, 6, 1))
    assert calculate_date_filters(cfg, now) == (d(2023, 6, 1), now)
    # PostRecency branch ignores MinPostDate; to == now
    cfg = CrawlerConfig(crawl_id="c", post_recency=d(2024, 6, 1),
                        min_post_date=d(2022, 1, 1))
    assert calculate_date_filters(cfg, now) == (d(2024, 6, 1), now)
    # MinPostDate fallback
    cfg = CrawlerConfig(crawl_id="c", min_post_date=d(2022, 1, 1))
    assert calculate_date_filters(cfg, now) == (d(2022, 1, 1), now)
    # all zero -> (None, ~now)
    cfg = CrawlerConfig(crawl_id="c")
    assert calculate_date_filters(cfg, now) == (None, now)


# --- common/utils_coverage_test.go:98-200 + utils_test.go:13-40 ---

def test_read_urls_from_file_reference_vectors(tmp_path):
    """ReadURLsFromFile: trims whitespace, drops empty lines, keeps
    order (the CLI's --url-file path)."""
    from crawler_amd.cli import resolve_urls

    class A:
        urls = ""
        url_file = ""
        url_file_url = ""

    f = tmp_path / "urls.txt"
    f.write_text(
        "https://example.com/channel1\n\t\n"
        "https://example.com/channel2\n"
        "   https://example.com/channel3   \n\n"
        "https://example.com/channel4\n")
    a = A()
    a.url_file = str(f)
    assert resolve_urls(a) == [
        "https://example.com/channel1",
        "https://example.com/channel2",
        "https://example.com/channel3",
        "https://example.com/channel4",
    ]
    # empty file -> no urls
    e = tmp_path / "empty.txt"
    e.write_text("")
    a.url_file = str(e)
    assert resolve_urls(a) == []
    # whitespace-only -> no urls
    ws = tmp_path / "ws.txt"
    ws.write_text("   \n\t\t\n   \t   \n")
    a.url_file = str(ws)
    assert resolve_urls(a) == []
    # file:// form of --url-file-url resolves identically
    a.url_file = ""
    a.url_file_url = f"file://{f}"
    assert len(resolve_urls(a)) == 4


def test_generate_crawl_id_reference_format():
    """utils_test.go:13-40: 14 digits, parseable as YYYYMMDDHHMMSS."""
    import datetime as dt
    import re

    from crawler_amd.config import generate_crawl_id

    cid = generate_crawl_id()
    assert re.fullmatch(r"\d{14}", cid)
    dt.datetime.strptime(cid, "%Y%m%d%H%M%S")  # parses back
    fixed = generate_crawl_id(dt.datetime(2024, 3, 5, 6, 7, 8))
    assert fixed == "20240305060708"


# --- crawl/flood_wait_test.go:31-140 TestParseFloodWaitSecs matrix ---

FLOOD_VECTORS = [
    ("nil error", "", 0, False),
    ("unrelated error", "connection refused", 0, False),
    ("bare FLOOD_WAIT with seconds", "FLOOD_WAIT_72560", 72560, True),
    ("prefixed as seen from TDLib", "[429] FLOOD_WAIT_300", 300, True),
    ("short ban below retire threshold", "FLOOD_WAIT_30", 30, True),
    ("exactly at retire threshold", "FLOOD_WAIT_300", 300, True),
    ("FLOOD_WAIT_0", "FLOOD_WAIT_0", 0, True),
    ("no trailing digits", "FLOOD_WAIT_", 0, True),
    ("embedded in longer message",
     "rpc error: code 429 FLOOD_WAIT_600 please wait", 600, True),
    ("wrapped error containing FLOOD_WAIT",
     "SearchPublicChat failed: FLOOD_WAIT_1800", 1800, True),
    ("HTTP 429 retry-after format",
     "429 Too Many Requests: retry after 72560", 72560, True),
    ("retry after with no digits",
     "429 Too Many Requests: retry after soon", 0, True),
]


@pytest.mark.parametrize("name,msg,want_secs,want_flood", FLOOD_VECTORS,
                         ids=[v[0].replace(" ", "_")
                              for v in FLOOD_VECTORS])
def test_parse_flood_wait_reference_vectors(name, msg, want_secs,
                                            want_flood):
    from crawler_amd.engine.errors import parse_flood_wait_secs

    secs, is_flood = parse_flood_wait_secs(msg)
    assert is_flood == want_flood, (msg, is_flood)
    assert secs == want_secs, (msg, secs)


def test_flood_wait_retire_threshold_boundary():
    """flood_wait_test.go:122-140 + crawl/runner.go:49: 300 s."""
    from crawler_amd.engine.errors import (
        FLOOD_WAIT_RETIRE_THRESHOLD_SECS, parse_flood_wait_secs)

    assert FLOOD_WAIT_RETIRE_THRESHOLD_SECS == 300
    below, _ = parse_flood_wait_secs("FLOOD_WAIT_299")
    at, _ = parse_flood_wait_secs("FLOOD_WAIT_300")
    above, _ = parse_flood_wait_secs("FLOOD_WAIT_301")
    assert below < 300 <= at < above


# --- crawl/runner_400_test.go:20-60, 841-900 isTDLib400 matrix ---

TDLIB400_VECTORS = [
    ("400 USERNAME_INVALID", True, "unbracketed prod-log format"),
    ("400 USERNAME_NOT_OCCUPIED", True, "unbracketed prod-log format"),
    ("[400] CHANNEL_INVALID", True, "bracketed TDLib format"),
    ("no messages found in the chat", True, "empty/inaccessible channel"),
    ("[404] USERNAME_NOT_OCCUPIED", False, "404 is not 400"),
    ("FLOOD_WAIT_300", False, "flood wait is not a 400"),
    ("network timeout", False, "unrelated error"),
    ("", False, "empty string"),
    ("getChannelInfo failed: 400 USERNAME_INVALID", True,
     "wrapped error still detected"),
]


@pytest.mark.parametrize("msg,want,comment", TDLIB400_VECTORS,
                         ids=[v[2].replace(" ", "_")
                              for v in TDLIB400_VECTORS])
def test_is_tdlib_400_reference_vectors(msg, want, comment):
    from crawler_amd.engine.errors import is_tdlib_400

    assert is_tdlib_400(msg) == want, (msg, comment)


# --- distributed/messages_test.go:103-170, 365-399 ---

def test_work_item_validation_reference_vectors():
    from crawler_amd.parallel import messages as M

    ok = M.WorkItem(id="test-id", url="https://t.me/testchannel",
                    platform="telegram", crawl_id="test-crawl")
    ok.validate()
    cases = [
        dict(url="https://t.me/testchannel", platform="telegram",
             crawl_id="test-crawl"),                       # missing ID
        dict(id="test-id", platform="telegram",
             crawl_id="test-crawl"),                       # missing URL
        dict(id="test-id", url="https://t.me/testchannel",
             platform="", crawl_id="test-crawl"),          # missing platform
        dict(id="test-id", url="https://example.com",
             platform="unsupported", crawl_id="test-crawl"),
    ]
    for kw in cases:
        with pytest.raises(ValueError):
            M.WorkItem(**kw).validate()


def test_trace_id_and_topic_constants_reference_shape():
    from crawler_amd.parallel import messages as M

    id1, id2 = M.new_trace_id(), M.new_trace_id()
    assert id1 and id2 and id1 != id2
    assert len(id1) > 16 and len(id2) > 16  # trace_YYYYMMDDHHMMSS_XXXX...
    for const in (M.TOPIC_WORK_QUEUE, M.TOPIC_RESULTS,
                  M.TOPIC_WORKER_STATUS, M.TOPIC_ORCHESTRATOR,
                  M.MSG_WORKER_STARTED, M.MSG_WORKER_STOPPING,
                  M.WORKER_IDLE, M.WORKER_BUSY, M.WORKER_ACTIVE,
                  M.WORKER_OFFLINE):
        assert const
    # topic names are the reference's (messages.go:53-58)
    assert M.TOPIC_WORK_QUEUE == "crawl-work-queue"
    assert M.TOPIC_RESULTS == "crawl-results"
    assert M.TOPIC_WORKER_STATUS == "worker-status"
    assert M.TOPIC_ORCHESTRATOR == "orchestrator-commands"


# --- worker/worker_test.go TestShouldRetryError matrix ---

RETRY_VECTORS = [
    ("channel not found", False),
    ("access denied", False),
    ("forbidden", False),
    ("connection failed", True),
    ("timeout occurred", True),
    ("temporary failure", True),
    ("something went wrong", True),  # unknown errors default to retry
]


@pytest.mark.parametrize("msg,want", RETRY_VECTORS,
                         ids=[v[0].replace(" ", "_")
                              for v in RETRY_VECTORS])
def test_should_retry_reference_vectors(msg, want):
    from crawler_amd.parallel.worker import Worker

    assert Worker.should_retry_error(Exception(msg)) == want


# --- BASELINE.md structural-constants table, pinned in one place ---

def test_structural_constants_match_reference():
    """Every hard-coded throttle/limit from BASELINE.md's table (the
    reference's measurable performance envelope), pinned so drift is a
    test failure, not a doc bug."""
    from crawler_amd.config import CrawlerConfig, TelegramRateLimitConfig
    from crawler_amd.engine import errors as E
    from crawler_amd.engine import validator as V
    from crawler_amd.engine.htmlvalidator import BODY_CAP
    from crawler_amd.engine.media import MEDIA_SIZE_CAP_MB
    from crawler_amd.engine.state import LocalStateManager, RandomWalkStore
    from crawler_amd.parallel import orchestrator as O

    rl = TelegramRateLimitConfig()
    assert rl.get_chat_history_rate == 30.0          # utils.go:35
    assert rl.search_public_chat_rate == 6.0         # utils.go:38
    assert rl.get_supergroup_info_rate == 20.0       # utils.go:39
    assert rl.get_message_server_hit_rate == 60.0    # utils.go:43
    assert rl.get_chat_history_jitter_ms == 500
    assert rl.search_public_chat_jitter_ms == 1500
    assert rl.get_supergroup_info_jitter_ms == 800

    cfg = CrawlerConfig(crawl_id="c")
    assert cfg.concurrency == 1                       # main.go:758
    assert cfg.max_pages == 108000                    # main.go:776
    assert cfg.walkback_rate == 15                    # main.go:783
    assert cfg.combine_trigger_size == 170            # main.go:800
    assert cfg.combine_hard_cap == 200                # main.go:801

    assert E.FLOOD_WAIT_RETIRE_THRESHOLD_SECS == 300  # runner.go:49
    assert MEDIA_SIZE_CAP_MB == 150.0                 # tdutils.go:293
    assert BODY_CAP == 64 * 1024                      # channelvalidator.go:103
    assert V.BLOCKED_THRESHOLD == 5                   # validator.go:36
    assert V.PROBE_INTERVAL_S == 300.0                # validator.go:37
    assert RandomWalkStore.MAX_ATTEMPTS == 3          # daprstate.go poison
    assert LocalStateManager.export_chunk_size_bytes == 100 * 1024 * 1024

    assert O.DISTRIBUTE_INTERVAL_S == 5.0             # orchestrator.go:163
    assert O.HEALTH_INTERVAL_S == 30.0                # orchestrator.go:475
    assert O.DEFAULT_WORKER_TIMEOUT_S == 300.0        # orchestrator.go:498
    assert O.Orchestrator.MAX_RETRIES == 3            # orchestrator_test.go:308
