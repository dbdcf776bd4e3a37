"""Golden hot-path op tests: link extraction, UTF-16 offsets, username filter.

Mirrors the reference's telegramhelper/channel_links_test.go and
username_filter_test.go coverage areas.
"""
import pytest

from crawler_amd.ops import (
    Entity,
    FormattedText,
    SynthMessage,
    build_telegram_link_and_message_id,
    extract_channel_links,
    extract_links_with_source,
    filter_username,
    utf16_offset_to_bytes,
)


def msg_text(text, entities=()):
    return SynthMessage(
        content_type="messageText",
        text=FormattedText(text=text, entities=list(entities)),
    )


# ---------- utf16_offset_to_bytes ----------

def test_utf16_ascii():
    b = "hello world".encode()
    assert utf16_offset_to_bytes(b, 6, 5) == (6, 11)


def test_utf16_cyrillic():
    s = "привет @chan1 дела"
    b = s.encode()
    off = s.index("@chan1")  # python str index == utf-16 units here (BMP)
    start, end = utf16_offset_to_bytes(b, off, 6)
    assert b[start:end].decode() == "@chan1"


def test_utf16_emoji_surrogate_pair():
    s = "🚀🚀 @chan1"
    b = s.encode()
    # each rocket = 2 UTF-16 units -> @chan1 starts at unit 5
    start, end = utf16_offset_to_bytes(b, 5, 6)
    assert b[start:end].decode() == "@chan1"


def test_utf16_offset_never_reached():
    assert utf16_offset_to_bytes(b"abc", 10, 2) == (0, 0)


def test_utf16_end_past_text():
    b = b"abcdef"
    assert utf16_offset_to_bytes(b, 3, 100) == (3, 6)


def test_utf16_zero_length():
    start, end = utf16_offset_to_bytes(b"abcdef", 2, 0)
    assert start == end == 2


# ---------- link extraction ----------

def test_plaintext_tme_link():
    m = msg_text("check out https://t.me/mychannel today")
    assert extract_channel_links(m) == ["mychannel"]


def test_plaintext_bare_tme():
    m = msg_text("go to t.me/some_chan now")
    assert extract_channel_links(m) == ["some_chan"]


def test_reserved_paths_filtered():
    m = msg_text("t.me/joinchat/abcdef and t.me/addstickers/xyz12")
    assert extract_channel_links(m) == []


def test_short_username_not_matched():
    # usernames must be >= 5 chars ([a-zA-Z][a-zA-Z0-9_]{4,31})
    m = msg_text("see t.me/abcd plus t.me/abcde")
    assert extract_channel_links(m) == ["abcde"]


def test_username_starting_with_digit_not_matched():
    m = msg_text("bad t.me/1abcde")
    # regex requires a letter first; "abcde" alone can't match since
    # 't.me/' precedes '1abcde' and the regex needs the name right after /
    assert extract_channel_links(m) == []


def test_lowercased():
    m = msg_text("https://t.me/MyChannel")
    assert extract_channel_links(m) == ["mychannel"]


def test_mention_entity():
    s = "hi @GoodChan1 there"
    m = msg_text(s, [Entity("mention", s.index("@"), len("@GoodChan1"))])
    links = extract_links_with_source(m)
    assert [(l.name, l.source_type) for l in links] == [("goodchan1", "mention")]


def test_mention_entity_nonascii_prefix():
    s = "привет @chan_one и ещё"
    m = msg_text(s, [Entity("mention", s.index("@"), 9)])
    links = extract_links_with_source(m)
    assert ("chan_one", "mention") in [(l.name, l.source_type) for l in links]


def test_text_url_entity():
    m = msg_text(
        "click here",
        [Entity("text_url", 0, 5, url="https://t.me/hidden_chan")],
    )
    links = extract_links_with_source(m)
    assert [(l.name, l.source_type) for l in links] == [
        ("hidden_chan", "text_url")
    ]


def test_url_entity_slices_text():
    s = "go to t.me/urlchan now"
    m = msg_text(s, [Entity("url", 6, len("t.me/urlchan"))])
    links = extract_links_with_source(m)
    # entity walk adds it as "url"; plaintext scan would add same name later
    assert [(l.name, l.source_type) for l in links] == [("urlchan", "url")]


def test_first_wins_entity_before_plaintext():
    s = "see https://t.me/dupchan"
    m = msg_text(s, [Entity("url", 4, len("https://t.me/dupchan"))])
    links = extract_links_with_source(m)
    assert links[0].source_type == "url"  # not plaintext


def test_dedup_multiple_plaintext():
    m = msg_text("t.me/samechan t.me/samechan t.me/other_1")
    assert extract_channel_links(m) == ["samechan", "other_1"]


def test_caption_extraction_for_media():
    m = SynthMessage(
        content_type="messagePhoto",
        caption=FormattedText(text="pic from t.me/photochan"),
    )
    assert extract_channel_links(m) == ["photochan"]


def test_no_formatted_text_types_have_no_links():
    m = SynthMessage(content_type="messagePoll", poll_question="t.me/pollchan?")
    assert extract_channel_links(m) == []


def test_max_length_username():
    name = "a" + "b" * 31  # 32 chars, max allowed
    m = msg_text(f"t.me/{name}x")  # 33 chars in text; regex takes first 32
    assert extract_channel_links(m) == [name]


# ---------- username filter ----------

@pytest.mark.parametrize(
    "name,valid,reason",
    [
        ("goodchannel", True, ""),
        ("abcd", False, "too_short"),
        ("a" * 33, False, "too_long"),
        ("1abcdef", False, "invalid_start_char"),
        ("_abcdef", False, "invalid_start_char"),
        ("abcde_", False, "ends_with_underscore"),
        ("abc-de", False, "invalid_char"),
        ("abc de", False, "invalid_char"),
        ("my_bot", False, "bot_suffix"),
        ("somerobot", False, "bot_suffix"),  # bare 'bot' suffix rejects
        ("botstuff", True, ""),
        ("ab_cd_ef", True, ""),
        ("A2345", True, ""),
    ],
)
def test_filter_username(name, valid, reason):
    v, r = filter_username(name)
    assert v == valid
    assert r == reason


def test_filter_username_path_chars_rejected_by_charset():
    v, r = filter_username("abc/def")
    assert not v and r == "invalid_char"


# ---------- link building ----------

def test_build_link_public_channel():
    m = SynthMessage(msg_id=7 << 20)
    link, public_id = build_telegram_link_and_message_id("mychan", m)
    assert link == "https://t.me/mychan/7"
    assert public_id == 7


def test_build_link_album_single():
    m = SynthMessage(msg_id=3 << 20, media_album_id=99)
    link, _ = build_telegram_link_and_message_id("mychan", m)
    assert link == "https://t.me/mychan/3?single"


def test_build_link_private_channel_empty():
    m = SynthMessage(msg_id=5 << 20)
    link, public_id = build_telegram_link_and_message_id("", m)
    assert link == ""
    assert public_id == 5


def test_extract_nonoverlap_cursor_go_findall():
    """Go FindAllSubmatch non-overlap: after a match the scan resumes at
    the END of the match — the embedded 't.me/xyzzy' starting inside the
    first match is skipped, later candidates still match."""
    from crawler_amd.ops import golden as G
    import re as _re

    ft = G.FormattedText(text="t.me/abcdet.me/xyzzy and t.me/tttttme")
    names = G.extract_channel_links(
        G.SynthMessage(content_type="messageText", text=ft))
    # exact oracle pin: Python finditer has the same non-overlap cursor
    pat = _re.compile(r"(https?://)?t\.me/([a-zA-Z][a-zA-Z0-9_]{4,31})")
    expect = []
    for m in pat.finditer(ft.text):
        nm = m.group(2).lower()
        if nm not in G.RESERVED_PATHS and nm not in expect:
            expect.append(nm)
    assert names == expect == ["abcdet", "tttttme"]
