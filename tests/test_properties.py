"""Property-based tests (hypothesis) for the encoding/parsing oracles.

These pin INVARIANTS rather than examples: the Go-JSON escaper must
produce strings that a JSON parser decodes back to the original; UTF-16
offset conversion must agree with an independent construction; the full
Post line must stay parseable JSON for arbitrary field content."""
import datetime as dt
import json

from hypothesis import given, settings
from hypothesis import strategies as st

from crawler_amd.models.post import Post, format_go_time, go_json_escape
from crawler_amd.ops.golden import filter_username, utf16_offset_to_bytes

UTC = dt.timezone.utc

# text including controls, quotes, JSON-hazard chars, astral planes and
# the U+2028/29 separators Go escapes specially
hazard_text = st.text(
    alphabet=st.one_of(
        st.characters(min_codepoint=0, max_codepoint=0x7F),
        st.sampled_from('"\\<>&\u2028\u2029\u0000\u001f'),
        st.characters(min_codepoint=0x80, max_codepoint=0x10FFFF,
                      exclude_categories=("Cs",)),
    ),
    max_size=200,
)


@settings(max_examples=300, deadline=None)
@given(hazard_text)
def test_escape_roundtrips_through_json(s):
    enc = '"' + go_json_escape(s) + '"'
    assert json.loads(enc) == s


@settings(max_examples=300, deadline=None)
@given(hazard_text)
def test_escape_output_is_ascii_safe_for_html(s):
    enc = go_json_escape(s)
    # Go's HTML-safe encoder never emits raw <, >, & or the line seps
    assert "<" not in enc and ">" not in enc and "&" not in enc
    assert "\u2028" not in enc and "\u2029" not in enc


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=60,
               alphabet=st.characters(min_codepoint=1,
                                      max_codepoint=0x10FFFF,
                                      exclude_categories=("Cs",))),
       st.integers(0, 80), st.integers(0, 80))
def test_utf16_offsets_match_independent_construction(s, off16, len16):
    """Compare against building the prefix via Python's UTF-16 encoder:
    byte_start must equal len(utf8(chars whose utf16 prefix length
    <= off16)) whenever the offset is reachable."""
    b = s.encode("utf-8")
    start, end = utf16_offset_to_bytes(b, off16, len16)
    # independent: walk characters accumulating utf16 units; "reached"
    # means the offset lands on a rune boundary BEFORE the end of text
    # (the mirrored tdutils.go walk only tests positions inside the loop)
    u = 0
    byte_pos = 0
    reached = False
    for ch in s:
        if u == off16:
            reached = True
            break
        if u > off16:
            break
        u += len(ch.encode("utf-16-le")) // 2
        byte_pos += len(ch.encode("utf-8"))
    if reached:
        assert start == byte_pos
        assert start <= end <= len(b)
    elif start == -1:
        # faithful reference quirk (tdutils.go:55-78): off16 falls
        # mid-surrogate-pair but off16+len16 lands on a boundary ->
        # rune_start stays -1. TDLib never emits such offsets.
        pass
    else:
        # never-reached offsets (incl. offset == end of text) -> (0, 0)
        assert (start, end) == (0, 0)


@settings(max_examples=300, deadline=None)
@given(st.text(max_size=40))
def test_filter_username_never_crashes_and_reasons_stable(s):
    ok, reason = filter_username(s)
    assert isinstance(ok, bool)
    if ok:
        # accepted names satisfy the structural rules
        assert 5 <= len(s) <= 32
        assert s[0].isascii() and s[0].isalpha()
        assert not s.endswith("_")
        assert all(c.isascii() and (c.isalnum() or c == "_") for c in s)
        assert not s.lower().endswith("bot")
    else:
        assert reason


@settings(max_examples=100, deadline=None)
@given(st.integers(0, 4102444800), st.integers(0, 999999))
def test_format_go_time_matches_strftime_for_modern_dates(secs, micros):
    t = dt.datetime.fromtimestamp(secs, UTC).replace(microsecond=micros)
    got = format_go_time(t)
    # RFC3339Nano: seconds part matches strftime; fraction trimmed of
    # trailing zeros; 'Z' suffix
    assert got.startswith(t.strftime("%Y-%m-%dT%H:%M:%S"))
    assert got.endswith("Z")
    if micros == 0:
        assert "." not in got
    else:
        frac = got[len("2006-01-02T15:04:05"):-1]
        assert frac.startswith(".")
        assert not frac.endswith("0")
        assert int(frac[1:].ljust(6, "0")) == micros


@settings(max_examples=100, deadline=None)
@given(hazard_text, hazard_text, st.integers(0, 2**31 - 1))
def test_post_jsonl_always_parses(desc, title, views):
    p = Post(
        post_link="https://t.me/x/1", channel_id="-1", post_uid="u",
        url="https://t.me/x/1",
        published_at=dt.datetime(2024, 1, 1, tzinfo=UTC),
        created_at=dt.datetime(2024, 1, 2, tzinfo=UTC),
        description=desc, view_count=views,
    )
    p.channel_data.channel_name = title
    line = p.to_jsonl()
    assert line.endswith("\n")
    obj = json.loads(line)
    assert obj["description"] == desc
    assert obj["channel_data"]["channel_name"] == title
    assert obj["view_count"] == views


@settings(max_examples=60, deadline=None)
@given(st.lists(
    st.tuples(
        hazard_text,                                   # body text
        st.sampled_from(["messageText", "messagePhoto", "messagePoll",
                         "messageSticker", "messageDocument"]),
        st.integers(0, 2**31 - 1),                     # views
        st.integers(0, 10_000),                        # forwards
        st.dictionaries(st.sampled_from(["👍", "🔥", "❤"]),
                        st.integers(1, 1000), max_size=3),
    ),
    min_size=1, max_size=12,
))
def test_pack_unpack_roundtrip(rows):
    """SoA batch pack -> unpack returns the same message fields
    (the packer is what every GPU test trusts for inputs)."""
    from crawler_amd.ops import batch as B
    from crawler_amd.ops import golden as G

    msgs = []
    for k, (text, ct, views, fwd, reacts) in enumerate(rows):
        kw = {}
        if ct == "messageText":
            kw["text"] = G.FormattedText(text=text)
        elif ct == "messagePhoto":
            kw["caption"] = G.FormattedText(text=text)
        elif ct == "messagePoll":
            kw["poll_question"] = text
        elif ct == "messageDocument":
            kw["document_name"] = text
        msgs.append(G.SynthMessage(
            chat_id=-1001000000042, msg_id=(k + 1) << 20,
            date=1_700_000_000 + k, content_type=ct, views=views,
            forwards=fwd, reactions=dict(reacts),
            poster_handle=f"user{k:04d}", **kw,
        ))
    ch = [B.ChannelRow(chat_id=-1001000000042, username="propchan",
                       title="P", member_count=1, post_count=len(msgs),
                       total_views=0)]
    batch = B.pack(msgs, ch, [0] * len(msgs))
    for i, orig in enumerate(msgs):
        back = B.unpack_message(batch, i)
        assert back.content_type == orig.content_type
        assert back.views == orig.views and back.forwards == orig.forwards
        assert back.reactions == orig.reactions
        assert back.msg_id == orig.msg_id and back.date == orig.date
        if orig.text:
            assert back.text.text == orig.text.text
        if orig.caption:
            assert back.caption.text == orig.caption.text
        assert back.poll_question == orig.poll_question
        assert back.document_name == orig.document_name
