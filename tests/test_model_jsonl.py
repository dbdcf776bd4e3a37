"""JSONL schema + Go-compatible encoding tests (reference model/data.go)."""
import datetime as dt
import json

from crawler_amd.models import (
    ChannelData,
    Comment,
    EngagementData,
    PerformanceScores,
    Post,
    format_go_time,
    go_json_escape,
)

UTC = dt.timezone.utc


def test_escape_basic():
    assert go_json_escape('he said "hi"') == 'he said \\"hi\\"'
    assert go_json_escape("a\\b") == "a\\\\b"
    assert go_json_escape("l1\nl2\tend\r") == "l1\\nl2\\tend\\r"


def test_escape_html_unsafe_matches_go():
    # Go encoding/json escapes <, >, & as < etc. by default
    assert go_json_escape("<b>&") == "\\u003cb\\u003e\\u0026"


def test_escape_control_chars():
    assert go_json_escape("\x00\x01\x1f") == "\\u0000\\u0001\\u001f"


def test_escape_line_separators():
    assert go_json_escape("a b ") == "a\\u2028b\\u2029"


def test_escape_unicode_passthrough():
    # Non-ASCII is NOT escaped (Go writes UTF-8 bytes through)
    assert go_json_escape("привет мир 🚀") == "привет мир 🚀"


def test_go_time_format():
    t = dt.datetime(2024, 1, 2, 3, 4, 5, tzinfo=UTC)
    assert format_go_time(t) == "2024-01-02T03:04:05Z"


def test_go_time_zero_value():
    assert format_go_time(None) == "0001-01-01T00:00:00Z"


def test_go_time_fractional_trimmed():
    t = dt.datetime(2024, 1, 2, 3, 4, 5, 120000, tzinfo=UTC)
    assert format_go_time(t) == "2024-01-02T03:04:05.12Z"


def test_go_time_offset():
    t = dt.datetime(2024, 1, 2, 3, 4, 5, tzinfo=dt.timezone(dt.timedelta(hours=2)))
    assert format_go_time(t) == "2024-01-02T03:04:05+02:00"


def test_default_post_is_valid_json_with_exact_field_order():
    p = Post()
    s = p.to_json()
    obj = json.loads(s)
    keys = list(obj.keys())
    # First and last few fields in Go struct declaration order
    assert keys[:6] == [
        "post_link", "channel_id", "post_uid", "url", "published_at",
        "created_at",
    ]
    assert keys[-6:] == [
        "media_url", "comments", "reactions", "outlinks", "capture_time",
        "handle",
    ]
    assert len(keys) == 65  # reference model/data.go Post json tag count


def test_nil_slices_marshal_null_empty_marshal_brackets():
    p = Post()
    obj = json.loads(p.to_json())
    assert obj["list_ids"] is None         # nil slice -> null
    assert obj["outlinks"] is None
    p2 = Post(outlinks=[], comments=[], reactions={})
    obj2 = json.loads(p2.to_json())
    assert obj2["outlinks"] == []
    assert obj2["comments"] == []
    assert obj2["reactions"] == {}


def test_reactions_sorted_like_go_maps():
    p = Post(reactions={"👍": 5, "🔥": 2, "a": 1})
    s = p.to_json()
    # Go sorts map keys lexicographically (by UTF-8 bytes)
    ordered = [k for k in json.loads(s)["reactions"]]
    assert ordered == sorted(["👍", "🔥", "a"])


def test_nested_channel_data_shape():
    cd = ChannelData(
        channel_id="123", channel_name="Test",
        channel_engagement_data=EngagementData(follower_count=10, post_count=3),
        channel_url="https://t.me/c/test",
    )
    obj = json.loads(cd.to_json())
    assert obj["channel_engagement_data"]["follower_count"] == 10
    assert obj["published_at"] == "0001-01-01T00:00:00Z"


def test_performance_scores_null_pattern():
    ps = PerformanceScores()
    assert ps.to_json() == '{"likes":null,"shares":null,"comments":null,"views":0}'


def test_comment_encoding():
    c = Comment(text='say "x"', reactions={"👍": 1}, view_count=2, handle="u")
    obj = json.loads(c.to_json())
    assert obj["text"] == 'say "x"'
    assert obj["reactions"] == {"👍": 1}


def test_jsonl_line_roundtrip():
    p = Post(
        post_link="https://t.me/chan/5",
        channel_id="42",
        published_at=dt.datetime(2024, 5, 1, tzinfo=UTC),
        description="hello <world> & 'stuff'\nnewline",
        outlinks=["abcde", "fghij"],
        post_type=["messageText"],
    )
    line = p.to_jsonl()
    assert line.endswith("\n")
    obj = json.loads(line)
    assert obj["description"] == "hello <world> & 'stuff'\nnewline"
    assert "\\u003cworld\\u003e" in line  # raw bytes show Go-style escapes
    assert obj["outlinks"] == ["abcde", "fghij"]


def test_reactions_sorted_map_keys():
    """Go encoding/json sorts map keys; reactions must serialize in
    byte-sorted emoji order regardless of insertion order."""
    import json as _json

    p = Post(
        post_link="x", channel_id="1", post_uid="u", url="x",
        published_at=dt.datetime(2024, 1, 1, tzinfo=UTC),
        created_at=dt.datetime(2024, 1, 1, tzinfo=UTC),
        reactions={"🔥": 2, "❤": 5, "👍": 1},
    )
    line = p.to_jsonl()
    obj = _json.loads(line)
    assert obj["reactions"] == {"🔥": 2, "❤": 5, "👍": 1}
    # raw byte order: keys appear sorted by their UTF-8 bytes
    raw = line[line.index('"reactions"'):]
    pos = {e: raw.index(_json.dumps(e, ensure_ascii=False))
           for e in ("❤", "👍", "🔥")}
    order = sorted(pos, key=pos.get)
    assert order == sorted(order, key=lambda s: s.encode())


def test_go_float_encoder_parity():
    """Go encoding/json floatEncoder rules: 'f' format for
    1e-6 <= |v| < 1e21, 'e' otherwise with e-0X -> e-X cleanup.
    Expected strings computed with Go strconv semantics."""
    from crawler_amd.models.post import _enc_float

    cases = [
        (0.0, "0"),
        (1.5, "1.5"),
        (-2.25, "-2.25"),
        (0.001, "0.001"),
        (0.0001, "0.0001"),
        (1e-05, "0.00001"),         # python repr says 1e-05
        (1.5e-05, "0.000015"),
        (-3.25e-05, "-0.0000325"),
        (1e-06, "0.000001"),        # boundary: still 'f' in Go
        (9.9e-07, "9.9e-7"),        # below 1e-6: 'e' + cleanup
        (1e-07, "1e-7"),
        (-1e-07, "-1e-7"),
        (1.25e-08, "1.25e-8"),
        (1e-10, "1e-10"),           # 2-digit exponent: no cleanup
        (5e-324, "5e-324"),         # denormal min
        (1e21, "1e+21"),            # >= 1e21: 'e' form
        (2.5e22, "2.5e+22"),
        (1e20, "100000000000000000000"),  # integral < 1e21: bare
        (123456789.5, "123456789.5"),
    ]
    for v, want in cases:
        assert _enc_float(v) == want, (v, _enc_float(v), want)
    import pytest

    for bad in (float("nan"), float("inf"), float("-inf")):
        with pytest.raises(ValueError):
            _enc_float(bad)
