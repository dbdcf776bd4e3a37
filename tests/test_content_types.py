"""parse_message content-type switch matrix — one test row per TDLib
message content type (reference tdutils.go:443-587 switch), pinning the
description / thumb / media-path rules for all 15 types."""
import datetime as dt

import pytest

from crawler_amd.ops.golden import (FormattedText, SynthMessage,
                                    parse_message)

UTC = dt.timezone.utc
NOW = dt.datetime(2026, 1, 1, tzinfo=UTC)


def msg(ct, **kw):
    return SynthMessage(chat_id=-1001000000007, msg_id=5 << 20,
                        date=1_700_000_000, content_type=ct, **kw)


def parse(m, skip_media=True, resolver=None):
    return parse_message(m, channel_username="chan5", chat_title="T",
                         member_count=10, post_count=5, total_views=100,
                         skip_media=skip_media,
                         media_path_resolver=resolver, now=NOW)


# description source per type (tdutils.go:443-587)
CASES = [
    ("messageText", {"text": FormattedText(text="body text")},
     "body text"),
    ("messagePhoto", {"caption": FormattedText(text="photo cap")},
     "photo cap"),
    ("messageVideo", {"caption": FormattedText(text="vid cap")},
     "vid cap"),
    ("messageAnimation", {"caption": FormattedText(text="gif cap")},
     "gif cap"),
    ("messagePaidMedia", {"caption": FormattedText(text="paid cap")},
     "paid cap"),
    ("messageAnimatedEmoji", {"emoji": "🔥"}, "🔥"),
    ("messagePoll", {"poll_question": "which?"}, "which?"),
    ("messageGiveaway", {"giveaway_prize": "prize!"}, "prize!"),
    ("messageDocument", {"document_name": "report.pdf"}, "report.pdf"),
    ("messageSticker", {}, ""),
    ("messageVideoNote", {}, ""),
    ("messageAudio", {}, ""),
    ("messageVoiceNote", {}, ""),
    ("messageGiveawayWinners", {}, ""),
    ("messageGiveawayCompleted", {}, ""),
]


@pytest.mark.parametrize("ct,fields,want", CASES,
                         ids=[c[0] for c in CASES])
def test_description_per_content_type(ct, fields, want):
    p = parse(msg(ct, **fields))
    assert p.description == want
    # the caption/body never leaks into other types
    if want == "":
        assert p.description == ""


def test_caption_ignored_for_text_type():
    p = parse(msg("messageText", text=FormattedText(text="body"),
                  caption=FormattedText(text="stray caption")))
    assert p.description == "body"


def test_text_ignored_for_media_types():
    p = parse(msg("messagePhoto", text=FormattedText(text="stray"),
                  caption=FormattedText(text="cap")))
    assert p.description == "cap"


def test_video_path_types_set_media_url_without_download():
    """VideoNote/Document surface the remote id as the video path even
    with skip_media (tdutils.go:545-587)."""
    for ct in ("messageVideoNote", "messageDocument"):
        p = parse(msg(ct, video_remote_id="RID123"))
        assert p.media_url == "RID123"
    # video content does NOT (it would need a download)
    p = parse(msg("messageVideo", video_remote_id="RID123"))
    assert p.media_url == ""


def test_thumb_resolved_only_when_media_enabled():
    calls = []

    def resolver(rid):
        calls.append(rid)
        return f"/media/{rid}"

    m = msg("messagePhoto", thumb_remote_id="TH9",
            caption=FormattedText(text="c"))
    p = parse(m, skip_media=True, resolver=resolver)
    assert p.thumb_url == "" and calls == []
    p = parse(m, skip_media=False, resolver=resolver)
    assert p.thumb_url == "/media/TH9" and calls == ["TH9"]


def test_thumbless_types_never_resolve():
    calls = []
    m = msg("messagePoll", thumb_remote_id="TH9", poll_question="q")
    p = parse(m, skip_media=False,
              resolver=lambda r: calls.append(r) or "x")
    assert p.thumb_url == "" and calls == []
