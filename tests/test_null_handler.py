"""Null-field validator tests (reference null_handler/main.go:70-475):
per-platform rule behaviors, nested paths, NullLogEvents, user-config
merge."""
import datetime as dt
import json

from crawler_amd.models import (
    ChannelData,
    EngagementData,
    NullValidator,
    PerformanceScores,
    Post,
)

UTC = dt.timezone.utc


def full_post(**kw):
    d = dict(
        post_link="https://t.me/c/x/1", channel_id="1", post_uid="1-x",
        url="https://t.me/c/x/1",
        published_at=dt.datetime(2024, 1, 1, tzinfo=UTC),
        created_at=dt.datetime(2024, 1, 1, tzinfo=UTC),
        platform_name="Telegram",
        channel_data=ChannelData(
            channel_id="1", channel_name="X",
            channel_url="https://t.me/c/x",
            channel_engagement_data=EngagementData(
                follower_count=10, post_count=5, views_count=100,
            ),
        ),
    )
    d.update(kw)
    return Post(**d)


def test_valid_post_passes():
    v = NullValidator("telegram")
    res = v.validate_post(full_post())
    assert res.valid
    assert not res.errors


def test_missing_critical_fields_fail():
    v = NullValidator("telegram")
    res = v.validate_post(full_post(post_link="", url=""))
    assert not res.valid
    assert "PostLink is required" in res.errors
    assert "URL is required" in res.errors


def test_nested_channel_critical():
    v = NullValidator("telegram")
    p = full_post()
    p.channel_data.channel_url = ""
    res = v.validate_post(p)
    assert not res.valid
    assert "ChannelURL is required" in res.errors


def test_zero_published_at_is_critical():
    v = NullValidator("telegram")
    res = v.validate_post(full_post(published_at=None))
    assert not res.valid
    assert "PublishedAt is required" in res.errors


def test_log_fields_warn_but_pass():
    v = NullValidator("telegram")
    res = v.validate_post(full_post(description=""))
    assert res.valid
    assert "Description is empty" in res.warnings


def test_unavailable_fields_tracked_with_platform_limit():
    v = NullValidator("telegram")
    res = v.validate_post(full_post())
    # LanguageCode "" is 'unavailable' on Telegram (main.go:197)
    assert "LanguageCode is empty" in res.unavailable_used
    ev = [e for e in res.null_log_events
          if e.field_name == "LanguageCode"][0]
    assert ev.is_platform_limit
    assert ev.strategy_used == "unavailable"


def test_platform_rule_differences():
    # LanguageCode: unavailable on Telegram, LOG on YouTube (main.go:95)
    p = full_post(platform_name="youtube")
    res_tg = NullValidator("telegram").validate_post(p)
    res_yt = NullValidator("youtube").validate_post(p)
    assert "LanguageCode is empty" in res_tg.unavailable_used
    assert "LanguageCode is empty" in res_yt.warnings


def test_channel_data_direct_validation():
    v = NullValidator("telegram")
    cd = ChannelData()  # everything empty
    res = v.validate_channel_data(cd)
    assert not res.valid
    assert "ChannelID is required" in res.errors


def test_user_config_merge_overrides():
    """MergeConfigs (main.go:258-292): user rules override defaults."""
    user = json.dumps({
        "rules": {
            "Description": {"behavior": "critical",
                            "message": "Description now required"},
            "PostLink": {"behavior": "log", "message": "PostLink relaxed"},
        }
    })
    v = NullValidator("telegram", user_config_json=user)
    res = v.validate_post(full_post(description="", post_link=""))
    assert "Description now required" in res.errors
    assert "PostLink relaxed" in res.warnings
    assert not any("PostLink is required" in e for e in res.errors)


def test_events_cover_every_empty_ruled_field():
    v = NullValidator("telegram")
    res = v.validate_post(full_post())
    names = {e.field_name for e in res.null_log_events}
    # a selection of known-empty fields with rules
    assert {"CrawlLabel", "ListIDs", "TranscriptText"} <= names
