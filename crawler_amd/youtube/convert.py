"""YouTube video -> unified Post conversion.

Parity: crawler/youtube/youtube_crawler.go convertVideoToPost (:530-836):
- engagement = likes + comments + views/100
- thumbnail quality preference maxres > high > medium > default
- ISO-8601 duration -> seconds; "P0D" treated as null
- OCRData rows per available thumbnail
- performance scores from likes/comments/views; has_embed_media = true
- outlinks = URLs extracted from the description (trailing punctuation
  trimmed, deduped; youtube_crawler.go:489-510)
- media_data document name "<id>-<sanitized title>.mp4"
- platform_name "youtube", post_type ["video"]
"""
from __future__ import annotations

import datetime as _dt
import re
from typing import List, Optional

from ..models.post import (
    ChannelData,
    EngagementData,
    MediaData,
    OCRData,
    PerformanceScores,
    Post,
)
from .synth import YouTubeChannel, YouTubeVideo

_ISO_DUR = re.compile(
    r"^P(?:(?P<days>\d+)D)?(?:T(?:(?P<hours>\d+)H)?(?:(?P<minutes>\d+)M)?"
    r"(?:(?P<seconds>\d+)S)?)?$"
)
_URL_RE = re.compile(r"https?://[^\s<>\"]+")
_SANITIZE_RE = re.compile(r"[^A-Za-z0-9._-]+")


def parse_iso8601_duration(s: str) -> Optional[int]:
    m = _ISO_DUR.match(s or "")
    if not m:
        return None
    total = 0
    total += int(m.group("days") or 0) * 86400
    total += int(m.group("hours") or 0) * 3600
    total += int(m.group("minutes") or 0) * 60
    total += int(m.group("seconds") or 0)
    return total


def extract_urls(text: str) -> List[str]:
    out = []
    for m in _URL_RE.finditer(text or ""):
        url = m.group(0).rstrip(",.;:!?()'\"")
        if url not in out:
            out.append(url)
    return out


def sanitize_filename(name: str) -> str:
    return _SANITIZE_RE.sub("_", name)[:80]


def convert_video_to_post(
    video: YouTubeVideo,
    channel: Optional[YouTubeChannel],
    crawl_label: str = "",
    now: Optional[_dt.datetime] = None,
) -> Post:
    now = now or _dt.datetime.now(_dt.timezone.utc)
    channel_name = channel.title if channel else video.channel_id
    engagement = int(video.like_count + video.comment_count
                     + video.view_count // 100)
    thumb = ""
    for q in ("maxres", "high", "medium", "default"):
        if video.thumbnails.get(q):
            thumb = video.thumbnails[q]
            break
    url = f"https://www.youtube.com/watch?v={video.id}"
    video_len = None
    if video.duration and video.duration != "P0D":
        video_len = parse_iso8601_duration(video.duration)
    ocr = [
        OCRData(ocr_text=f"YouTube thumbnail: {q} quality", thumb_url=u)
        for q, u in sorted(video.thumbnails.items()) if u
    ] or None
    cd = ChannelData(
        channel_id=video.channel_id,
        channel_name=channel_name,
        channel_description=channel.description if channel else "",
        channel_profile_image=(
            channel.thumbnails.get("default", "") if channel else ""
        ),
        channel_engagement_data=EngagementData(
            follower_count=channel.subscriber_count if channel else 0,
            post_count=channel.video_count if channel else 0,
            views_count=channel.view_count if channel else 0,
        ),
        channel_url_external=(
            f"https://www.youtube.com/channel/{video.channel_id}"
        ),
        channel_url=f"https://www.youtube.com/channel/{video.channel_id}",
        country_code=channel.country if channel else "",
        published_at=channel.published_at if channel else None,
    )
    return Post(
        post_link=url,
        channel_id=video.channel_id,
        post_uid=video.id,
        url=url,
        published_at=video.published_at,
        created_at=now.replace(microsecond=0),
        language_code=video.language,
        engagement=engagement,
        view_count=video.view_count,
        like_count=video.like_count,
        comment_count=video.comment_count,
        crawl_label=crawl_label,
        channel_name=channel_name,
        video_length=video_len,
        channel_data=cd,
        platform_name="youtube",
        ocr_data=ocr,
        performance_scores=PerformanceScores(
            likes=video.like_count, comments=video.comment_count,
            views=float(video.view_count),
        ),
        has_embed_media=True,
        description=video.description,
        post_type=["video"],
        post_title=video.title,
        media_data=MediaData(
            document_name=f"{video.id}-{sanitize_filename(video.title)}.mp4"
        ),
        likes_count=video.like_count,
        comments_count=video.comment_count,
        views_count=video.view_count,
        searchable_text=f"{video.title} {video.description}",
        all_text=f"{video.title} {video.description}",
        thumb_url=thumb,
        # the reference's convertVideoToPost never sets Comments/Reactions
        # (nil slices -> null); comment COUNTS come from the API stats
        comments=None,
        reactions=None,
        outlinks=extract_urls(video.description),
        capture_time=now,
        handle=channel_name,
    )
