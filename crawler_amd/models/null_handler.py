"""Null/empty-field validation with per-platform rule tables.

Parity target: reference null_handler/main.go —
- behaviors critical / log / unavailable / optional (main.go:20-36),
- per-platform default rule tables (DefaultConfigs, main.go:70-254),
- user-config JSON merge (MergeConfigs, main.go:258-292),
- recursive struct walk with nested-struct paths, pointer/slice/map nil
  checks and zero-value checks (validateStruct, main.go:377-419),
- NullLogEvent emission per empty field (main.go:59-67, 444-475).

The walk is driven by each dataclass's field list with an explicit
python-attr -> Go-name map so rule paths match the reference exactly
(e.g. "ChannelData.ChannelEngagementData.FollowerCount").
"""
from __future__ import annotations

import dataclasses
import datetime as _dt
import json
from typing import Dict, List

from . import post as _post

BEHAVIOR_CRITICAL = "critical"
BEHAVIOR_LOG = "log"
BEHAVIOR_UNAVAILABLE = "unavailable"
BEHAVIOR_OPTIONAL = "optional"


@dataclasses.dataclass
class NullLogEvent:
    platform: str
    data_type: str
    field_name: str
    strategy_used: str
    is_platform_limit: bool
    message: str


@dataclasses.dataclass
class ValidationResult:
    valid: bool = True
    errors: List[str] = dataclasses.field(default_factory=list)
    warnings: List[str] = dataclasses.field(default_factory=list)
    unavailable_used: List[str] = dataclasses.field(default_factory=list)
    null_log_events: List[NullLogEvent] = dataclasses.field(default_factory=list)


# python attr -> Go field name, per dataclass (order matches emission order).
_GO_NAMES = {
    _post.Post: {
        "post_link": "PostLink", "channel_id": "ChannelID", "post_uid": "PostUID",
        "url": "URL", "published_at": "PublishedAt", "created_at": "CreatedAt",
        "language_code": "LanguageCode", "engagement": "Engagement",
        "view_count": "ViewCount", "like_count": "LikeCount",
        "share_count": "ShareCount", "comment_count": "CommentCount",
        "crawl_label": "CrawlLabel", "list_ids": "ListIDs",
        "channel_name": "ChannelName", "search_terms": "SearchTerms",
        "search_term_ids": "SearchTermIDs", "project_ids": "ProjectIDs",
        "exercise_ids": "ExerciseIDs", "label_data": "LabelData",
        "labels_metadata": "LabelsMetadata",
        "project_labeled_post_ids": "ProjectLabeledPostIDs",
        "labeler_ids": "LabelerIDs", "all_labels": "AllLabels",
        "label_ids": "LabelIDs", "is_ad": "IsAd",
        "transcript_text": "TranscriptText", "image_text": "ImageText",
        "video_length": "VideoLength", "is_verified": "IsVerified",
        "channel_data": "ChannelData", "platform_name": "PlatformName",
        "shared_id": "SharedID", "quoted_id": "QuotedID",
        "replied_id": "RepliedID", "ai_label": "AILabel",
        "root_post_id": "RootPostID",
        "engagement_steps_count": "EngagementStepsCount",
        "ocr_data": "OCRData", "performance_scores": "PerformanceScores",
        "has_embed_media": "HasEmbedMedia", "description": "Description",
        "repost_channel_data": "RepostChannelData", "post_type": "PostType",
        "inner_link": "InnerLink", "post_title": "PostTitle",
        "media_data": "MediaData", "is_reply": "IsReply",
        "ad_fields": "AdFields", "likes_count": "LikesCount",
        "shares_count": "SharesCount", "comments_count": "CommentsCount",
        "views_count": "ViewsCount", "searchable_text": "SearchableText",
        "all_text": "AllText",
        "contrast_agent_project_ids": "ContrastAgentProjectIDs",
        "agent_ids": "AgentIDs", "segment_ids": "SegmentIDs",
        "thumb_url": "ThumbURL", "media_url": "MediaURL",
        "comments": "Comments", "reactions": "Reactions",
        "outlinks": "Outlinks", "capture_time": "CaptureTime",
        "handle": "Handle",
    },
    _post.ChannelData: {
        "channel_id": "ChannelID", "channel_name": "ChannelName",
        "channel_description": "ChannelDescription",
        "channel_profile_image": "ChannelProfileImage",
        "channel_engagement_data": "ChannelEngagementData",
        "channel_url_external": "ChannelURLExternal",
        "channel_url": "ChannelURL", "country_code": "CountryCode",
        "published_at": "PublishedAt",
    },
    _post.EngagementData: {
        "follower_count": "FollowerCount", "following_count": "FollowingCount",
        "like_count": "LikeCount", "post_count": "PostCount",
        "views_count": "ViewsCount", "comment_count": "CommentCount",
        "share_count": "ShareCount",
    },
    _post.PerformanceScores: {
        "likes": "Likes", "shares": "Shares", "comments": "Comments",
        "views": "Views",
    },
    _post.InnerLink: {},
    _post.MediaData: {"document_name": "DocumentName"},
}

_C, _L, _U, _O = (
    BEHAVIOR_CRITICAL, BEHAVIOR_LOG, BEHAVIOR_UNAVAILABLE, BEHAVIOR_OPTIONAL
)


def _channel_rules(platform: str, prefix: str = "ChannelData.") -> Dict[str, tuple]:
    """Shared ChannelData rule shape (null_handler/main.go:76-89, 166-180)."""
    tg = platform == "telegram"
    return {
        prefix + "ChannelID": (_C, "ChannelID is required"),
        prefix + "ChannelName": (_C, "ChannelName is required"),
        prefix + "ChannelDescription": (_L, "ChannelDescription is empty"),
        prefix + "ChannelProfileImage": (_L, "ChannelProfileImage is empty"),
        prefix + "ChannelEngagementData.FollowerCount": (_L, "FollowerCount is zero"),
        prefix + "ChannelEngagementData.FollowingCount":
            (_U, f"FollowingCount not available on {'Telegram' if tg else 'YouTube'}"),
        prefix + "ChannelEngagementData.LikeCount":
            (_U, f"LikeCount not available on {'Telegram' if tg else 'YouTube'}"),
        prefix + "ChannelEngagementData.PostCount": (_L, "PostCount is zero"),
        prefix + "ChannelEngagementData.ViewsCount": (_L, "ViewsCount is zero"),
        prefix + "ChannelEngagementData.CommentCount":
            (_U, f"CommentCount not available on {'Telegram' if tg else 'YouTube'}"),
        prefix + "ChannelEngagementData.ShareCount":
            (_U, "ShareCount is zero" if tg else "ShareCount not available on YouTube"),
        prefix + "ChannelURLExternal": (_L, "ChannelURLExternal is empty"),
        prefix + "ChannelURL": (_C, "ChannelURL is required"),
        prefix + "CountryCode":
            (_U, "CountryCode is empty") if tg else (_O, "CountryCode is empty"),
        prefix + "PublishedAt":
            (_U, "PublishedAt is zero") if tg else (_L, "PublishedAt is zero"),
    }


def _telegram_rules() -> Dict[str, tuple]:
    """null_handler/main.go:162-254."""
    r = _channel_rules("telegram")
    r.update({
        "PostLink": (_C, "PostLink is required"),
        "ChannelID": (_C, "ChannelID is required"),
        "PostUID": (_C, "PostUID is required"),
        "URL": (_C, "URL is required"),
        "PublishedAt": (_C, "PublishedAt is required"),
        "CreatedAt": (_L, "CreatedAt is zero"),
        "LanguageCode": (_U, "LanguageCode is empty"),
        "Engagement": (_L, "Engagement is zero"),
        "ViewCount": (_L, "ViewCount is zero"),
        "LikeCount": (_U, "LikeCount not directly available on Telegram"),
        "ShareCount": (_L, "ShareCount is zero"),
        "CommentCount": (_L, "CommentCount is zero"),
        "CrawlLabel": (_L, "CrawlLabel is empty"),
        "ListIDs": (_U, "ListIDs is empty"),
        "ChannelName": (_L, "ChannelName is empty"),
        "SearchTerms": (_U, "SearchTerms is empty"),
        "SearchTermIDs": (_U, "SearchTermIDs is empty"),
        "ProjectIDs": (_U, "ProjectIDs is empty"),
        "ExerciseIDs": (_U, "ExerciseIDs is empty"),
        "LabelData": (_U, "LabelData is empty"),
        "LabelsMetadata": (_U, "LabelsMetadata is empty"),
        "ProjectLabeledPostIDs": (_U, "ProjectLabeledPostIDs is empty"),
        "LabelerIDs": (_U, "LabelerIDs is empty"),
        "AllLabels": (_U, "AllLabels is empty"),
        "LabelIDs": (_U, "LabelIDs is empty"),
        "IsAd": (_L, "IsAd is false"),
        "TranscriptText": (_U, "TranscriptText is empty"),
        "ImageText": (_U, "ImageText is empty"),
        "VideoLength": (_U, "VideoLength is null"),
        "IsVerified": (_U, "IsVerified is null"),
        "PlatformName": (_C, "PlatformName is required"),
        "SharedID": (_U, "SharedID is null"),
        "QuotedID": (_U, "QuotedID is null"),
        "RepliedID": (_U, "RepliedID is null"),
        "AILabel": (_U, "AILabel is null"),
        "RootPostID": (_U, "RootPostID is null"),
        "EngagementStepsCount": (_U, "EngagementStepsCount is zero"),
        "OCRData": (_U, "OCRData is empty"),
        "PerformanceScores.Likes":
            (_U, "PerformanceScores.Likes not directly available on Telegram"),
        "PerformanceScores.Shares": (_U, "PerformanceScores.Shares is null"),
        "PerformanceScores.Comments": (_U, "PerformanceScores.Comments is null"),
        "PerformanceScores.Views": (_U, "PerformanceScores.Views is zero"),
        "HasEmbedMedia": (_U, "HasEmbedMedia is null"),
        "Description": (_L, "Description is empty"),
        "RepostChannelData": (_U, "RepostChannelData is null"),
        "PostType": (_L, "PostType is empty"),
        "InnerLink": (_U, "InnerLink is empty"),
        "PostTitle": (_U, "PostTitle is null"),
        "MediaData.DocumentName": (_U, "MediaData.DocumentName is empty"),
        "IsReply": (_U, "IsReply is null"),
        "AdFields": (_U, "AdFields is null"),
        "LikesCount": (_U, "LikesCount not directly available on Telegram"),
        "SharesCount": (_L, "SharesCount is zero"),
        "CommentsCount": (_L, "CommentsCount is zero"),
        "ViewsCount": (_L, "ViewsCount is zero"),
        "SearchableText": (_U, "SearchableText is empty"),
        "AllText": (_U, "AllText is empty"),
        "ContrastAgentProjectIDs": (_U, "ContrastAgentProjectIDs is empty"),
        "AgentIDs": (_U, "AgentIDs is empty"),
        "SegmentIDs": (_U, "SegmentIDs is empty"),
        "ThumbURL": (_L, "ThumbURL is empty"),
        "MediaURL": (_L, "MediaURL is empty"),
        "Comments": (_L, "Comments is empty"),
        "Reactions": (_L, "Reactions is empty"),
        "Outlinks": (_L, "Outlinks is empty"),
        "CaptureTime": (_L, "CaptureTime is zero"),
        "Handle": (_L, "Handle is empty"),
    })
    return r


def _youtube_rules() -> Dict[str, tuple]:
    """null_handler/main.go:72-160 — differs from Telegram in the marked rows."""
    r = _channel_rules("youtube")
    r.update(_telegram_rules())
    # Re-apply channel rules (telegram update clobbered the ChannelData.*)
    r.update(_channel_rules("youtube"))
    # YouTube-specific deltas (main.go:95-159)
    r.update({
        "LanguageCode": (_L, "LanguageCode is empty"),
        "LikeCount": (_L, "LikeCount is zero"),
        "ShareCount": (_U, "ShareCount not available on YouTube"),
        "IsAd": (_U, "IsAd is false"),
        "VideoLength": (_L, "VideoLength is null"),
        "IsVerified": (_U, "IsVerified is null"),
        "OCRData": (_L, "OCRData is empty"),
        "PerformanceScores.Likes": (_L, "PerformanceScores.Likes is null"),
        "PerformanceScores.Shares":
            (_U, "PerformanceScores.Shares not available on YouTube"),
        "PerformanceScores.Comments": (_L, "PerformanceScores.Comments is null"),
        "PerformanceScores.Views": (_L, "PerformanceScores.Views is zero"),
        "HasEmbedMedia": (_L, "HasEmbedMedia is null"),
        "PostTitle": (_L, "PostTitle is null"),
        "Comments": (_U, "Comments is empty"),
        "Reactions": (_L, "Reactions not available on YouTube"),
        "LikesCount": (_L, "LikesCount is zero"),
        "SearchableText": (_L, "SearchableText is empty"),
        "AllText": (_L, "AllText is empty"),
    })
    return r


DEFAULT_RULES = {
    "telegram": _telegram_rules,
    "youtube": _youtube_rules,
}

# Unavailable rules represent platform limits (NullLogEvent.IsPlatformLimit).
_PLATFORM_LIMIT_BEHAVIORS = {BEHAVIOR_UNAVAILABLE}


class NullValidator:
    """Reflection-driven field validator (null_handler/main.go:294-419)."""

    def __init__(self, platform: str = "telegram", user_config_json: str = "{}"):
        self.platform = platform
        rules_fn = DEFAULT_RULES.get(platform, _telegram_rules)
        self.rules: Dict[str, tuple] = rules_fn()
        if user_config_json and user_config_json != "{}":
            user = json.loads(user_config_json)
            for path, cfg in user.get("rules", {}).items():
                behavior = cfg.get("behavior", BEHAVIOR_LOG)
                message = cfg.get("message", f"{path} is empty")
                self.rules[path] = (behavior, message)

    # -- public API (mirrors Validator.ValidatePost / ValidateChannelData) --

    def validate_post(self, p: _post.Post) -> ValidationResult:
        result = ValidationResult()
        self._walk("", "post", p, result)
        return result

    def validate_channel_data(self, cd: _post.ChannelData) -> ValidationResult:
        result = ValidationResult()
        self._walk("ChannelData", "channel", cd, result)
        return result

    # -- internals --

    def _walk(self, prefix: str, data_type: str, obj, result: ValidationResult):
        go_names = _GO_NAMES.get(type(obj))
        if go_names is None:
            return
        for f in dataclasses.fields(obj):
            go_name = go_names.get(f.name)
            if go_name is None:
                continue
            path = f"{prefix}.{go_name}" if prefix else go_name
            value = getattr(obj, f.name)
            if dataclasses.is_dataclass(value) and not isinstance(
                value, _dt.datetime
            ):
                # Nested struct (not time.Time): recurse, except InnerLink
                # which has no fields and is rule-checked as a whole.
                if isinstance(value, _post.InnerLink):
                    self._handle_empty(path, data_type, result)
                else:
                    self._walk(path, data_type, value, result)
                continue
            if self._is_empty(value):
                self._handle_empty(path, data_type, result)

    @staticmethod
    def _is_empty(value) -> bool:
        if value is None:
            return True
        if isinstance(value, str):
            return value == ""
        if isinstance(value, bool):
            return not value
        if isinstance(value, (int, float)):
            return value == 0
        if isinstance(value, (list, dict)):
            return len(value) == 0
        if isinstance(value, _dt.datetime):
            return value == _post.ZERO_TIME or value.replace(
                tzinfo=_dt.timezone.utc
            ) == _post.ZERO_TIME
        return False

    def _handle_empty(self, path: str, data_type: str, result: ValidationResult):
        rule = self.rules.get(path)
        if rule is None:
            return  # unknown field: no rule, no action (main.go:444-448)
        behavior, message = rule
        event = NullLogEvent(
            platform=self.platform,
            data_type=data_type,
            field_name=path,
            strategy_used=behavior,
            is_platform_limit=behavior in _PLATFORM_LIMIT_BEHAVIORS,
            message=message,
        )
        if behavior == BEHAVIOR_CRITICAL:
            result.valid = False
            result.errors.append(message)
            result.null_log_events.append(event)
        elif behavior == BEHAVIOR_LOG:
            result.warnings.append(message)
            result.null_log_events.append(event)
        elif behavior == BEHAVIOR_UNAVAILABLE:
            result.unavailable_used.append(message)
            result.null_log_events.append(event)
        # optional: no action
