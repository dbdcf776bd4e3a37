from .post import (  # noqa: F401
    ChannelData,
    Comment,
    EngagementData,
    InnerLink,
    MediaData,
    OCRData,
    PerformanceScores,
    Post,
    ZERO_TIME,
    format_go_time,
    go_json_escape,
)
from .null_handler import NullValidator, ValidationResult  # noqa: F401
