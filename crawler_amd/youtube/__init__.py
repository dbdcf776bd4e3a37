"""crawler_amd.youtube — YouTube platform support over a synthetic index.

Replaces the reference's google-api-go client (client/youtube_client.go)
with a deterministic synthetic video index while preserving the sampling
semantics: random prefix sampling (McGrady et al. 2023 style,
client/youtube_client.go:1112-1543), snowball, channel mode, the
>min-videos channel gate, 50-ID list batches, and YouTube API quota
economics (search=100 units, list=1 unit, 10k/day)."""
from .synth import SyntheticYouTubeIndex, YouTubeChannel, YouTubeVideo  # noqa: F401
from .client import SyntheticYouTubeClient, QuotaExceeded  # noqa: F401
from .convert import convert_video_to_post, parse_iso8601_duration  # noqa: F401
