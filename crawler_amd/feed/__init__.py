from .synth import SyntheticFeed, FeedConfig  # noqa: F401
