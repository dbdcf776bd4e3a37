from .state import (  # noqa: F401
    BaseStateManager,
    CrawlMetadata,
    EdgeRecord,
    LocalStateManager,
    Page,
    PageMessage,
    PendingEdge,
    PendingEdgeBatch,
    RandomWalkStore,
    StateManagerFactory,
)
