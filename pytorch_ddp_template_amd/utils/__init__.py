from .dist import get_rank, get_world_size, is_main_process
from .logging import (
    RankFilter,
    StructuredFormatter,
    TqdmLoggingHandler,
    getLoggerWithRank,
    redirect_warnings_to_logger,
)

__all__ = [
    "get_rank",
    "get_world_size",
    "is_main_process",
    "getLoggerWithRank",
    "redirect_warnings_to_logger",
    "StructuredFormatter",
    "TqdmLoggingHandler",
    "RankFilter",
]
