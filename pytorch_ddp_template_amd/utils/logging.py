"""Structured, rank-aware logging.

MI355X-native re-implementation of the reference's observability layer
(reference: utils.py:9-101).  Preserves the reference's public surface and
conventions:

* log format ``[time][level][node_rank ^ local_rank][module][file:line][msg]``
  (reference utils.py:9),
* the ``log.info("msg", dict(k=v))`` structured-args convention: when
  ``record.args`` is a Mapping the formatter appends ``[k=repr(v)]`` pairs
  (reference utils.py:16-21),
* tz-aware millisecond timestamps (reference utils.py:23-31),
* tqdm-safe emission so progress bars don't tear (reference utils.py:34-46),
* a rank filter injecting node/local rank into every record
  (reference utils.py:49-58),
* non-main ranks quieted to WARNING (reference utils.py:67-68) and
  ``propagate=False`` (reference utils.py:74).

Additions over the reference (the reference left ``# TODO: Add File Handler``
at utils.py:61): an optional file handler via ``log_file=``.
"""

from __future__ import annotations

import datetime
import logging
import sys
from collections.abc import Mapping

try:
    from tqdm import tqdm

    _HAVE_TQDM = True
except Exception:  # pragma: no cover - tqdm is present in this environment
    _HAVE_TQDM = False

LOG_FORMAT = (
    "[%(asctime)s][%(levelname)s][%(node_rank)s ^ %(local_rank)s]"
    "[%(module)s][%(filename)s:%(lineno)d][%(message)s]"
)


class StructuredFormatter(logging.Formatter):
    """Formatter appending ``[k=repr(v)]`` pairs for Mapping args.

    Call style: ``log.info("msg", dict(k=v))`` (reference utils.py:16-21).
    """

    def format(self, record: logging.LogRecord) -> str:
        structured = None
        if isinstance(record.args, Mapping):
            structured = record.args
            record.args = None  # prevent %-interpolation against a Mapping
        out = super().format(record)
        if structured:
            out += "".join(f"[{k}={v!r}]" for k, v in structured.items())
        return out

    def formatTime(self, record, datefmt=None):  # noqa: N802 (stdlib API)
        # tz-aware, millisecond precision (reference utils.py:23-31).
        dt = datetime.datetime.fromtimestamp(
            record.created, tz=datetime.timezone.utc
        ).astimezone()
        if datefmt:
            return dt.strftime(datefmt)
        return dt.isoformat(timespec="milliseconds")


class TqdmLoggingHandler(logging.Handler):
    """Emit through ``tqdm.write`` so active progress bars do not tear
    (reference utils.py:34-46)."""

    def emit(self, record: logging.LogRecord) -> None:
        try:
            msg = self.format(record)
            if _HAVE_TQDM:
                tqdm.write(msg, file=sys.stderr)
            else:
                sys.stderr.write(msg + "\n")
            self.flush()
        except Exception:  # pragma: no cover
            self.handleError(record)


class RankFilter(logging.Filter):
    """Inject node_rank/local_rank into every record (reference utils.py:49-58)."""

    def __init__(self, node_rank: int, local_rank: int):
        super().__init__()
        self.node_rank = node_rank
        self.local_rank = local_rank

    def filter(self, record: logging.LogRecord) -> bool:
        record.node_rank = self.node_rank
        record.local_rank = self.local_rank
        return True


def getLoggerWithRank(  # noqa: N802 - keep the reference's camelCase API
    name: str,
    node_rank: int = 0,
    local_rank: int = -1,
    log_file: str | None = None,
) -> logging.Logger:
    """Build the structured rank-aware logger (reference utils.py:61-75).

    Non-main ranks (local_rank not in {-1, 0}) are quieted to WARNING so an
    8-rank job does not emit 8 copies of every line.
    """
    logger = logging.getLogger(f"{name}.rank{node_rank}.{local_rank}")
    if getattr(logger, "_ddp_amd_configured", False):
        return logger
    logger._ddp_amd_configured = True

    level = logging.INFO if local_rank in (-1, 0) else logging.WARNING
    logger.setLevel(level)
    formatter = StructuredFormatter(LOG_FORMAT)
    handler = TqdmLoggingHandler()
    handler.setFormatter(formatter)
    logger.addHandler(handler)
    if log_file:
        fh = logging.FileHandler(log_file)
        fh.setFormatter(formatter)
        logger.addHandler(fh)
    logger.addFilter(RankFilter(node_rank, local_rank))
    logger.propagate = False
    return logger


def redirect_warnings_to_logger(logger: logging.Logger) -> None:
    """Route ``warnings.warn`` into the structured logger
    (reference utils.py:78-82)."""
    import warnings

    def showwarning(message, category, filename, lineno, file=None, line=None):
        logger.warning(
            "%s", f"{category.__name__}: {message} ({filename}:{lineno})"
        )

    warnings.showwarning = showwarning
