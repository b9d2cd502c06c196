"""Logger format parity with the reference (utils.py:9-75)."""

import logging

from pytorch_ddp_template_amd.utils import (
    StructuredFormatter,
    getLoggerWithRank,
)
from pytorch_ddp_template_amd.utils.logging import LOG_FORMAT, RankFilter


def make_record(msg, args=None):
    rec = logging.LogRecord("mod", logging.INFO, "f.py", 12, msg, args, None)
    rec.node_rank = 3
    rec.local_rank = 1
    return rec


def test_structured_kv_pairs_appended():
    f = StructuredFormatter(LOG_FORMAT)
    out = f.format(make_record("hello", {"loss": 0.5, "step": 7}))
    assert "[loss=0.5]" in out
    assert "[step=7]" in out
    assert "[3 ^ 1]" in out
    assert "[f.py:12]" in out
    assert "[hello]" in out


def test_plain_message_unchanged():
    f = StructuredFormatter(LOG_FORMAT)
    out = f.format(make_record("plain %d", (5,)))
    assert "[plain 5]" in out


def test_rank_filter_injects_ranks():
    flt = RankFilter(2, 0)
    rec = logging.LogRecord("m", logging.INFO, "f", 1, "x", None, None)
    assert flt.filter(rec)
    assert rec.node_rank == 2 and rec.local_rank == 0


def test_logger_levels_by_rank():
    main = getLoggerWithRank("t_main", 0, 0)
    other = getLoggerWithRank("t_other", 0, 3)
    assert main.level == logging.INFO
    assert other.level == logging.WARNING
    assert main.propagate is False


def test_log_call_convention_does_not_raise(caplog):
    log = getLoggerWithRank("t_conv", 0, -1)
    log.info("msg", dict(k=1))  # the reference's call style (utils.py:18-21)
