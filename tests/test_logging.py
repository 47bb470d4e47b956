"""utils/logging: rank-scoped loggers (SURVEY.md §5 metrics row)."""
import logging

from lstm_tensorspark_amd.utils.logging import get_logger


def test_logger_caching_and_rank_scoping(capsys):
    a = get_logger(rank=0)
    assert get_logger(rank=0) is a          # cached
    b = get_logger(rank=3)
    assert b is not a and "r3" in b.name
    assert len(a.handlers) == 1             # no handler duplication
    assert len(get_logger(rank=0).handlers) == 1
    a.info("hello-log-test")
    err = capsys.readouterr().err
    assert "hello-log-test" in err and "r0]" in err
    assert not a.propagate                  # no double-emit via root
