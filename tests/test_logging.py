"""utils/logging: rank-scoped loggers (SURVEY.md §5 metrics row)."""
import logging

from lstm_tensorspark_amd.utils.logging import get_logger


def test_logger_caching_and_rank_scoping(capsys):
    # unique name: the logger must be CREATED inside this test, so its
    # stderr handler binds the capsys-replaced stream (a logger cached by
    # an earlier test would hold the real stderr and capsys would miss it)
    a = get_logger(rank=0, name="logtest_a")
    assert get_logger(rank=0, name="logtest_a") is a    # cached
    b = get_logger(rank=3, name="logtest_a")
    assert b is not a and "r3" in b.name
    assert len(a.handlers) == 1             # no handler duplication
    assert len(get_logger(rank=0, name="logtest_a").handlers) == 1
    a.info("hello-log-test")
    err = capsys.readouterr().err
    assert "hello-log-test" in err and "r0]" in err
    assert not a.propagate                  # no double-emit via root
