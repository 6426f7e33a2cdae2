"""CPU tests for ProcLog write + the load_by_filename/load_by_pid readers
(reference python/bifrost/proclog.py surface)."""

import os

from bifrost_amd import proclog


def test_proclog_update_and_load_by_pid():
    pl = proclog.ProcLog("plt_block/status")
    pl.update({"n": 42, "rate": 3.25, "state": "running"})
    got = proclog.load_by_pid(os.getpid())
    assert got["plt_block"]["status"] == {"n": 42, "rate": 3.25,
                                          "state": "running"}


def test_proclog_update_replaces_contents():
    pl = proclog.ProcLog("plt_block2/perf")
    pl.update({"a": 1})
    pl.update({"b": 2})
    got = proclog.load_by_pid(os.getpid())["plt_block2"]["perf"]
    assert got == {"b": 2}


def test_load_by_filename_types():
    pl = proclog.ProcLog("plt_block3/info")
    pl.update("i : 7\nf : 1.5\ns : hello world\nmalformed line")
    base = os.path.join(proclog.PROCLOG_DIR, str(os.getpid()))
    path = os.path.join(base, "plt_block3", "info")
    if not os.path.exists(path):
        path += ".log"
    got = proclog.load_by_filename(path)
    assert got == {"i": 7, "f": 1.5, "s": "hello world"}


def test_load_by_pid_missing_raises():
    import pytest
    with pytest.raises(RuntimeError):
        proclog.load_by_pid(99999999)
