"""Mirror of reference test/test_version.py: the version CLI must run."""

import subprocess
import sys


def test_plain_version():
    out = subprocess.check_output([sys.executable, "-m",
                                   "bifrost_amd.version"])
    assert b"bifrost_amd" in out


def test_version_config():
    out = subprocess.check_output([sys.executable, "-m",
                                   "bifrost_amd.version", "--config"])
    assert b"Configuration:" in out
    assert b"gfx950" in out
