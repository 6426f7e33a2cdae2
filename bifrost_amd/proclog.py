"""ProcLog: per-block status files (reference python/bifrost/proclog.py
surface; files under /dev/shm/bifrost_amd/<pid>/ or $BIFROST_PROCLOG_DIR)."""

import os
import time

from bifrost_amd.libbifrost import _bf, _check, BifrostObject

__all__ = ["ProcLog", "load_by_filename", "load_by_pid", "PROCLOG_DIR"]

PROCLOG_DIR = os.environ.get("BIFROST_PROCLOG_DIR", "/dev/shm/bifrost_amd")


def _multi_convert(value):
    """Best-effort str -> int -> float conversion (reference
    proclog.py:_multi_convert)."""
    try:
        return int(value, 10)
    except ValueError:
        try:
            return float(value)
        except ValueError:
            return value


def load_by_filename(filename):
    """Parse one ProcLog file ('key : value' lines) into a dict."""
    contents = {}
    for _ in range(5):
        if os.path.getsize(filename) != 0:
            break
        time.sleep(0.001)
    with open(filename, "r") as fh:
        for line in fh.read().split("\n"):
            try:
                key, value = line.split(":", 1)
            except ValueError:
                continue
            contents[key.strip()] = _multi_convert(value.strip())
    return contents


def load_by_pid(pid, include_rings=False):
    """Parse every ProcLog file of a process into
    {block: {log: {key: value}}}."""
    base_dir = os.path.join(PROCLOG_DIR, str(pid))
    if not os.path.isdir(base_dir):
        raise RuntimeError("Cannot find log directory associated with PID "
                           "%s" % pid)
    contents = {}
    for parent, _, filenames in os.walk(base_dir):
        for filename in filenames:
            path = os.path.join(parent, filename)
            log_name = os.path.splitext(os.path.basename(path))[0]
            block_name = os.path.relpath(parent, base_dir)
            if block_name.split(os.sep)[0] == "rings" and not include_rings:
                continue
            contents.setdefault(block_name, {})[log_name] = \
                load_by_filename(path)
    return contents


class ProcLog(BifrostObject):
    def __init__(self, name):
        BifrostObject.__init__(self, _bf.bfProcLogCreate, _bf.bfProcLogDestroy,
                               name.encode())

    def update(self, contents):
        """Updates (replaces) the contents of the log.
        contents: string, or dict of key/value pairs."""
        if contents is None:
            raise ValueError("Contents cannot be None")
        if isinstance(contents, dict):
            contents = "\n".join("%s : %s" % (k, v)
                                 for k, v in contents.items())
        _check(_bf.bfProcLogUpdate(self.obj, contents.encode()))
