"""ProcLog: per-block status files (reference python/bifrost/proclog.py
surface; files under /dev/shm/bifrost_amd/<pid>/ or $BIFROST_PROCLOG_DIR)."""

from bifrost_amd.libbifrost import _bf, _check, BifrostObject

__all__ = ["ProcLog"]


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
