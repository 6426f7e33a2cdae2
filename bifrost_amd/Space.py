"""Space name <-> BFspace enum (reference python/bifrost/Space.py surface)."""

from bifrost_amd.libbifrost import _bf, _string2space, _space2string

SPACEMAP = {"auto": _bf.BF_SPACE_AUTO, "system": _bf.BF_SPACE_SYSTEM,
            "cuda": _bf.BF_SPACE_CUDA, "cuda_host": _bf.BF_SPACE_CUDA_HOST,
            "cuda_managed": _bf.BF_SPACE_CUDA_MANAGED}


class Space(object):
    def __init__(self, s):
        if isinstance(s, Space):
            self._space = s._space
        elif isinstance(s, str):
            if s not in SPACEMAP:
                raise ValueError("Invalid space: %r" % (s,))
            self._space = s
        else:
            self._space = _space2string(s)

    def as_BFspace(self):
        return _string2space(self._space)

    def __str__(self):
        return self._space

    def __repr__(self):
        return "Space('%s')" % self._space

    def __eq__(self, other):
        return str(self) == str(other)
