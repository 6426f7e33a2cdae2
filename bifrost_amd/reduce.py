"""reduce: axis reductions / scrunching (reference python/bifrost/reduce.py
surface; op-name -> BFreduce_op mapping per python/typehinting.py:30-33,
'POWER_' -> 'pwr', lowercased)."""

from bifrost_amd.libbifrost import _bf, _check
from bifrost_amd.ndarray import asarray

__all__ = ["reduce"]

_OPMAP = {
    "sum": 0, "mean": 1, "min": 2, "max": 3, "stderr": 4,
    "pwrsum": 5, "pwrmean": 6, "pwrmin": 7, "pwrmax": 8, "pwrstderr": 9,
}


def reduce(idata, odata, op="sum"):
    try:
        op = _OPMAP[op]
    except KeyError:
        raise ValueError("Invalid reduce op: " + str(op))
    _check(_bf.bfReduce(asarray(idata).as_BFarray(),
                        asarray(odata).as_BFarray(), op))
    return odata
