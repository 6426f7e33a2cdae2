"""Oracle restatement of bfReduce semantics (TEST INFRASTRUCTURE ONLY —
imported by tests/; never by the product path).

Restates reference src/reduce.cu:880-919 + test/test_reduce.py:47-66:
the reduced axis is the one where out.shape < in.shape; factor must
divide; accumulate in float32 (float64 for stderr scale precision is NOT
used — the reference accumulates in f32); mean = sum/n, stderr =
sum/sqrt(n); power ops square magnitudes first.
"""

import numpy as np

_REAL_OPS = {
    "sum": lambda a, ax: a.sum(axis=ax),
    "mean": lambda a, ax: a.sum(axis=ax) / a.shape[ax],
    "min": lambda a, ax: a.min(axis=ax),
    "max": lambda a, ax: a.max(axis=ax),
    "stderr": lambda a, ax: a.sum(axis=ax) / np.sqrt(a.shape[ax]),
}


def scrunch(data, factor, axis, op="sum"):
    """Reduce `axis` of `data` by `factor` (None = whole axis) with `op`.

    Returns float32 output for real/power ops, complex64 for complex
    non-power ops — the bfReduce output dtype contract.
    """
    data = np.asarray(data)
    if factor is None:
        factor = data.shape[axis]
    if data.shape[axis] % factor != 0:
        raise ValueError("factor does not divide axis length")
    axis = axis % data.ndim
    s = data.shape
    split = s[:axis] + (s[axis] // factor, factor) + s[axis + 1:]
    power = op.startswith("pwr")
    if power:
        op = op[3:]
        work = (np.abs(data.astype(np.complex64 if np.iscomplexobj(data)
                                   else np.float32)) ** 2).astype(np.float32)
    elif np.iscomplexobj(data):
        work = data.astype(np.complex64)
    else:
        work = data.astype(np.float32)
    work = work.reshape(split)
    out = _REAL_OPS[op](work, axis + 1)
    if np.iscomplexobj(out):
        return out.astype(np.complex64)
    return out.astype(np.float32)
