"""Standard ring-header validation (reference
python/bifrost/header_standard.py surface).

Required parameters: nchans/nifs/nbits (int, >=1), fch1 (float, >=0),
foff (float), tstart/tsamp (float, >=0).
"""

import numpy as np

__all__ = ["STANDARD_HEADER", "enforce_header_standard"]

# 'parameter name': (accepted types, minimum or None)
STANDARD_HEADER = {
    "nchans": ((int, np.int64), 1),
    "nifs": ((int, np.int64), 1),
    "nbits": ((int, np.int64), 1),
    "fch1": ((float, np.float64), 0),
    "foff": ((float, np.float64), None),
    "tstart": ((float, np.float64), 0),
    "tsamp": ((float, np.float64), 0),
}


def enforce_header_standard(header_dict):
    """Return True iff `header_dict` satisfies the standard above."""
    if type(header_dict) != dict:
        return False
    for parameter, (types, minimum) in STANDARD_HEADER.items():
        if parameter not in header_dict:
            return False
        if not isinstance(header_dict[parameter], types):
            return False
        if minimum is not None and header_dict[parameter] < minimum:
            return False
    return True
