"""Unit conversion helpers (reference python/bifrost/units.py surface).

The reference uses pint; this restatement handles the SI-prefixed units
the pipeline actually passes around (time and frequency scales) without
an external dependency.
"""

__all__ = ["convert_units", "transform_units"]

_PREFIXES = {"f": 1e-15, "p": 1e-12, "n": 1e-9, "u": 1e-6, "m": 1e-3,
             "": 1.0, "k": 1e3, "M": 1e6, "G": 1e9, "T": 1e12}
_BASES = ("s", "Hz", "m", "B")


def _parse(units):
    if units is None:
        return None, 1.0
    u = str(units).strip()
    for base in _BASES:
        if u == base:
            return base, 1.0
        if u.endswith(base) and u[:-len(base)] in _PREFIXES:
            return base, _PREFIXES[u[:-len(base)]]
    return u, 1.0  # unknown unit: only identity conversions allowed


def convert_units(value, old_units, new_units):
    if old_units is None or new_units is None or old_units == new_units:
        return value
    ob, of = _parse(old_units)
    nb, nf = _parse(new_units)
    if ob != nb:
        raise ValueError("Cannot convert units %s to %s"
                         % (old_units, new_units))
    return value * (of / nf)


def transform_units(units, exponent):
    if exponent == 1 or units is None:
        return units
    return "%s^%s" % (units, exponent)
