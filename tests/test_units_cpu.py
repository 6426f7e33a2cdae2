"""CPU tests for the pint-free units helpers (reference
python/bifrost/units.py surface; SI-prefixed s/Hz/m/B)."""

import pytest

from bifrost_amd.units import convert_units, transform_units


def test_si_prefix_conversions():
    assert convert_units(1.0, "MHz", "kHz") == pytest.approx(1000.0)
    assert convert_units(2.0, "ms", "s") == pytest.approx(0.002)
    assert convert_units(1.0, "GHz", "Hz") == pytest.approx(1e9)
    assert convert_units(1.5, "km", "m") == pytest.approx(1500.0)


def test_identity_and_none():
    assert convert_units(3.0, None, "Hz") == 3.0
    assert convert_units(3.0, "Hz", None) == 3.0
    assert convert_units(3.0, "Hz", "Hz") == 3.0
    assert convert_units(5, "beam", "beam") == 5  # unknown unit, identity


def test_incompatible_units_raise():
    with pytest.raises(ValueError):
        convert_units(1.0, "Hz", "s")


def test_transform_units():
    assert transform_units("s", 1) == "s"
    assert transform_units(None, 2) is None
    assert transform_units("s", 2) == "s^2"
