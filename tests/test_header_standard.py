"""CPU tests for header_standard (reference test_header_standard.py)."""

import numpy as np

from bifrost_amd.header_standard import enforce_header_standard

GOOD = {"nchans": 1, "nifs": 1, "nbits": 8, "fch1": 100.0, "foff": 1e-5,
        "tstart": 1e5, "tsamp": 1e-5}


def test_simple_header():
    assert enforce_header_standard(dict(GOOD))


def test_numpy_types():
    hdr = dict(GOOD)
    hdr["nchans"] = np.int64(1)
    hdr["fch1"] = np.float64(100.0)
    assert enforce_header_standard(hdr)


def test_extra_parameters():
    hdr = dict(GOOD, my_extra_param=50)
    assert enforce_header_standard(hdr)


def test_empty_header():
    assert not enforce_header_standard({})


def test_missing_parameter():
    hdr = dict(GOOD)
    del hdr["foff"]
    assert not enforce_header_standard(hdr)


def test_bad_type():
    hdr = dict(GOOD, nchans=1.5)
    assert not enforce_header_standard(hdr)


def test_below_minimum():
    hdr = dict(GOOD, nbits=0)
    assert not enforce_header_standard(hdr)


def test_not_a_dict():
    assert not enforce_header_standard([1, 2, 3])
