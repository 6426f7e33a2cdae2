"""CPU tests for the views package (reference views/basic_views.py
surface): header transforms applied between pipeline blocks."""

import numpy as np

import bifrost_amd as bf
from bifrost_amd import views
from tests.test_pipeline_cpu import CollectBlock, NumpySourceBlock


def _run(data, view_fn, labels=None):
    out = []
    with bf.Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=4, labels=labels)
        viewed = view_fn(src)
        sink = CollectBlock(viewed, out)
        pipe.run()
    return np.concatenate(out, axis=0), sink.headers[0]


def test_rename_axis():
    data = np.arange(32, dtype=np.float32).reshape(8, 4)
    got, hdr = _run(data, lambda b: views.rename_axis(b, "d0", "chan"))
    assert hdr["_tensor"]["labels"] == ["time", "chan"]
    np.testing.assert_array_equal(got, data)


def test_add_delete_axis():
    data = np.arange(32, dtype=np.float32).reshape(8, 4)
    got, hdr = _run(data, lambda b: views.add_axis(b, 1, label="pol"))
    assert hdr["_tensor"]["shape"] == [-1, 1, 4]
    assert hdr["_tensor"]["labels"][1] == "pol"

    got, hdr = _run(data,
                    lambda b: views.delete_axis(views.add_axis(b, 1), 1))
    assert hdr["_tensor"]["shape"] == [-1, 4]


def test_split_axis():
    data = np.arange(64, dtype=np.float32).reshape(8, 8)
    got, hdr = _run(data, lambda b: views.split_axis(b, 1, 4,
                                                     label="fine"))
    assert hdr["_tensor"]["shape"] == [-1, 2, 4]
    assert hdr["_tensor"]["labels"] == ["time", "d0", "fine"]
    np.testing.assert_array_equal(got.reshape(8, 8), data)


def test_merge_axes():
    data = np.arange(64, dtype=np.float32).reshape(8, 2, 4)
    got, hdr = _run(
        data, lambda b: views.merge_axes(
            views.reinterpret_axis(
                views.reinterpret_axis(b, 1, "coarse", [0, 4], "s"),
                2, "fine", [0, 1], "s"),
            1, 2, label="chan"))
    assert hdr["_tensor"]["shape"] == [-1, 8]
    assert hdr["_tensor"]["labels"] == ["time", "chan"]
    np.testing.assert_array_equal(got, data.reshape(8, 8))


def test_astype_view():
    data = np.arange(32, dtype=np.float32).reshape(8, 4)
    got, hdr = _run(data, lambda b: views.astype(b, "cf32"))
    assert hdr["_tensor"]["dtype"] == "cf32"
    assert hdr["_tensor"]["shape"] == [-1, 2]
    np.testing.assert_array_equal(got.view(np.float32).reshape(8, 4),
                                  data)


def test_custom_and_reverse_scale():
    data = np.arange(32, dtype=np.float32).reshape(8, 4)

    def bump_time_tag(hdr):
        hdr["time_tag"] = 42
        return hdr

    got, hdr = _run(data, lambda b: views.custom(b, bump_time_tag))
    assert hdr["time_tag"] == 42

    got, hdr = _run(data,
                    lambda b: views.reverse_scale(b, "d0"),
                    labels=["time", "d0"])
    assert hdr["_tensor"]["scales"][1][1] == -1


def test_units_convert():
    from bifrost_amd.units import convert_units
    assert convert_units(1.0, "MHz", "kHz") == 1000.0
    assert convert_units(2.0, "ms", "s") == 0.002
    assert convert_units(5, None, "s") == 5
    import pytest
    with pytest.raises(ValueError):
        convert_units(1.0, "MHz", "s")
