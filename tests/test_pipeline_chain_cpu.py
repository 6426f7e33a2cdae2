"""Long-chain CPU pipeline stress tests (reference test_pipeline_cpu.py
run_test_simple_copy model): a synthesized 16-bit sigproc filterbank
through 20+ chained copy blocks with mixed gulp sizes, the full views
gauntlet, and CallbackBlock header/data assertions."""

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd import sigproc
from bifrost_amd.blocks.copy import CopyBlock

NFRAME = 2150  # fills the minimum ring a few times over
NCHANS = 2
TSTART_MJD = 50000.0
TSAMP = 8e-05
FCH1, FOFF = 433.968, -0.062


@pytest.fixture()
def fil_file(tmp_path):
    fname = str(tmp_path / "2chan16bit.fil")
    rng = np.random.RandomState(0)
    data = rng.randint(0, 65536, size=(NFRAME, 1, NCHANS)) \
        .astype(np.uint16)
    hdr = {"telescope_id": 0, "machine_id": 0, "data_type": 1,
           "nchans": NCHANS, "nbits": 16, "nifs": 1,
           "tstart": TSTART_MJD, "tsamp": TSAMP,
           "fch1": FCH1, "foff": FOFF}
    with open(fname, "wb") as f:
        sigproc.write_header(hdr, f)
        data.tofile(f)
    return fname, data


class CallbackBlock(CopyBlock):
    """Calls user callbacks on sequence/data (reference
    test_pipeline_cpu.py:36-55)."""

    def __init__(self, iring, seq_callback, data_callback, data_ref=None,
                 *args, **kwargs):
        super(CallbackBlock, self).__init__(iring, *args, **kwargs)
        self.seq_callback = seq_callback
        self.data_callback = data_callback
        self.data_ref = data_ref

    def on_sequence(self, iseq):
        self.seq_callback(iseq)
        return super(CallbackBlock, self).on_sequence(iseq)

    def on_data(self, ispan, ospan):
        self.data_callback(ispan, ospan)
        if self.data_ref is not None:
            self.data_ref["odata"] = np.array(ispan.data)
        return super(CallbackBlock, self).on_data(ispan, ospan)


def _expected_scales():
    tstart_unix = (TSTART_MJD - 40587) * 86400
    return [[tstart_unix, TSAMP], None, [FCH1, FOFF]]


def run_simple_copy(fil_file, guarantee, test_views=False,
                    gulp_nframe_inc=0):
    fname, fdata = fil_file

    def check_sequence(seq):
        tensor = seq.header["_tensor"]
        assert tensor["shape"] == [-1, 1, NCHANS]
        assert tensor["dtype"] == "u16"
        assert tensor["labels"] == ["time", "pol", "freq"]
        assert tensor["units"] == ["s", None, "MHz"]
        np.testing.assert_allclose(
            np.array(tensor["scales"][0], dtype=np.float64),
            _expected_scales()[0])
        np.testing.assert_allclose(
            np.array(tensor["scales"][2], dtype=np.float64),
            _expected_scales()[2])

    collected = []

    def check_data(ispan, ospan):
        assert ospan.nframe == ispan.nframe
        assert ispan.data.shape == (ispan.nframe, 1, NCHANS)
        collected.append((ispan.frame_offset, np.array(ispan.data)))

    gulp_nframe = 101
    with bf.Pipeline() as pipeline:
        data = bf.blocks.read_sigproc([fname], gulp_nframe)
        if test_views:
            v = bf.views
            data = v.split_axis(data, "freq", 2, "fine_freq")
            data = v.merge_axes(data, "freq", "fine_freq")
            data = v.rename_axis(data, "freq", "chan")
            data = v.rename_axis(data, "chan", "freq")
            data = v.reverse_scale(data, "freq")
            data = v.reverse_scale(data, "freq")
            data = v.reinterpret_axis(data, "freq", label="chan",
                                      scale=[0, 1], units="THz")
            data = v.reinterpret_axis(data, "chan", label="freq",
                                      scale=[FCH1, FOFF], units="MHz")
            data = v.astype(data, "i16")
            data = v.astype(data, "u16")
            data = v.add_axis(data, -1, "phony_axis", scale=(0, 1),
                              units="imaginary")
            data = v.delete_axis(data, "phony_axis")
            data = v.add_axis(data, 0, "phony_axis")
            data = v.delete_axis(data, "phony_axis")
            data = v.custom(data, lambda hdr: hdr)
        for i in range(20):
            if gulp_nframe_inc != 0:
                data = bf.blocks.copy(
                    data, guarantee=guarantee,
                    gulp_nframe=gulp_nframe + i * gulp_nframe_inc)
            else:
                data = bf.blocks.copy(data, guarantee=guarantee)
        data = bf.blocks.copy(data, guarantee=guarantee,
                              gulp_nframe=gulp_nframe)
        CallbackBlock(data, check_sequence, check_data)
        pipeline.run()

    # Guaranteed chains must deliver every frame intact.  Unguaranteed
    # reads may drop (skip) frames when a reader is overrun and may even
    # forward torn data (the reference accepts this too — its test checks
    # shapes only): assert no duplication and sane bounds.
    nframes = sum(d.shape[0] for _, d in collected)
    if guarantee:
        assert nframes == NFRAME
        for off, d in collected:
            np.testing.assert_array_equal(d, fdata[off:off + d.shape[0]])
        got = np.concatenate([d for _, d in collected], axis=0)
        np.testing.assert_array_equal(got, fdata)
    else:
        assert 0 < nframes <= NFRAME
        for off, d in collected:
            assert 0 <= off and off + d.shape[0] <= NFRAME


def test_simple_copy(fil_file):
    run_simple_copy(fil_file, guarantee=True)


def test_simple_copy_unguaranteed(fil_file):
    run_simple_copy(fil_file, guarantee=False)


def test_simple_copy_mixed_gulp_nframe(fil_file):
    run_simple_copy(fil_file, guarantee=True, gulp_nframe_inc=1)


def test_simple_copy_mixed_gulp_nframe_unguaranteed(fil_file):
    run_simple_copy(fil_file, guarantee=False, gulp_nframe_inc=1)


def test_simple_views(fil_file):
    run_simple_copy(fil_file, guarantee=True, test_views=True)


def test_simple_views_unguaranteed(fil_file):
    run_simple_copy(fil_file, guarantee=False, test_views=True)


def test_block_chainer(fil_file):
    fname, fdata = fil_file
    collected = []
    with bf.Pipeline() as pipeline:
        bc = bf.BlockChainer()
        bc.blocks.read_sigproc([fname], 101)
        bc.blocks.copy()
        bc.views.rename_axis("freq", "chan")
        bc.views.rename_axis("chan", "freq")
        bc.custom(lambda b: CallbackBlock(
            b, lambda s: None,
            lambda i, o: collected.append(np.array(i.data))))()
        pipeline.run()
    got = np.concatenate(collected, axis=0)
    np.testing.assert_array_equal(got, fdata)


def test_unguaranteed_chain_no_duplication(fil_file):
    """Regression: the C acquire must never deliver data beyond the
    requested window — a 20-deep unguaranteed chain used to REPLAY the
    stream ~2x per hop (exponential frame duplication)."""
    fname, fdata = fil_file
    collected = []
    with bf.Pipeline() as pipeline:
        d = bf.blocks.read_sigproc([fname], 101)
        for _ in range(20):
            d = bf.blocks.copy(d, guarantee=False)
        CallbackBlock(d, lambda s: None,
                      lambda i, o: collected.append(np.array(i.data)))
        pipeline.run()
    nframes = sum(x.shape[0] for x in collected)
    assert nframes <= NFRAME
