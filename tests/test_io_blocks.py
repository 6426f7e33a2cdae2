"""CPU tests for the I/O blocks: serialize /
deserialize, binary read/write, GUPPI RAW reader, sigproc read/write."""

import glob
import json
import os
import struct

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd import guppi_raw, sigproc
from tests.test_pipeline_cpu import CollectBlock, NumpySourceBlock


# ---------------------------------------------------------------------------
# serialize / deserialize

class TestSerialize:
    def _roundtrip(self, tmp_path, data, max_file_size=None):
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([data], gulp_nframe=4)
            bf.blocks.serialize(src, path=str(tmp_path),
                                max_file_size=max_file_size)
            pipe.run()
        basenames = glob.glob(str(tmp_path / "*.bf.json"))
        assert len(basenames) == 1
        basename = basenames[0][:-5]  # strip .json

        out = []
        with bf.Pipeline() as pipe:
            src2 = bf.blocks.deserialize([basename], gulp_nframe=4)
            CollectBlock(src2, out)
            pipe.run()
        got = np.concatenate(out, axis=0)
        np.testing.assert_array_equal(got, data)
        return basename

    def test_roundtrip(self, tmp_path):
        data = np.arange(32 * 3 * 5, dtype=np.float32).reshape(32, 3, 5)
        basename = self._roundtrip(tmp_path, data)
        with open(basename + ".json") as f:
            hdr = json.load(f)
        assert hdr["_tensor"]["shape"] == [-1, 3, 5]
        assert hdr["_tensor"]["dtype"] == "f32"

    def test_roundtrip_multifile(self, tmp_path):
        # file rolls at max_file_size: several .dat files, same data back
        data = np.arange(64 * 8, dtype=np.int16).reshape(64, 8)
        basename = self._roundtrip(tmp_path, data, max_file_size=256)
        dats = glob.glob(basename + ".*.dat")
        assert len(dats) > 1

    def test_roundtrip_ci8(self, tmp_path):
        raw = np.random.RandomState(0).randint(
            -100, 100, size=(16, 4, 2)).astype(np.int8)
        data = bf.ndarray(raw.view(bf.DataType.ci8)).reshape(16, 4)
        self._roundtrip(tmp_path, data)

    def test_serialize_unnamed_uses_time_tag(self, tmp_path):
        # An unnamed sequence serializes under its zero-padded time_tag
        # (reference test_serialize.py:108-126 / blocks/serialize.py).
        class UnnamedSource(NumpySourceBlock):
            def on_sequence(self, reader, sourcename):
                hdrs = super(UnnamedSource, self).on_sequence(reader,
                                                              sourcename)
                for h in hdrs:
                    h["name"] = ""
                    h["time_tag"] = 1234
                return hdrs

        data = np.arange(8 * 3, dtype=np.float32).reshape(8, 3)
        with bf.Pipeline() as pipe:
            src = UnnamedSource([data], gulp_nframe=4)
            bf.blocks.serialize(src, path=str(tmp_path))
            pipe.run()
        base = str(tmp_path / ("%020i.bf" % 1234))
        assert glob.glob(base + ".json"), "expected time-tag-named header"

        out = []
        with bf.Pipeline() as pipe:
            src2 = bf.blocks.deserialize([base], gulp_nframe=4)
            CollectBlock(src2, out)
            pipe.run()
        np.testing.assert_array_equal(np.concatenate(out, axis=0), data)


# ---------------------------------------------------------------------------
# binary read / write

class TestBinaryIO:
    def test_read(self, tmp_path):
        fname = str(tmp_path / "input.bin")
        data = np.random.RandomState(1).standard_normal(240) \
            .astype(np.float32)
        data.tofile(fname)
        out = []
        with bf.Pipeline() as pipe:
            src = bf.blocks.binary_read([fname], gulp_size=60,
                                        gulp_nframe=1, dtype="f32")
            CollectBlock(src, out)
            pipe.run()
        got = np.concatenate(out, axis=0).ravel()
        np.testing.assert_array_equal(got, data)

    def test_write(self, tmp_path, monkeypatch):
        data = np.arange(128, dtype=np.float32).reshape(16, 8)
        monkeypatch.chdir(tmp_path)
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([data], gulp_nframe=4)
            bf.blocks.binary_write(src, file_ext="out")
            pipe.run()
        files = glob.glob(str(tmp_path / "*.out"))
        assert len(files) == 1
        got = np.fromfile(files[0], dtype=np.float32).reshape(16, 8)
        np.testing.assert_array_equal(got, data)


# ---------------------------------------------------------------------------
# GUPPI RAW

def _write_guppi(fname, blocks, nchan, npol, nbit=8, directio=False):
    """Synthesize a GUPPI RAW file: blocks is a list of int8 arrays shaped
    [chan][time][2*pol] (complex components interleaved in the last dim)."""
    with open(fname, "wb") as f:
        for data in blocks:
            blocsize = data.size  # int8: one byte per component
            records = {
                "BACKEND": "'GUPPI   '",
                "NBITS": str(nbit),
                "OBSNCHAN": str(nchan),
                "NPOL": str(npol * 2),  # GUPPI convention: 4 => dual pol
                "OBSFREQ": "1400.0",
                "OBSBW": "100.0",
                "BLOCSIZE": str(blocsize),
                "PKTIDX": "0",
                "PKTSIZE": "8192",
                "STT_IMJD": "57000",
                "STT_SMJD": "12345",
            }
            if directio:
                records["DIRECTIO"] = "1"
            for key, val in records.items():
                rec = "%-8s= %s" % (key, val)
                f.write(rec.ljust(80).encode())
            f.write(b"END" + b" " * 77)
            if directio:
                pad = (-f.tell()) % 512
                f.write(b"\x00" * pad)
            f.write(data.astype(np.int8).tobytes())


class TestGuppiRaw:
    def test_read_header(self, tmp_path):
        fname = str(tmp_path / "test.raw")
        rng = np.random.RandomState(2)
        data = rng.randint(-100, 100, size=(4, 16, 2 * 2)).astype(np.int8)
        _write_guppi(fname, [data], nchan=4, npol=2)
        with open(fname, "rb") as f:
            hdr = guppi_raw.read_header(f)
        assert hdr["OBSNCHAN"] == 4
        assert hdr["NPOL"] == 2
        assert hdr["NTIME"] == 16
        assert hdr["BACKEND"] == "GUPPI"

    def test_read_header_directio(self, tmp_path):
        fname = str(tmp_path / "test_dio.raw")
        data = np.zeros((2, 8, 2 * 2), dtype=np.int8)
        _write_guppi(fname, [data], nchan=2, npol=2, directio=True)
        with open(fname, "rb") as f:
            hdr = guppi_raw.read_header(f)
            assert f.tell() % 512 == 0
        assert hdr["NTIME"] == 8

    def test_pipeline_read(self, tmp_path):
        fname = str(tmp_path / "pipe.raw")
        rng = np.random.RandomState(3)
        blocks = [rng.randint(-100, 100, size=(4, 16, 2 * 2))
                  .astype(np.int8) for _ in range(3)]
        _write_guppi(fname, blocks, nchan=4, npol=2)
        out = []
        with bf.Pipeline() as pipe:
            src = bf.blocks.read_guppi_raw([fname])
            CollectBlock(src, out)
            pipe.run()
        got = np.concatenate(out, axis=0)
        assert got.shape == (3, 4, 16, 2)  # [block][chan][time][pol] ci8
        raw = np.stack([b.reshape(4, 16, 2, 2) for b in blocks])
        want_re, want_im = raw[..., 0], raw[..., 1]
        np.testing.assert_array_equal(got["re"], want_re)
        np.testing.assert_array_equal(got["im"], want_im)


# ---------------------------------------------------------------------------
# sigproc

class TestSigprocModule:
    def test_header_roundtrip(self, tmp_path):
        fname = str(tmp_path / "hdr.fil")
        hdr = {"telescope_id": 6, "machine_id": 0, "data_type": 1,
               "source_name": "B0329+54", "nchans": 128, "nbits": 8,
               "nifs": 2, "tstart": 57000.5, "tsamp": 1e-4,
               "fch1": 1400.0, "foff": -0.5}
        with open(fname, "wb") as f:
            sigproc.write_header(hdr, f)
        with open(fname, "rb") as f:
            got = sigproc.read_header(f)
        for k, v in hdr.items():
            assert got[k] == v, k
        assert got["header_size"] > 0

    @pytest.mark.parametrize("nbit", [1, 2, 4])
    def test_pack_unpack_roundtrip(self, nbit):
        rng = np.random.RandomState(4)
        vals = rng.randint(0, 1 << nbit, size=64).astype(np.uint8)
        packed = sigproc.pack(vals, nbit)
        assert packed.size == 64 * nbit // 8
        unpacked = sigproc.unpack(packed, nbit)
        np.testing.assert_array_equal(unpacked, vals)

    def test_unpack_signed(self):
        # 4-bit signed: nibbles LSB-first, sign-extended
        packed = np.array([0xF1, 0x7F], dtype=np.uint8) \
            .view(np.int8)  # [1,-1], [-1,7]
        got = sigproc.unpack(packed, 4)
        np.testing.assert_array_equal(got, [1, -1, -1, 7])

    def test_file_read(self, tmp_path):
        fname = str(tmp_path / "file.fil")
        rng = np.random.RandomState(5)
        data = rng.randint(0, 255, size=(100, 2, 16)).astype(np.uint8)
        hdr = {"telescope_id": 0, "machine_id": 0, "data_type": 1,
               "nchans": 16, "nbits": 8, "nifs": 2, "tstart": 57000.0,
               "tsamp": 1e-3, "fch1": 1400.0, "foff": -1.0}
        with open(fname, "wb") as f:
            sigproc.write_header(hdr, f)
            data.tofile(f)
        sf = sigproc.SigprocFile(fname)
        assert sf.nframe() == 100
        got = sf.read(40)
        np.testing.assert_array_equal(got, data[:40])
        got = sf.read(100)  # short read at EOF
        np.testing.assert_array_equal(got, data[40:])
        sf.close()


class TestSigprocBlocks:
    def test_write_then_read(self, tmp_path):
        ntime, npol, nchan = 24, 2, 8
        rng = np.random.RandomState(6)
        data = rng.randint(0, 255, size=(ntime, npol, nchan)) \
            .astype(np.uint8)
        hdr_updates = {
            "name": "testseq",
            "source_name": "J0000+0000",
            "telescope": "GBT",
            "machine": "FAKE",
            "coord_frame": "topocentric",
        }

        class _Source(NumpySourceBlock):
            def on_sequence(self, reader, sourcename):
                hdrs = super(_Source, self).on_sequence(reader, sourcename)
                hdrs[0].update(hdr_updates)
                t = hdrs[0]["_tensor"]
                t["labels"] = ["time", "pol", "freq"]
                t["scales"] = [[1.4e9, 1e-4], None, [1400.0, -0.5]]
                t["units"] = ["s", None, "MHz"]
                return hdrs

        with bf.Pipeline() as pipe:
            src = _Source([data], gulp_nframe=8)
            bf.blocks.write_sigproc(src, path=str(tmp_path))
            pipe.run()

        fil = str(tmp_path / "testseq.fil")
        assert os.path.exists(fil)

        out = []
        with bf.Pipeline() as pipe:
            src2 = bf.blocks.read_sigproc([fil], gulp_nframe=8)
            sink = CollectBlock(src2, out)
            pipe.run()
        got = np.concatenate(out, axis=0)
        np.testing.assert_array_equal(got, data)
        shdr = sink.headers[0]
        assert shdr["_tensor"]["labels"] == ["time", "pol", "freq"]
        assert shdr["source_name"] == "J0000+0000"
        assert shdr["telescope"] == "GBT"


# ---------------------------------------------------------------------------
# DADA files

def _write_dada(fname, hdr_pairs, data):
    with open(fname, "wb") as f:
        txt = "".join("%s %s\n" % kv for kv in hdr_pairs.items())
        f.write(txt.encode().ljust(4096, b"\0"))
        data.tofile(f)


class TestDadaFile:
    def test_read_pipeline(self, tmp_path):
        nframe, nchan = 12, 16
        rng = np.random.RandomState(7)
        data = rng.standard_normal((nframe, nchan)).astype(np.float32)
        f1 = str(tmp_path / "cap_0000000000.000000.dada")
        f2 = str(tmp_path / "cap_0000000001.000000.dada")
        hdr = {"NCHAN": nchan, "NBIT": 32, "TSAMP": "1.0"}
        _write_dada(f1, hdr, data[:6])
        _write_dada(f2, hdr, data[6:])

        def header_callback(dada_hdr):
            nch = int(dada_hdr["NCHAN"])
            return {
                "name": "dada-test",
                "time_tag": 0,
                "_tensor": {
                    "dtype": "f32",
                    "shape": [-1, nch],
                    "labels": ["time", "freq"],
                    "scales": [[0, 1], [0, 1]],
                    "units": [None, None],
                },
            }

        out = []
        with bf.Pipeline() as pipe:
            src = bf.blocks.read_dada_file([f1], header_callback,
                                           gulp_nframe=1)
            CollectBlock(src, out)
            pipe.run()
        got = np.concatenate(out, axis=0)
        np.testing.assert_array_equal(got, data)  # spans both files


@pytest.mark.gpu
def test_accumulate_device_whole_window_reduce():
    """Round 2: device-space accumulate with gulp_nframe == window goes
    through ONE bfReduce per output instead of per-frame map launches
    (the C5 pipeline's configuration)."""
    import bifrost_amd as bf
    from tests.test_pipeline_cpu import NumpySourceBlock, CollectBlock

    rng = np.random.RandomState(3)
    data = rng.standard_normal((24, 5, 7)).astype(np.float32)
    out = []
    with bf.Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=8)
        dev = bf.blocks.copy(src, space="cuda")
        acc = bf.blocks.accumulate(dev, 8, gulp_nframe=8)
        host = bf.blocks.copy(acc, space="system")
        CollectBlock(host, out)
        pipe.run()
    got = np.concatenate(out, axis=0)
    gold = data.reshape(3, 8, 5, 7).sum(axis=1)
    np.testing.assert_allclose(got, gold, rtol=1e-5, atol=1e-5)
