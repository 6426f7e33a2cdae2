"""CPU tests for the classic byte-oriented `bifrost.ring` API
(reference python/bifrost/ring.py surface over our C ring ABI)."""

import threading

import pytest
import numpy as np

from bifrost_amd.ring import Ring


def test_classic_ring_properties():
    ring = Ring(space="system", name="classic-props")
    assert ring.name == "classic-props"
    assert ring.space == "system"
    assert not ring.writing_ended()


def test_classic_write_read_roundtrip():
    ring = Ring(space="system", name="classic-rt")
    gulp = 64
    ring.resize(gulp, 4 * gulp)
    chunks = [np.full(gulp, i, dtype=np.uint8) for i in range(3)]

    def writer():
        with ring.begin_writing() as ow:
            with ow.begin_sequence(name="seq0", time_tag=42,
                                   header="hello") as oseq:
                for c in chunks:
                    with oseq.reserve(gulp) as wspan:
                        wspan.data_view()[0, :] = c

    t = threading.Thread(target=writer)
    t.start()

    with ring.open_earliest_sequence(guarantee=True) as iseq:
        assert iseq.name == "seq0"
        assert iseq.time_tag == 42
        assert iseq.header.tobytes() == b"hello"
        assert iseq.nringlet == 1
        got = [bytes(span.data.tobytes()) for span in iseq.read(gulp)]
    t.join()
    assert got == [c.tobytes() for c in chunks]
    assert ring.writing_ended()


def test_classic_partial_commit_and_typed_view():
    ring = Ring(space="system", name="classic-commit")
    gulp = 64
    ring.resize(gulp, 4 * gulp)

    def writer():
        with ring.begin_writing() as ow:
            with ow.begin_sequence(name="s") as oseq:
                span = oseq.reserve(gulp)
                view = span.data_view(np.float32)
                assert view.shape == (1, gulp // 4)
                view[0, :8] = np.arange(8, dtype=np.float32)
                span.commit(32)   # commit only the 8 floats
                span.close()

    t = threading.Thread(target=writer)
    t.start()
    with ring.open_earliest_sequence(guarantee=True) as iseq:
        spans = list(iseq.read(32))
        assert len(spans) == 1
    t.join()


def test_classic_sequence_generator_across_sequences():
    ring = Ring(space="system", name="classic-seqs")
    gulp = 16
    ring.resize(gulp, 8 * gulp)

    def writer():
        with ring.begin_writing() as ow:
            for s in range(3):
                with ow.begin_sequence(name="seq%d" % s) as oseq:
                    with oseq.reserve(gulp) as wspan:
                        wspan.data_view()[0, :] = s

    t = threading.Thread(target=writer)
    t.start()
    names, payloads = [], []
    for seq in ring.read(whence="earliest", guarantee=True):
        names.append(seq.name)
        for span in seq.read(gulp):
            payloads.append(int(span.data[0, 0]))
    t.join()
    assert names == ["seq0", "seq1", "seq2"]
    assert payloads == [0, 1, 2]


def test_classic_late_resize_preserves_data():
    """Reference test_resizing.py behaviour: growing the ring AFTER a read
    sequence is open (the ModResizeAsciiBlock pattern — resize between
    opening the sequence and acquiring spans) must preserve live data.
    The resize waits for span quiescence (no open spans), so it must be
    called while no span is held."""
    ring = Ring(space="system", name="classic-lateresize")
    gulp = 32
    ring.resize(gulp, 2 * gulp)

    def writer():
        with ring.begin_writing() as ow:
            with ow.begin_sequence(name="s") as oseq:
                for i in range(2):
                    with oseq.reserve(gulp) as wspan:
                        wspan.data_view()[0, :] = i

    t = threading.Thread(target=writer)
    t.start()
    with ring.open_earliest_sequence(guarantee=True) as iseq:
        # late resize: after the sequence is open, before any span is held
        ring.resize(4 * gulp, 16 * gulp)
        got = [s.data.tobytes() for s in iseq.read(gulp)]
    t.join()
    assert got == [bytes([0] * gulp), bytes([1] * gulp)]


def test_classic_resize_preserves_committed_data():
    """Growing the ring after data is fully written re-places the live
    bytes at their new positions (no torn reads)."""
    ring = Ring(space="system", name="classic-resize-keep")
    gulp = 48
    ring.resize(gulp, 4 * gulp)

    def writer():
        with ring.begin_writing() as ow:
            with ow.begin_sequence(name="s") as oseq:
                for i in range(4):
                    with oseq.reserve(gulp) as wspan:
                        wspan.data_view()[0, :] = 10 + i

    t = threading.Thread(target=writer)
    t.start()
    t.join()
    ring.resize(2 * gulp, 16 * gulp)   # grow: ghost AND capacity
    with ring.open_earliest_sequence(guarantee=True) as iseq:
        got = [s.data.tobytes() for s in iseq.read(gulp)]
    assert got == [bytes([10 + i] * gulp) for i in range(4)]


def test_classic_gulp_larger_than_ring_grows_capacity():
    """Reference TestLargeGulpSize: requesting a contiguous span larger
    than the ring's total size must grow the ring, not fault."""
    ring = Ring(space="system", name="classic-largegulp")
    ring.resize(12)            # tiny: total 48 bytes
    ring.resize(1024, 1024)    # "gulp" bigger than the old ring
    payload = np.resize(np.arange(256, dtype=np.uint8), 1024)

    def writer():
        with ring.begin_writing() as ow:
            with ow.begin_sequence(name="big") as oseq:
                with oseq.reserve(1024) as wspan:
                    wspan.data_view()[0, :] = payload

    t = threading.Thread(target=writer)
    t.start()
    with ring.open_earliest_sequence(guarantee=True) as iseq:
        got = [s.data.tobytes() for s in iseq.read(1024)]
    t.join()
    assert got == [payload.tobytes()]


def test_multi_ringlet_roundtrip():
    """Round 2: nringlet > 1 — N parallel lanes sharing offsets; span
    data is ringlet-strided (stride = capacity + ghost).  Writes fill
    each lane with a distinct pattern; reads must see them laned, also
    across the wrap (per-ringlet ghost fix-up)."""
    import threading

    import bifrost_amd.ring as bring

    NRINGLET, GULP, NGULP = 4, 32, 12
    ring = bring.Ring(name="mrl")
    ring.resize(GULP, GULP * 4, nringlet=NRINGLET)
    # a reader's guarantee protects data only from its OPEN onwards
    # (reference semantics): gate the writer on the reader being open so
    # the 4-gulp window can't lap it under suite load
    reader_open = threading.Event()

    def writer():
        with ring.begin_writing() as ow:
            with ow.begin_sequence(name="mr", time_tag=1,
                                   nringlet=NRINGLET) as oseq:
                assert reader_open.wait(timeout=30)
                for g in range(NGULP):
                    with oseq.reserve(GULP) as wspan:
                        v = wspan.data_view(np.uint8)
                        assert v.shape == (NRINGLET, GULP)
                        for r in range(NRINGLET):
                            v[r, :] = (np.arange(GULP) + 64 * r + g) % 256

    t = threading.Thread(target=writer)
    t.start()
    seen = 0
    with ring.open_earliest_sequence(guarantee=True) as iseq:
        reader_open.set()
        for g, span in enumerate(iseq.read(GULP)):
            v = span.data_view(np.uint8)
            assert v.shape == (NRINGLET, GULP)
            for r in range(NRINGLET):
                np.testing.assert_array_equal(
                    v[r], (np.arange(GULP) + 64 * r + g) % 256)
            seen += 1
            if seen == NGULP:
                break
    t.join()
    assert seen == NGULP


def test_multi_ringlet_resize_rules():
    import bifrost_amd.ring as bring
    from bifrost_amd.libbifrost import _bf

    ring = bring.Ring(name="mrl2")
    ring.resize(16, 64, nringlet=3)
    # changing lanes while empty is fine
    ring.resize(16, 64, nringlet=2)
    with ring.begin_writing() as ow:
        with ow.begin_sequence(name="s", nringlet=2) as oseq:
            with oseq.reserve(16) as wspan:
                wspan.data_view(np.uint8)[...] = 7
    # ring now holds data: changing nringlet must be refused
    import pytest as _pytest
    with _pytest.raises(RuntimeError):
        ring.resize(16, 64, nringlet=5)
    # sequence lanes may not exceed the ring's
    with _pytest.raises(RuntimeError):
        with ring.begin_writing() as ow:
            ow.begin_sequence(name="s2", nringlet=8)


@pytest.mark.gpu
def test_multi_ringlet_device_ring():
    """Multi-ringlet lanes on a cuda-space ring: per-ringlet ghost
    fix-up runs device-side copies (with the round-2 stream fence)."""
    import bifrost_amd.ring as bring
    from bifrost_amd import memory as bf_memory
    import bifrost_amd as bf

    NR, GULP = 3, 64
    ring = bring.Ring(name="mrl_dev", space="cuda")
    ring.resize(GULP, GULP * 3, nringlet=NR)
    pats = [(np.arange(GULP, dtype=np.uint8) + 17 * r) % 251
            for r in range(NR)]
    # persistent host buffers: bfMemcpy H2D is async on the thread
    # stream, so the source must outlive the call until the sync below
    hosts = [[bf.ndarray((pats[r] + g) % 251) for r in range(NR)]
             for g in range(5)]
    from bifrost_amd import device as bf_device
    with ring.begin_writing() as ow:
        with ow.begin_sequence(name="d", nringlet=NR) as oseq:
            # 5 gulps on a 3-gulp window forces wrap + ghost flush
            for g in range(5):
                with oseq.reserve(GULP) as wspan:
                    v = wspan.data_view(np.uint8)
                    assert v.shape == (NR, GULP)
                    for r in range(NR):
                        bf_memory.memcpy(v[r], hosts[g][r])
                    bf_device.stream_synchronize()
    # v[r] of the LAST gulp should still hold its pattern.  (An
    # abandoned read generator is ALSO safe — bfRingSequenceClose defers
    # the reader free to the last span release, tested by leaving
    # `spans` un-closed here.)
    with ring.open_latest_sequence(guarantee=True) as iseq:
        spans = iseq.read(GULP, begin=4 * GULP)
        span = next(iter(spans))
        v = span.data_view(np.uint8)
        for r in range(NR):
            got = np.asarray(bf.ndarray(v[r]).copy("system"))
            np.testing.assert_array_equal(got, (pats[r] + 4) % 251)
