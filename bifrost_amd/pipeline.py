"""Pipeline runtime: blocks connected by rings, one host thread per block.

Source-compatible surface with the reference python/bifrost/pipeline.py
(Pipeline / SourceBlock / SinkBlock / TransformBlock / MultiTransformBlock
and the on_sequence / on_data / define_* hook API; block scopes carry the
inheritable gulp_nframe / buffer_nframe / buffer_factor / core / gpu
attributes).  Re-implemented for this backend: same execution model — the
gulp loop acquires input spans, reserves output spans, calls on_data, then
stream-synchronizes before committing (device rings stay async-safe).
"""

import threading
import time
from contextlib import ExitStack

from bifrost_amd import affinity, device, memory
from bifrost_amd.ndarray import memset_array
from bifrost_amd.proclog import ProcLog
from bifrost_amd.ring2 import Ring

__all__ = ["Pipeline", "get_default_pipeline", "block_scope", "BlockScope",
           "Block", "SourceBlock", "SinkBlock", "TransformBlock",
           "MultiTransformBlock"]


def izip(*iterables):
    while True:
        try:
            yield [next(it) for it in iterables]
        except StopIteration:
            return


_SCOPE_STACK = []


def get_default_pipeline():
    return _default_pipeline


def get_current_block_scope():
    return _SCOPE_STACK[-1] if _SCOPE_STACK else None


def block_scope(*args, **kwargs):
    return BlockScope(*args, **kwargs)


class BlockScope(object):
    """Hierarchical settings scope: attribute lookups fall through to the
    enclosing scope (gulp_nframe, buffer_nframe, buffer_factor, core, gpu,
    share_temp_storage, fuse)."""

    _INHERITED = ("gulp_nframe", "buffer_nframe", "buffer_factor", "core",
                  "gpu", "share_temp_storage", "fuse")

    def __init__(self, name=None, gulp_nframe=None, buffer_nframe=None,
                 buffer_factor=None, core=None, gpu=None, fuse=False,
                 share_temp_storage=False):
        self._parent_scope = get_current_block_scope()
        self._scope_name = name
        self.gulp_nframe = gulp_nframe
        self.buffer_nframe = buffer_nframe
        self.buffer_factor = buffer_factor
        self.core = core
        self.gpu = gpu
        self.fuse = fuse
        self.share_temp_storage = share_temp_storage

    def __enter__(self):
        _SCOPE_STACK.append(self)
        return self

    def __exit__(self, t, v, tb):
        _SCOPE_STACK.pop()

    def __getattribute__(self, name):
        val = object.__getattribute__(self, name)
        if val is None and name in BlockScope._INHERITED:
            parent = object.__getattribute__(self, "_parent_scope")
            if parent is not None:
                return getattr(parent, name)
        return val

    def is_fused_with(self, other):
        return False  # block fusion not implemented on this backend

    def cache_scope_hierarchy(self):
        pass


class PipelineInitError(Exception):
    pass


class Pipeline(BlockScope):
    def __init__(self, name=None, **kwargs):
        Pipeline.instance_count = getattr(Pipeline, "instance_count", 0) + 1
        self.name = name or ("pipeline_%d" % Pipeline.instance_count)
        BlockScope.__init__(self, **kwargs)
        self._parent_scope = None
        self.blocks = []
        self._initialized = []
        self._init_lock = threading.Lock()
        self.all_blocks_finished_initializing_event = threading.Event()
        self._block_error = None

    def as_default(self):
        global _default_pipeline
        _default_pipeline = self
        return self

    def _block_initialized(self, block, ok):
        with self._init_lock:
            self._initialized.append((block, ok))
            if not ok:
                self._block_error = block
                self.all_blocks_finished_initializing_event.set()
            elif len(self._initialized) == len(self.blocks):
                self.all_blocks_finished_initializing_event.set()

    def run(self):
        threads = [threading.Thread(target=b.run, name=b.name, daemon=True)
                   for b in self.blocks]
        for t in threads:
            t.start()
        try:
            for t in threads:
                while t.is_alive():
                    t.join(timeout=0.1)
        except KeyboardInterrupt:
            self.shutdown()
            for t in threads:
                t.join(timeout=5)
            raise
        for b in self.blocks:
            if b._error is not None:
                raise b._error

    def shutdown(self):
        for b in self.blocks:
            b.shutdown()

    def dot_graph(self):
        """DOT source for the block/ring graph (reference
        pipeline.py:163-201 dot_graph; emitted as plain DOT text rather
        than a graphviz.Digraph so no external package is needed —
        render with `dot -Tsvg`)."""
        space_colors = {"system": "orange", "cuda": "limegreen",
                        "cuda_host": "deepskyblue"}
        lines = ["digraph \"cluster_%s\" {" % self.name]
        rings = {}
        for block in self.blocks:
            label = block.name.split("/", 1)[-1]
            fill = ("lightsteelblue"
                    if block.__class__.__name__ == "CopyBlock" else "white")
            lines.append('  "%s" [label="%s", shape=box, style=filled, '
                         'fillcolor=%s];' % (block.name, label, fill))
            for oring in block.orings:
                rings[oring.name] = oring
                lines.append('  "%s" -> "%s";' % (block.name, oring.name))
            for iring in block.irings:
                rings[iring.name] = iring
                lines.append('  "%s" -> "%s";' % (iring.name, block.name))
        for name, ring in sorted(rings.items()):
            color = space_colors.get(ring.space, "white")
            lines.append('  "%s" [shape=ellipse, style=filled, '
                         'fillcolor=%s];' % (name, color))
        lines.append("}")
        return "\n".join(lines)

    def __enter__(self):
        self._prev_default = get_default_pipeline()
        self.as_default()
        BlockScope.__enter__(self)
        return self

    def __exit__(self, t, v, tb):
        BlockScope.__exit__(self, t, v, tb)
        global _default_pipeline
        _default_pipeline = self._prev_default


_default_pipeline = Pipeline(name="default_pipeline")


def get_ring(block_or_ring):
    if isinstance(block_or_ring, Ring):
        return block_or_ring
    if hasattr(block_or_ring, "orings"):
        if len(block_or_ring.orings) != 1:
            raise ValueError("Block must have exactly one output ring here")
        return block_or_ring.orings[0]
    raise TypeError("Expected a Block or Ring, got %r" % (block_or_ring,))


def block_view(block, header_transform):
    from copy import copy as _copy
    new_block = _copy(block)
    new_block.orings = [ring.view() for ring in block.orings]
    for oring in new_block.orings:
        old = oring.header_transform
        if old is not None:
            ht = header_transform
            oring.header_transform = lambda hdr, f=ht, g=old: f(g(hdr))
        else:
            oring.header_transform = header_transform
    return new_block


class Block(BlockScope):
    instance_counts = {}

    def __init__(self, irings, name=None, type_=None, **kwargs):
        self.type = type_ or self.__class__.__name__
        count = Block.instance_counts.get(self.type, 0)
        self.name = name or ("%s_%d" % (self.type, count))
        Block.instance_counts[self.type] = count + 1
        BlockScope.__init__(self, **kwargs)
        self.pipeline = get_default_pipeline()
        self.pipeline.blocks.append(self)
        self.irings = [get_ring(r) for r in irings]
        valid = self._define_valid_input_spaces()
        for i, (iring, spaces) in enumerate(zip(self.irings, valid)):
            if spaces != "any" and not memory.space_accessible(iring.space,
                                                               spaces):
                raise ValueError(
                    "Block %s input %d space %r not accessible from %r"
                    % (self.name, i, iring.space, spaces))
        self.orings = []
        self.shutdown_event = threading.Event()
        self._error = None
        self.bind_proclog = ProcLog(self.name + "/bind")

    def shutdown(self):
        self.shutdown_event.set()

    def create_ring(self, *args, **kwargs):
        return Ring(*args, owner=self, **kwargs)

    def run(self):
        try:
            if self.core is not None:
                affinity.set_core(self.core if isinstance(self.core, int)
                                  else self.core[0])
            self.bind_proclog.update({"ncore": 1,
                                      "core0": affinity.get_core()})
            if self.gpu is not None:
                device.set_device(self.gpu)
            with ExitStack() as oring_stack:
                active_orings = self.begin_writing(oring_stack, self.orings)
                self.main(active_orings)
        except Exception as e:
            self._error = e
            self.pipeline._block_initialized(self, False)
            self.pipeline.shutdown()
            # unblock peers stuck on our rings
            for ring in self.orings:
                try:
                    ring.end_writing()
                except Exception:
                    pass

    def num_outputs(self):
        return len(self.orings)

    def begin_writing(self, exit_stack, orings):
        return [exit_stack.enter_context(oring.begin_writing())
                for oring in orings]

    def begin_sequences(self, exit_stack, orings, oheaders, igulp_nframes,
                        istride_nframes):
        ostride_nframes = self._define_output_nframes(istride_nframes)
        for ohdr, ostride in zip(oheaders, ostride_nframes):
            ohdr["gulp_nframe"] = ostride
        ogulp_nframes = self._define_output_nframes(igulp_nframes)
        oseqs = [exit_stack.enter_context(
                     oring.begin_sequence(ohdr, ogulp_nframe, ogulp_nframe))
                 for oring, ohdr, ogulp_nframe
                 in zip(orings, oheaders, ogulp_nframes)]
        self.pipeline._block_initialized(self, True)
        self.pipeline.all_blocks_finished_initializing_event.wait()
        ogulp_overlaps = [g - s for g, s in zip(ogulp_nframes,
                                                ostride_nframes)]
        return oseqs, ogulp_overlaps

    def reserve_spans(self, exit_stack, oseqs, igulp_nframes=[]):
        ogulp_nframes = self._define_output_nframes(igulp_nframes)
        return [exit_stack.enter_context(oseq.reserve(nf))
                for oseq, nf in zip(oseqs, ogulp_nframes)]

    def commit_spans(self, ospans, ostrides_actual, ogulp_overlaps):
        if ostrides_actual is None:
            ostrides_actual = [None] * len(ospans)
        for ospan, ostride, overlap in zip(ospans, ostrides_actual,
                                           ogulp_overlaps):
            if ostride is None:
                ostride = max(ospan.nframe - overlap, 0)
            ospan.commit(ostride)

    def _define_output_nframes(self, input_nframes):
        return self.define_output_nframes(input_nframes)

    def define_output_nframes(self, input_nframes):
        raise NotImplementedError

    def _define_valid_input_spaces(self):
        return self.define_valid_input_spaces()

    def define_valid_input_spaces(self):
        return ["any"] * len(self.irings)


class SourceBlock(Block):
    def __init__(self, sourcenames, gulp_nframe, space=None, *args, **kwargs):
        super(SourceBlock, self).__init__([], *args, gulp_nframe=gulp_nframe,
                                          **kwargs)
        self.sourcenames = sourcenames
        if space is None:
            space = "system"
        self.orings = [self.create_ring(space=space)]
        self._seq_count = 0
        self.perf_proclog = ProcLog(self.name + "/perf")

    def main(self, orings):
        for sourcename in self.sourcenames:
            if self.shutdown_event.is_set():
                break
            with self.create_reader(sourcename) as reader:
                oheaders = self.on_sequence(reader, sourcename)
                for ohdr in oheaders:
                    ohdr.setdefault("time_tag", self._seq_count)
                    ohdr.setdefault("name",
                                    "unnamed-sequence-%d" % self._seq_count)
                self._seq_count += 1
                with ExitStack() as oseq_stack:
                    oseqs, ogulp_overlaps = self.begin_sequences(
                        oseq_stack, orings, oheaders, [], [])
                    while not self.shutdown_event.is_set():
                        t0 = time.time()
                        with ExitStack() as ospan_stack:
                            ospans = self.reserve_spans(ospan_stack, oseqs)
                            ostrides = self.on_data(reader, ospans)
                            device.stream_synchronize()
                            self.commit_spans(ospans, ostrides,
                                              ogulp_overlaps)
                            if any(o == 0 for o in ostrides):
                                break
                        self.perf_proclog.update(
                            {"process_time": time.time() - t0})

    def define_output_nframes(self, _):
        return [self.gulp_nframe] * self.num_outputs()

    def define_valid_input_spaces(self):
        return []

    def create_reader(self, sourcename):
        raise NotImplementedError

    def on_sequence(self, reader, sourcename):
        raise NotImplementedError

    def on_data(self, reader, ospans):
        raise NotImplementedError


class MultiTransformBlock(Block):
    def __init__(self, irings_, guarantee=True, *args, **kwargs):
        super(MultiTransformBlock, self).__init__(irings_, *args, **kwargs)
        self.guarantee = guarantee
        self.orings = [self.create_ring(space=iring.space)
                       for iring in self.irings]
        self._seq_count = 0
        self.perf_proclog = ProcLog(self.name + "/perf")

    def main(self, orings):
        for iseqs in izip(*[iring.read(guarantee=self.guarantee)
                            for iring in self.irings]):
            if self.shutdown_event.is_set():
                break
            oheaders = self._on_sequence(iseqs)
            for ohdr in oheaders:
                ohdr.setdefault("time_tag", self._seq_count)
            self._seq_count += 1

            igulp_nframes = [self.gulp_nframe or iseq.header["gulp_nframe"]
                             for iseq in iseqs]
            igulp_overlaps = self._define_input_overlap_nframe(iseqs)
            istride_nframes = igulp_nframes[:]
            igulp_nframes = [g + o for g, o in zip(igulp_nframes,
                                                   igulp_overlaps)]
            for iseq, igulp_nframe in zip(iseqs, igulp_nframes):
                iseq.resize(gulp_nframe=igulp_nframe,
                            buf_nframe=self.buffer_nframe,
                            buffer_factor=self.buffer_factor)

            with ExitStack() as oseq_stack:
                oseqs, ogulp_overlaps = self.begin_sequences(
                    oseq_stack, orings, oheaders, igulp_nframes,
                    istride_nframes)
                if self.shutdown_event.is_set():
                    break
                prev_time = time.time()
                for ispans in izip(*[iseq.read(g, s, 0)
                                     for iseq, g, s
                                     in zip(iseqs, igulp_nframes,
                                            istride_nframes)]):
                    if self.shutdown_event.is_set():
                        return
                    if any(ispan.nframe_skipped for ispan in ispans):
                        with ExitStack() as ospan_stack:
                            nskips = [isp.nframe_skipped for isp in ispans]
                            ospans = self.reserve_spans(ospan_stack, oseqs,
                                                        nskips)
                            islices = [slice(0, n) for n in nskips]
                            ostrides = self._on_skip(islices, ospans)
                            device.stream_synchronize()
                            self.commit_spans(ospans, ostrides,
                                              ogulp_overlaps)
                    if all(ispan.nframe == 0 for ispan in ispans):
                        continue
                    acquire_time = time.time() - prev_time
                    with ExitStack() as ospan_stack:
                        cur_igulps = [ispan.nframe for ispan in ispans]
                        ospans = self.reserve_spans(ospan_stack, oseqs,
                                                    cur_igulps)
                        reserve_time = time.time() - prev_time - acquire_time
                        ostrides = self._on_data(ispans, ospans)
                        device.stream_synchronize()
                        self.commit_spans(ospans, ostrides, ogulp_overlaps)
                    now = time.time()
                    self.perf_proclog.update({
                        "acquire_time": acquire_time,
                        "reserve_time": reserve_time,
                        "process_time": now - prev_time})
                    prev_time = now
            self._on_sequence_end(iseqs)

    def _on_sequence(self, iseqs):
        return self.on_sequence(iseqs)

    def _on_sequence_end(self, iseqs):
        return self.on_sequence_end(iseqs)

    def _on_data(self, ispans, ospans):
        return self.on_data(ispans, ospans)

    def _on_skip(self, islices, ospans):
        return self.on_skip(islices, ospans)

    def _define_input_overlap_nframe(self, iseqs):
        return self.define_input_overlap_nframe(iseqs)

    def define_input_overlap_nframe(self, iseqs):
        return [0] * len(self.irings)

    def define_output_nframes(self, input_nframes):
        return input_nframes

    def on_sequence(self, iseqs):
        raise NotImplementedError

    def on_sequence_end(self, iseqs):
        pass

    def on_data(self, ispans, ospans):
        raise NotImplementedError

    def on_skip(self, islices, ospans):
        for ospan in ospans:
            memset_array(ospan.data, 0)
        return None


class TransformBlock(MultiTransformBlock):
    def __init__(self, iring, *args, **kwargs):
        super(TransformBlock, self).__init__([iring], *args, **kwargs)
        self.iring = self.irings[0]

    def _define_valid_input_spaces(self):
        return [self.define_valid_input_spaces()]

    def define_valid_input_spaces(self):
        return "any"

    def _define_input_overlap_nframe(self, iseqs):
        return [self.define_input_overlap_nframe(iseqs[0])]

    def define_input_overlap_nframe(self, iseq):
        return 0

    def _define_output_nframes(self, input_nframes):
        return [self.define_output_nframes(input_nframes[0])]

    def define_output_nframes(self, input_nframe):
        return input_nframe

    def _on_sequence(self, iseqs):
        return [self.on_sequence(iseqs[0])]

    def on_sequence(self, iseq):
        raise NotImplementedError

    def _on_sequence_end(self, iseqs):
        return [self.on_sequence_end(iseqs[0])]

    def on_sequence_end(self, iseq):
        pass

    def _on_data(self, ispans, ospans):
        return [self.on_data(ispans[0], ospans[0])]

    def on_data(self, ispan, ospan):
        raise NotImplementedError

    def _on_skip(self, islices, ospans):
        return [self.on_skip(islices[0], ospans[0])]

    def on_skip(self, islice, ospan):
        memset_array(ospan.data, 0)
        return None


class SinkBlock(MultiTransformBlock):
    def __init__(self, iring, *args, **kwargs):
        super(SinkBlock, self).__init__([iring], *args, **kwargs)
        self.orings = []
        self.iring = self.irings[0]

    def _define_valid_input_spaces(self):
        return [self.define_valid_input_spaces()]

    def define_valid_input_spaces(self):
        return "any"

    def _define_input_overlap_nframe(self, iseqs):
        return [self.define_input_overlap_nframe(iseqs[0])]

    def define_input_overlap_nframe(self, iseq):
        return 0

    def _define_output_nframes(self, input_nframes):
        return []

    def _on_sequence(self, iseqs):
        self.on_sequence(iseqs[0])
        return []

    def on_sequence(self, iseq):
        raise NotImplementedError

    def _on_sequence_end(self, iseqs):
        return [self.on_sequence_end(iseqs[0])]

    def on_sequence_end(self, iseq):
        pass

    def _on_data(self, ispans, ospans):
        self.on_data(ispans[0])
        return []

    def on_data(self, ispan):
        raise NotImplementedError

    def _on_skip(self, islices, ospans):
        return []
