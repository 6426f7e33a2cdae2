"""Block library (hot-path subset per SURVEY.md §2: copy, transpose,
unpack, quantize, correlate, accumulate, fft, detect)."""

from bifrost_amd.blocks.accumulate import AccumulateBlock, accumulate  # noqa: F401
from bifrost_amd.blocks.binary_io import (BinaryFileReadBlock,  # noqa: F401
                                          BinaryFileWriteBlock,
                                          binary_read, binary_write)
from bifrost_amd.blocks.convert_visibilities import (  # noqa: F401
    ConvertVisibilitiesBlock, convert_visibilities)
from bifrost_amd.blocks.copy import CopyBlock, copy  # noqa: F401
from bifrost_amd.blocks.dada_file import (DadaFileReadBlock,  # noqa: F401
                                          read_dada_file)
from bifrost_amd.blocks.guppi_raw import GuppiRawSourceBlock, read_guppi_raw  # noqa: F401
from bifrost_amd.blocks.serialize import (DeserializeBlock,  # noqa: F401
                                          SerializeBlock, deserialize,
                                          serialize)
from bifrost_amd.blocks.sigproc import (SigprocSourceBlock,  # noqa: F401
                                        SigprocSinkBlock, read_sigproc,
                                        write_sigproc)
from bifrost_amd.blocks.correlate import CorrelateBlock, correlate  # noqa: F401
from bifrost_amd.blocks.detect import DetectBlock, detect  # noqa: F401
from bifrost_amd.blocks.fft import FftBlock, fft  # noqa: F401
from bifrost_amd.blocks.fftshift import FftShiftBlock, fftshift  # noqa: F401
from bifrost_amd.blocks.print_header import (PrintHeaderBlock,  # noqa: F401
                                             print_header)
from bifrost_amd.blocks.reverse import ReverseBlock, reverse  # noqa: F401
from bifrost_amd.blocks.scrunch import ScrunchBlock, scrunch  # noqa: F401
from bifrost_amd.blocks.quantize import QuantizeBlock, quantize  # noqa: F401
from bifrost_amd.blocks.reduce import ReduceBlock, reduce  # noqa: F401
from bifrost_amd.blocks.transpose import TransposeBlock, transpose  # noqa: F401
from bifrost_amd.blocks.unpack import UnpackBlock, unpack  # noqa: F401
