"""bifrost_amd: MI355X-native libbifrost DSP backend.

Drop-in for the bifrost Python API on the linalg hot path: same module
layout and call surface as `bifrost` (use `import bifrost_amd as bf`, or
alias `sys.modules['bifrost'] = bifrost_amd` for unmodified pipelines —
see INTEGRATION.md).
"""

from bifrost_amd import affinity, core, device, memory  # noqa: F401
from bifrost_amd.version import __version__  # noqa: F401
from bifrost_amd import pipeline  # noqa: F401
from bifrost_amd.pipeline import Pipeline, block_scope, get_default_pipeline  # noqa: F401
from bifrost_amd import ring  # noqa: F401  (classic byte-span API)
from bifrost_amd.ring2 import Ring  # noqa: F401
from bifrost_amd import blocks  # noqa: F401
from bifrost_amd import views  # noqa: F401
from bifrost_amd.block_chainer import BlockChainer  # noqa: F401
from bifrost_amd.DataType import DataType  # noqa: F401
from bifrost_amd.ndarray import (asarray, copy_array, empty, empty_like,  # noqa: F401
                                 memset_array, ndarray, zeros, zeros_like)
from bifrost_amd.quantize import quantize  # noqa: F401
from bifrost_amd.transpose import transpose  # noqa: F401
from bifrost_amd.unpack import unpack  # noqa: F401
from bifrost_amd.map import clear_map_cache, list_map_cache, map  # noqa: F401
from bifrost_amd.fft import Fft  # noqa: F401
from bifrost_amd.reduce import reduce  # noqa: F401

# Make `import bifrost_amd as bf; bf.DataType.ci8` work like the reference's
# module attribute access (bifrost.DataType is a module there).
from bifrost_amd import DataType as _DataType_module_names  # noqa: F401
from bifrost_amd.DataType import ci4, ci8, ci16, ci32, ci64, cf16  # noqa: F401
DataType.ci4 = ci4
DataType.ci8 = ci8
DataType.ci16 = ci16
DataType.ci32 = ci32
DataType.ci64 = ci64
DataType.cf16 = cf16
