"""Compatibility shim: the reference exposes build-time flags through a
ctypesgen-generated module (python/Makefile.in:23-30); user code imports
e.g. BF_CUDA_ENABLED from it.  This backend exports the same names from the
hand-written FFI layer."""

from bifrost_amd.libbifrost import _bf, BFarray as struct_BFarray_  # noqa: F401

BF_CUDA_ENABLED = _bf.BF_CUDA_ENABLED
bfTestSuite = _bf.bfTestSuite  # C self-test entry (reference test_library.py)
BF_FLOAT128_ENABLED = _bf.BF_FLOAT128_ENABLED
BF_DEBUG_ENABLED = _bf.BF_DEBUG_ENABLED
BF_TRACE_ENABLED = _bf.BF_TRACE_ENABLED

# Re-export every BF_* constant for parity with the generated module.
for _name in dir(_bf):
    if _name.startswith("BF_"):
        globals()[_name] = getattr(_bf, _name)
del _name
