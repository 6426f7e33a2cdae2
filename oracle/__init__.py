"""CPU oracle for the bifrost linalg hot path — TEST INFRASTRUCTURE ONLY.

This package is a plain-numpy restatement of the reference algorithms of
ledatelescope/bifrost's linalg correlation/beamforming path and its
unpack/quantize/transpose feeders.  It exists solely so that `tests/`,
`__graft_entry__.smoke()` and `bench.py`'s `cpu_baseline` leg can check the
HIP product path against an independent CPU implementation.  Nothing in the
product path (`bifrost_amd/`) may import, call, link or execute anything in
here; the product path must fail loudly when the HIP extension is missing.

Pinning: the bit-twiddle restatements in `bitops.py` are pinned against the
reference's own known-answer test vectors (test/test_unpack.py:33-95,
test/test_quantize.py:33-50 of the reference tree), committed as fixtures
under tests/golden/; they can additionally be cross-validated against the
reference's *own* CPU implementations compiled unmodified from
/root/reference/src/{unpack,quantize}.cpp by `oracle/ref_build/Makefile`
(output in oracle/_ref/, see `oracle.refcpu`).  The linalg gold restates the
numpy-gold recipe embedded in the reference's tests
(test/test_linalg.py:41-71,136-151,168-185): that recipe IS the reference's
only CPU statement of the correlator/beamformer math (the reference has no
C++ CPU linalg path).
"""

from .bitops import unpack as unpack            # noqa: F401
from .bitops import quantize as quantize        # noqa: F401
from .linalg import matmul_aa, matmul_ab, correlator_gold, beamformer_gold  # noqa: F401
from .linalg import transpose as transpose      # noqa: F401
