"""Numpy restatement of the bfLinAlgMatMul semantics — TEST ORACLE ONLY.

The reference has no C++ CPU linalg path; its ground truth is
the numpy-gold recipe inside its own tests, restated here:

  - matmul_aa:  c = alpha * (A . A^H) + beta * c on the LOWER triangle only,
                upper triangle of c left untouched (row-major; the reference
                computes this with cublas herk uplo=UPPER on col-major data,
                src/linalg.cu:190-240; gold recipe test_linalg.py:52-71).
  - matmul_ab:  c = alpha * (A . B) + beta * c (test_linalg.py:90-135).
  - correlator_gold / beamformer_gold: the exact input synthesis + gold
    computation of test_linalg.py:168-185 and :136-151.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may use
this module.
"""

import numpy as np

__all__ = ["H", "matmul_aa", "matmul_ab", "transpose",
           "correlator_gold", "beamformer_gold", "make_ci8_voltages"]


def H(c):
    """Hermitian transpose of the last two dims (test_linalg.py:44-45)."""
    return np.swapaxes(c, -1, -2).conj()


def matmul_aa(alpha, a, beta, c):
    """c = alpha * (a @ H(a)) + beta * c, lower triangle written only.

    a: complex array [..., n, k]; c: complex array [..., n, n] (modified copy
    returned).  Matches bfMatMul_aa (src/linalg.cu:242-357): the diagonal and
    strictly-lower elements are updated, the strictly-upper part of c is
    returned unchanged.
    """
    a = np.asarray(a)
    c = np.array(c, copy=True)
    full = alpha * np.matmul(a, H(a))
    n = c.shape[-1]
    il = np.tril_indices(n)
    c[..., il[0], il[1]] = full[..., il[0], il[1]] + beta * c[..., il[0], il[1]]
    return c


def matmul_ab(alpha, a, b, beta, c):
    """c = alpha * (a @ b) + beta * c (bfMatMul_ab, src/linalg.cu:723-862)."""
    return alpha * np.matmul(np.asarray(a), np.asarray(b)) + beta * np.asarray(c)


def transpose(a, axes):
    """bfTranspose oracle: exact permutation copy (test_transpose.py:46-54)."""
    return np.ascontiguousarray(np.transpose(np.asarray(a), axes))


# ---------------------------------------------------------------------------
# Synthesis + gold recipes, exactly as the reference tests write them
# ---------------------------------------------------------------------------

def make_ci8_voltages(ntime, nchan, nstand, seed=1234):
    """Synthetic 8-bit voltages, the reference's recipe.

    test_linalg.py:51,171: np.random.seed(seed);
    x8 = ((random(size=(ntime,nchan,nstand*2,2))*2-1)*127).astype(int8).
    Returns (x8 int8 [t,c,s,2], x complex64 [t,c,s]).  Values lie in
    [-127,127] (the kernel contract excludes -128, test_linalg.py:55).
    """
    np.random.seed(seed)
    shape = (ntime, nchan, nstand * 2, 2)
    x8 = ((np.random.random(size=shape) * 2 - 1) * 127).astype(np.int8)
    x = x8.astype(np.float32).view(np.complex64).reshape(shape[:-1])
    return x8, x


def correlator_gold(ntime, nstand, nchan, misalign=0, seed=1234):
    """Inputs + gold of run_test_matmul_aa_correlator_kernel
    (test_linalg.py:168-185).

    Returns (x8 int8 [t,c,s,2], b_gold complex64 [c, n, n]) where
    n = nstand*2 - misalign; b_gold = H(x) @ x per channel with the strictly
    upper triangle zeroed (x viewed [c,t,s][..., misalign:]).
    """
    x8, x = make_ci8_voltages(ntime, nchan, nstand, seed)
    xv = x.transpose(1, 0, 2)[..., misalign:]
    b_gold = np.matmul(H(xv), xv)
    triu = np.triu_indices(xv.shape[-1], 1)
    b_gold[..., triu[0], triu[1]] = 0
    return x8, b_gold


def beamformer_gold(ntime, nbeam, nstand, nchan, seed=1234):
    """Inputs + gold of run_test_matmul_ab_beamformer_kernel
    (test_linalg.py:136-151).

    Returns (x8 int8 [t,c,s,2], w complex64 [beam,c,s],
    b_gold complex64 [c, beam, t]) with
    b_gold = w.transpose(1,0,2) @ x.transpose(1,2,0).
    """
    np.random.seed(seed)
    x_shape = (ntime, nchan, nstand * 2)
    w_shape = (nbeam, nchan, nstand * 2)
    x8 = ((np.random.random(size=x_shape + (2,)) * 2 - 1) * 127).astype(np.int8)
    x = x8.astype(np.float32).view(np.complex64).reshape(x_shape)
    w = ((np.random.random(size=w_shape + (2,)) * 2 - 1) * 127).astype(np.int8) \
        .astype(np.float32).view(np.complex64).reshape(w_shape)
    b_gold = np.matmul(w.transpose(1, 0, 2), x.transpose(1, 2, 0))
    return x8, w, b_gold
