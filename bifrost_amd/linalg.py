"""LinAlg wrapper over bfLinAlgMatMul (reference python/bifrost/linalg.py
surface; backed by hand-written CDNA4 HIP kernels)."""

from bifrost_amd.libbifrost import _bf, _check, BifrostObject
from bifrost_amd.ndarray import asarray

__all__ = ["LinAlg"]


class LinAlg(BifrostObject):
    def __init__(self):
        BifrostObject.__init__(self, _bf.bfLinAlgCreate, _bf.bfLinAlgDestroy)

    def matmul(self, alpha, a, b, beta, c):
        """Computes c = alpha*a.b + beta*c, or a.a^H / b^H.b when b / a is
        None.  numpy.matmul batch semantics over leading dims; the herk
        forms fill only the lower triangle of c."""
        if alpha is None:
            alpha = 1.0
        if beta is None:
            beta = 0.0
        a_arr = asarray(a).as_BFarray() if a is not None else None
        b_arr = asarray(b).as_BFarray() if b is not None else None
        c_arr = asarray(c).as_BFarray()
        _check(_bf.bfLinAlgMatMul(self.obj, float(alpha), a_arr, b_arr,
                                  float(beta), c_arr))
        return c
