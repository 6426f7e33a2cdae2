"""GPU-box probe: ndarray.tofile on a device array (stages via system)."""
import numpy as np
import bifrost_amd as bf

host = np.arange(1024, dtype=np.float32).reshape(32, 32)
dev = bf.asarray(bf.ndarray(host), space="cuda")
with open("/tmp/dev.dat", "wb") as f:
    dev.tofile(f)
back = np.fromfile("/tmp/dev.dat", dtype=np.float32).reshape(32, 32)
np.testing.assert_array_equal(back, host)
print("device tofile: OK", flush=True)
