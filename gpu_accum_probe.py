"""GPU-box probe: the accumulate block's device path (bf.map casts)."""
import numpy as np
import bifrost_amd as bf
from tests.test_pipeline_cpu import NumpySourceBlock, CollectBlock

raw = np.zeros((8, 4), dtype=[("re", np.int8), ("im", np.int8)])
raw["re"] = (np.arange(32).reshape(8, 4) % 11) - 5
raw["im"] = (np.arange(32).reshape(8, 4) % 7) - 3
out = []
with bf.Pipeline() as pipe:
    src = NumpySourceBlock([raw], gulp_nframe=1)
    dev = bf.blocks.copy(src, space="cuda")
    acc = bf.blocks.accumulate(dev, 4, dtype="cf32", gulp_nframe=1)
    host = bf.blocks.copy(acc, space="system")
    CollectBlock(host, out)
    pipe.run()
got = np.concatenate(out, axis=0)
want = raw["re"].astype(np.float32) + 1j * raw["im"].astype(np.float32)
np.testing.assert_allclose(got[0], want[0:4].sum(axis=0))
np.testing.assert_allclose(got[1], want[4:8].sum(axis=0))
print("accumulate cuda path: OK", flush=True)
