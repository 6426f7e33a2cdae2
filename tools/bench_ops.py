# Feeder + beamformer throughput measurement (committed evidence for
# DESIGN.md; run on the GPU box)
import time, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
import bifrost_amd as bf
from bifrost_amd import device as bfdev
from bifrost_amd.linalg import LinAlg

torch.cuda.init()
bfdev.set_stream(torch.cuda.current_stream().cuda_stream)

def timeit(fn, n=30, warm=5):
    for _ in range(warm): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n

# unpack ci4->ci8: 256 MB packed input
nel = 256 * 1024 * 1024
raw = np.random.randint(0, 256, size=nel, dtype=np.uint8)
i4 = bf.asarray(bf.ndarray(raw.view(bf.DataType.ci4)), space="cuda")
o8 = bf.ndarray(shape=i4.shape, dtype="ci8", space="cuda")
dt = timeit(lambda: bf.unpack(i4, o8))
print("unpack ci4->ci8: %.3f ms, %.1f GB/s (in+out)" % (dt*1e3, (nel*3)/dt/1e9))

# quantize cf32->ci8: 1G floats
n2 = 128 * 1024 * 1024
qin = bf.ndarray(shape=(n2,), dtype="cf32", space="cuda")
qout = bf.ndarray(shape=(n2,), dtype="ci8", space="cuda")
dt = timeit(lambda: bf.quantize(qin, qout, 0.1))
print("quantize cf32->ci8: %.3f ms, %.1f GB/s" % (dt*1e3, (n2*10)/dt/1e9))

# transpose feeder [t,c,sp]->[c,t,sp] 2 GB
t_, c_, s_ = 4096, 512, 1024
src = bf.ndarray(shape=(t_, c_, s_), dtype="i8", space="cuda")
dst = bf.ndarray(shape=(c_, t_, s_), dtype="i8", space="cuda")
dt = timeit(lambda: bf.transpose(dst, src, (1, 0, 2)), n=10)
gb = t_*c_*s_*2
print("transpose feeder 2GB: %.3f ms, %.1f GB/s (rw)" % (dt*1e3, gb/dt/1e9))

# general transpose (fastest dim moves): [a,b] -> [b,a] 1 GB f32
A = bf.ndarray(shape=(16384, 16384), dtype="f32", space="cuda")
B = bf.ndarray(shape=(16384, 16384), dtype="f32", space="cuda")
dt = timeit(lambda: bf.transpose(B, A, (1, 0)), n=10)
gb = 16384*16384*4*2
print("transpose 2D 1GBx2: %.3f ms, %.1f GB/s (rw)" % (dt*1e3, gb/dt/1e9))

# beamformer config-5-ish: nbeam=64, nstand=256, nchan=512, ntime=1024
la = LinAlg()
ntime, nbeam, ks, nchan = 1024, 64, 512, 512
x8 = np.random.randint(-127, 128, size=(ntime, nchan, ks, 2), dtype=np.int8)
x = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8).reshape(ntime, nchan, ks)), space="cuda")
w = bf.ndarray(shape=(nbeam, nchan, ks), dtype="cf32", space="cuda")
bout = bf.ndarray(shape=(nchan, nbeam, ntime), dtype="cf32", space="cuda")
xv = x.transpose(1, 2, 0); wv = w.transpose(1, 0, 2)
dt = timeit(lambda: la.matmul(1, wv, xv, 0, bout), n=10)
samp = ntime * nchan
flops = ntime*nchan*ks*nbeam*8
print("beamform b64 s256 c512 t1024: %.3f ms, %.3f Gsamp/s, %.1f TFLOP/s"
      % (dt*1e3, samp/dt/1e9, flops/dt/1e12))

# reduce: [8192, 4096] f32 time-scrunch by 16 (wave kernel path)
A = bf.ndarray(shape=(8192, 4096), dtype="f32", space="cuda")
B = bf.ndarray(shape=(8192, 256), dtype="f32", space="cuda")
dt = timeit(lambda: bf.reduce(A, B, "mean"), n=10)
gb = 8192*4096*4
print("reduce f32 128MB axis-scrunch x16: %.3f ms, %.1f GB/s (read)"
      % (dt*1e3, gb/dt/1e9))

# fft: batched c2c 4096-point, 256 MB
X = bf.ndarray(shape=(8192, 4096), dtype="cf32", space="cuda")
Y = bf.ndarray(shape=(8192, 4096), dtype="cf32", space="cuda")
f = bf.Fft()
f.init(X, Y, axes=[1])
dt = timeit(lambda: f.execute(X, Y), n=10)
fft_flops = 5 * 8192 * 4096 * 12  # 5 N log2 N
print("fft c2c 8192x4096 batched: %.3f ms, %.1f GFLOP/s, %.1f GB/s (rw)"
      % (dt*1e3, fft_flops/dt/1e9, 2*8192*4096*8/dt/1e9))

# beamformer with ci4 input (config C5's 4-bit width)
x4buf = np.random.randint(0, 256, size=(ntime, nchan, ks), dtype=np.uint8)
x4 = bf.asarray(bf.ndarray(x4buf.view(bf.DataType.ci4)), space="cuda")
x4v = x4.transpose(1, 2, 0)
dt = timeit(lambda: la.matmul(1, wv, x4v, 0, bout), n=10)
print("beamform ci4 b64 s256 c512 t1024: %.3f ms, %.3f Gsamp/s, %.1f TFLOP/s"
      % (dt*1e3, samp/dt/1e9, flops/dt/1e12))
