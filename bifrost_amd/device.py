"""Device/stream control (reference python/bifrost/device.py surface).
On this backend a device is a HIP device and streams are hipStream_t."""

import ctypes

from bifrost_amd.libbifrost import _bf, _check

__all__ = ["set_device", "get_device", "set_devices_no_spin_cpu",
           "stream_synchronize", "set_stream", "get_stream"]


def set_device(device):
    if isinstance(device, int):
        _check(_bf.bfDeviceSet(device))
    else:
        _check(_bf.bfDeviceSetById(str(device).encode()))


def get_device():
    d = ctypes.c_int()
    _check(_bf.bfDeviceGet(ctypes.byref(d)))
    return d.value


def set_devices_no_spin_cpu():
    _check(_bf.bfDevicesSetNoSpinCPU())


def stream_synchronize():
    _check(_bf.bfStreamSynchronize())


def set_stream(stream):
    s = ctypes.c_void_p(int(stream))
    _check(_bf.bfStreamSet(ctypes.byref(s)))


def get_stream():
    s = ctypes.c_void_p()
    _check(_bf.bfStreamGet(ctypes.byref(s)))
    return s.value if s.value is not None else 0


class ExternalStream(object):
    """Context manager to run bifrost ops on a stream created outside
    (e.g. a torch.cuda.Stream): the original stream is restored on exit.

        with bf.device.ExternalStream(torch.cuda.current_stream()):
            ...
    """

    def __init__(self, stream):
        self._stream = stream

    def use(self):
        self._orig_stream = get_stream()
        stream = getattr(self._stream, "cuda_stream", None)  # torch
        if stream is None:
            stream = getattr(self._stream, "ptr", None)  # cupy
        if stream is None:
            stream = getattr(self._stream, "handle", None)  # pycuda
        if stream is None:
            stream = self._stream
        set_stream(stream)

    def __enter__(self):
        self.use()
        return self

    def __exit__(self, t, v, tb):
        set_stream(self._orig_stream)
        del self._orig_stream

    def __del__(self):
        try:
            set_stream(self._orig_stream)
        except AttributeError:
            pass
