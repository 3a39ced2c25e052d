"""Measure H2D bandwidth: 1 vs 2 concurrent copy streams (2 SDMA engines)."""
import ctypes
import time

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import pumiumtally_amd as pt  # loads the runtime


def main():
    hip = ctypes.CDLL("libamdhip64.so")
    n = 512 * 1024 * 1024
    half = n // 2
    src = pt.pinned_array((n,), "uint8")
    src[:] = 1
    p = src.ctypes.data_as(ctypes.c_void_p)
    p2 = ctypes.c_void_p(p.value + half)
    d1, d2 = ctypes.c_void_p(), ctypes.c_void_p()
    assert hip.hipMalloc(ctypes.byref(d1), ctypes.c_size_t(half)) == 0
    assert hip.hipMalloc(ctypes.byref(d2), ctypes.c_size_t(half)) == 0
    s1, s2 = ctypes.c_void_p(), ctypes.c_void_p()
    assert hip.hipStreamCreateWithFlags(ctypes.byref(s1), 1) == 0
    assert hip.hipStreamCreateWithFlags(ctypes.byref(s2), 1) == 0

    def copy(stream, dst, srcp, bytes_):
        assert hip.hipMemcpyAsync(dst, srcp, ctypes.c_size_t(bytes_), 1, stream) == 0

    # warm
    copy(s1, d1, p, half); copy(s2, d2, p2, half)
    hip.hipDeviceSynchronize()

    reps = 6
    t0 = time.time()
    for _ in range(reps):
        copy(s1, d1, p, half)
        copy(s1, d2, p2, half)
    hip.hipDeviceSynchronize()
    one = (time.time() - t0) / reps
    t0 = time.time()
    for _ in range(reps):
        copy(s1, d1, p, half)
        copy(s2, d2, p2, half)
    hip.hipDeviceSynchronize()
    two = (time.time() - t0) / reps
    print(f"1 stream: {n/one/1e9:.1f} GB/s   2 streams: {n/two/1e9:.1f} GB/s")


if __name__ == "__main__":
    main()
