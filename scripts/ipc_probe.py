"""Probe hipIpc memory-handle sharing across processes (dmabuf IPC mode).
Parent hipMallocs + fills a buffer, child opens the IPC handle and
verifies the bytes — feasibility for cross-process GPU short-circuit."""
import ctypes
import os
import subprocess
import sys

libhip = ctypes.CDLL("libamdhip64.so")
HANDLE_SZ = 64


class IpcHandle(ctypes.Structure):
    # hipIpcMemHandle_t: 64 reserved bytes, passed BY VALUE to open
    _fields_ = [("reserved", ctypes.c_char * HANDLE_SZ)]


def check(rc, what):
    if rc != 0:
        raise RuntimeError(f"{what} -> {rc}")

if len(sys.argv) > 1 and sys.argv[1] == "child":
    h = bytes.fromhex(sys.argv[2])
    check(libhip.hipSetDevice(0), "child setdev")
    ptr = ctypes.c_void_p()
    hd = IpcHandle()
    ctypes.memmove(hd.reserved, h, HANDLE_SZ)
    libhip.hipIpcOpenMemHandle.argtypes = [
        ctypes.POINTER(ctypes.c_void_p), IpcHandle, ctypes.c_uint]
    rc = libhip.hipIpcOpenMemHandle(ctypes.byref(ptr), hd,
                                    ctypes.c_uint(1))  # lazy peer access
    check(rc, "hipIpcOpenMemHandle")
    out = ctypes.create_string_buffer(16)
    check(libhip.hipMemcpy(out, ptr, 16, 2), "hipMemcpy D2H")  # D2H=2
    print("child got:", out.raw.hex())
    assert out.raw == bytes(range(16)), out.raw
    check(libhip.hipIpcCloseMemHandle(ptr), "close")
    print("CHILD_OK")
    sys.exit(0)

dev = ctypes.c_void_p()
check(libhip.hipSetDevice(0), "setdev")
check(libhip.hipMalloc(ctypes.byref(dev), 4096), "malloc")
src = ctypes.create_string_buffer(bytes(range(16)), 16)
check(libhip.hipMemcpy(dev, src, 16, 1), "hipMemcpy H2D")
handle = ctypes.create_string_buffer(HANDLE_SZ)
rc = libhip.hipIpcGetMemHandle(handle, dev)
print("hipIpcGetMemHandle rc:", rc)
if rc != 0:
    print("IPC_UNSUPPORTED")
    sys.exit(0)
r = subprocess.run([sys.executable, __file__, "child", handle.raw.hex()],
                   capture_output=True, text=True, timeout=120)
print(r.stdout, r.stderr[-400:])
print("PARENT_OK" if "CHILD_OK" in r.stdout else "IPC_CHILD_FAILED")
