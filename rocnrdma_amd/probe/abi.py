"""ctypes mirror of the rocp2p_probe ioctl ABI
(module/include/rocp2p_probe_abi.h) — userspace client side.

The reference defined an ioctl ABI but never shipped a userspace client
(SURVEY.md §4); this module + tools/rocp2p_probe_cli.c are that client.
"""
from __future__ import annotations

import ctypes
import os

DEVICE_PATH = "/dev/rocp2p_probe"
MAGIC = ord("R")

_IOC_NRBITS = 8
_IOC_TYPEBITS = 8
_IOC_SIZEBITS = 14
_IOC_NRSHIFT = 0
_IOC_TYPESHIFT = _IOC_NRSHIFT + _IOC_NRBITS
_IOC_SIZESHIFT = _IOC_TYPESHIFT + _IOC_TYPEBITS
_IOC_DIRSHIFT = _IOC_SIZESHIFT + _IOC_SIZEBITS
_IOC_WRITE = 1
_IOC_READ = 2


def _ioc(direction: int, nr: int, size: int) -> int:
    return ((direction << _IOC_DIRSHIFT) | (MAGIC << _IOC_TYPESHIFT)
            | (nr << _IOC_NRSHIFT) | (size << _IOC_SIZESHIFT))


class PageSizeParam(ctypes.Structure):
    _fields_ = [("addr", ctypes.c_uint64), ("length", ctypes.c_uint64),
                ("page_size", ctypes.c_uint64)]


class PinParam(ctypes.Structure):
    _fields_ = [("addr", ctypes.c_uint64), ("length", ctypes.c_uint64)]


class UnpinParam(ctypes.Structure):
    _fields_ = [("addr", ctypes.c_uint64), ("length", ctypes.c_uint64),
                ("released", ctypes.c_uint64)]


class IsGpuParam(ctypes.Structure):
    _fields_ = [("addr", ctypes.c_uint64), ("is_gpu", ctypes.c_uint64)]


class InfoParam(ctypes.Structure):
    _fields_ = [("addr", ctypes.c_uint64), ("length", ctypes.c_uint64),
                ("nents", ctypes.c_uint64), ("total_bytes", ctypes.c_uint64),
                ("first_dma_addr", ctypes.c_uint64),
                ("max_seg_bytes", ctypes.c_uint64)]


GET_PAGE_SIZE = _ioc(_IOC_READ | _IOC_WRITE, 1, ctypes.sizeof(PageSizeParam))
GET_PAGES = _ioc(_IOC_WRITE, 2, ctypes.sizeof(PinParam))
PUT_PAGES = _ioc(_IOC_READ | _IOC_WRITE, 3, ctypes.sizeof(UnpinParam))
IS_GPU_ADDRESS = _ioc(_IOC_READ | _IOC_WRITE, 4, ctypes.sizeof(IsGpuParam))
GET_INFO = _ioc(_IOC_READ | _IOC_WRITE, 5, ctypes.sizeof(InfoParam))


class ProbeDevice:
    """Userspace client for /dev/rocp2p_probe (requires the module)."""

    def __init__(self, path: str = DEVICE_PATH):
        import fcntl  # noqa: F401  (ensure availability early)

        self.fd = os.open(path, os.O_RDWR)

    def close(self):
        os.close(self.fd)

    def _ioctl(self, code: int, param) -> None:
        import fcntl

        fcntl.ioctl(self.fd, code, param)

    def is_gpu_address(self, addr: int) -> bool:
        p = IsGpuParam(addr=addr)
        self._ioctl(IS_GPU_ADDRESS, p)
        return bool(p.is_gpu)

    def get_page_size(self, addr: int, length: int) -> int:
        p = PageSizeParam(addr=addr, length=length)
        self._ioctl(GET_PAGE_SIZE, p)
        return p.page_size

    def pin(self, addr: int, length: int) -> None:
        self._ioctl(GET_PAGES, PinParam(addr=addr, length=length))

    def unpin(self, addr: int, length: int) -> int:
        p = UnpinParam(addr=addr, length=length)
        self._ioctl(PUT_PAGES, p)
        return p.released

    def info(self, addr: int, length: int) -> InfoParam:
        p = InfoParam(addr=addr, length=length)
        self._ioctl(GET_INFO, p)
        return p
