"""IB-verbs capability probe (PeerDirect path).

The verbs data plane lives in the native harness
(harness/src/verbs_backend.cpp): loopback QP pair, client/server mode
with TCP OOB bootstrap, peer/dmabuf/host MR modes — CI-executed through
the fake-verbs layer and activated for real on HCA-equipped hosts.
This module only answers "is a usable verbs stack present?" (dlopen +
device count) so auto-transport selection and the runbook's layer
isolation work; it deliberately implements no python data plane.
"""
from __future__ import annotations

import ctypes
import ctypes.util
import os

from .base import Transport

_LIBNAMES = ("libibverbs.so.1", "libibverbs.so")


def _load_libibverbs():
    for name in _LIBNAMES:
        try:
            return ctypes.CDLL(name)
        except OSError:
            continue
    found = ctypes.util.find_library("ibverbs")
    if found:
        try:
            return ctypes.CDLL(found)
        except OSError:
            pass
    return None


def verbs_available() -> bool:
    """True iff libibverbs loads AND at least one IB device exists."""
    lib = _load_libibverbs()
    if lib is None:
        return False
    try:
        lib.ibv_get_device_list.restype = ctypes.POINTER(ctypes.c_void_p)
        lib.ibv_get_device_list.argtypes = [ctypes.POINTER(ctypes.c_int)]
        n = ctypes.c_int(0)
        devs = lib.ibv_get_device_list(ctypes.byref(n))
        ok = bool(devs) and n.value > 0
        if devs:
            lib.ibv_free_device_list(devs)
        return ok
    except AttributeError:
        return False


def harness_binary() -> str | None:
    """Path of the built C++ harness CLI, if present."""
    here = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    cand = os.path.join(here, "harness", "build", "rocp2p_bw")
    return cand if os.path.exists(cand) else None


class VerbsTransport(Transport):
    name = "verbs"

    def __init__(self, *a, **kw):
        if not verbs_available():
            raise RuntimeError(
                "verbs transport: libibverbs or an IB device is missing on "
                "this host (expected on the GPU pool; use sdma).")
        raise NotImplementedError(
            "in-process verbs data plane is driven via the C++ harness "
            "(harness/build/rocp2p_bw) on verbs-equipped hosts")
