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
    """Path of the C++ harness CLI that drives the verbs data plane.
    ROCNR_VERBS_HARNESS overrides (tests point it at the fake-verbs
    build; an HCA box uses the real one)."""
    env = os.environ.get("ROCNR_VERBS_HARNESS")
    if env:
        return env if os.path.exists(env) else None
    here = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    cand = os.path.join(here, "harness", "build", "rocp2p_bw")
    return cand if os.path.exists(cand) else None


def run_harness_steps(*, msg_bytes: int, region_bytes: int, steps: int,
                      warmup: int, direction: str = "write",
                      mr: str = "auto", device_index: int = 0,
                      seed: int = 0xC0FFEE,
                      timeout: float = 600.0) -> dict:
    """One native step-mode run (rocp2p_bw --steps): W untimed warmup +
    K timed steps through the REAL verbs data plane (ibv_reg_mr on the
    region per the MR mode, chained RDMA WRITE/READ WRs, CQ drain).
    Returns the harness's JSON record (keys: secs, msgs, gbps,
    integrity, ...).  Raises RuntimeError with the harness stderr on
    failure."""
    import json as _json
    import subprocess

    binary = harness_binary()
    if binary is None:
        raise RuntimeError(
            "verbs harness binary not built (make -C harness), and "
            "ROCNR_VERBS_HARNESS is not set")
    cmd = [binary, "--transport", "verbs", "--mr", mr,
           "--msg", str(msg_bytes), "--region", str(region_bytes),
           "--dir", direction, "--steps", str(steps),
           "--warmup", str(warmup), "--device", str(device_index),
           "--seed", str(seed), "--json"]
    out = subprocess.run(cmd, capture_output=True, text=True,
                         timeout=timeout)
    if out.returncode != 0:
        raise RuntimeError(
            f"verbs harness failed (rc={out.returncode}): "
            f"{out.stderr.strip() or out.stdout.strip()}")
    return _json.loads(out.stdout.strip().splitlines()[-1])


class VerbsTransport(Transport):
    name = "verbs"

    def __init__(self, *a, **kw):
        if not verbs_available():
            raise RuntimeError(
                "verbs transport: libibverbs or an IB device is missing on "
                "this host (expected on the GPU pool; use sdma).")
        raise NotImplementedError(
            "in-process verbs data plane is driven via the C++ harness "
            "(harness/build/rocp2p_bw) on verbs-equipped hosts — "
            "bench.py routes through run_harness_steps()")
