"""gfx950 payload kernels (fill / verify / CRC32 / copy).

The native extension is mandatory on GPU: if torch sees a device but the
extension is missing, every entry point raises — no silent eager
fallback (CPU references for tests live in rocnrdma_amd.utils.pattern).
"""
from __future__ import annotations

import torch

try:
    from . import _p2p_ext  # built in-tree by setup.py (gfx950)
except ImportError:  # pragma: no cover - exercised only on broken installs
    _p2p_ext = None


def have_ext() -> bool:
    return _p2p_ext is not None


def _require():
    if _p2p_ext is None:
        raise RuntimeError(
            "rocnrdma_amd.ops._p2p_ext is not built. Run "
            "`python __graft_entry__.py build` (PYTORCH_ROCM_ARCH=gfx950). "
            "The GPU path never falls back to eager."
        )
    return _p2p_ext


def fill_(buf: torch.Tensor, seed: int) -> None:
    """In-place splitmix64 pattern fill (word i = mix(seed+(i+1)*PHI))."""
    _require().fill_(buf, seed)


def verify(buf: torch.Tensor, seed: int) -> int:
    """Count 8-byte words deviating from the fill pattern (on-GPU)."""
    return _require().verify(buf, seed)


def crc32_pages(buf: torch.Tensor) -> torch.Tensor:
    """zlib-compatible CRC32 of each 4 KiB page; int32 tensor on GPU."""
    return _require().crc32_pages(buf)


def copy_(dst: torch.Tensor, src: torch.Tensor) -> None:
    """Streaming 16 B/lane device copy (bandwidth ceiling probe)."""
    _require().copy_(dst, src)


def copy_nt_(dst: torch.Tensor, src: torch.Tensor) -> None:
    """Nontemporal streaming copy (streamed-once data stays out of L2)."""
    _require().copy_nt_(dst, src)


def gather_(region: torch.Tensor, dst_offs: torch.Tensor,
            src_addrs: torch.Tensor, msg_bytes: int) -> None:
    """Batched message engine: host-pinned sources -> HBM region offsets
    (one launch retires the whole batch — the NIC-WQE analog)."""
    _require().gather_(region, dst_offs, src_addrs, msg_bytes)


def scatter_(region: torch.Tensor, src_offs: torch.Tensor,
             dst_addrs: torch.Tensor, msg_bytes: int) -> None:
    """Batched message engine: HBM region offsets -> host-pinned dests."""
    _require().scatter_(region, src_offs, dst_addrs, msg_bytes)


def dmabuf_fd(buf: torch.Tensor) -> int:
    """Export an HBM tensor as a dmabuf fd (ibv_reg_dmabuf_mr's GPU
    half). Caller must os.close() the fd."""
    return _require().dmabuf_fd(buf)
