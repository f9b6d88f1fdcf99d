"""The python ioctl mirror must match the C ABI header bit for bit.
A tiny C program compiled against module/include/rocp2p_probe_abi.h
prints the authoritative sizes/codes; the test diffs them."""
import ctypes
import os
import shutil
import subprocess

import pytest

from rocnrdma_amd.probe import abi

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

C_CHECK = r"""
#include <stdio.h>
#include <stdint.h>
#include "rocp2p_probe_abi.h"
int main(void) {
  printf("%zu %zu %zu %zu %zu\n",
         sizeof(struct rocp2p_probe_page_size),
         sizeof(struct rocp2p_probe_pin),
         sizeof(struct rocp2p_probe_unpin),
         sizeof(struct rocp2p_probe_is_gpu),
         sizeof(struct rocp2p_probe_info));
  printf("%#lx %#lx %#lx %#lx %#lx\n",
         (unsigned long)ROCP2P_PROBE_GET_PAGE_SIZE,
         (unsigned long)ROCP2P_PROBE_GET_PAGES,
         (unsigned long)ROCP2P_PROBE_PUT_PAGES,
         (unsigned long)ROCP2P_PROBE_IS_GPU_ADDRESS,
         (unsigned long)ROCP2P_PROBE_GET_INFO);
  return 0;
}
"""


def test_struct_sizes_python_side():
    assert ctypes.sizeof(abi.PageSizeParam) == 24
    assert ctypes.sizeof(abi.PinParam) == 16
    assert ctypes.sizeof(abi.UnpinParam) == 24
    assert ctypes.sizeof(abi.IsGpuParam) == 16
    assert ctypes.sizeof(abi.InfoParam) == 48


@pytest.mark.skipif(shutil.which("gcc") is None, reason="no gcc")
def test_abi_matches_c_header(tmp_path):
    src = tmp_path / "abicheck.c"
    src.write_text(C_CHECK)
    exe = tmp_path / "abicheck"
    subprocess.run(
        ["gcc", "-I", os.path.join(ROOT, "module", "include"),
         str(src), "-o", str(exe)],
        check=True)
    out = subprocess.run([str(exe)], capture_output=True, text=True,
                         check=True).stdout.splitlines()
    sizes = [int(x) for x in out[0].split()]
    codes = [int(x, 16) for x in out[1].split()]
    assert sizes == [24, 16, 24, 16, 48]
    assert codes == [abi.GET_PAGE_SIZE, abi.GET_PAGES, abi.PUT_PAGES,
                     abi.IS_GPU_ADDRESS, abi.GET_INFO]


def test_device_path_matches_header():
    hdr = open(os.path.join(ROOT, "module", "include",
                            "rocp2p_probe_abi.h")).read()
    assert f'"{abi.DEVICE_PATH}"' in hdr
    assert f"'{chr(abi.MAGIC)}'" in hdr
