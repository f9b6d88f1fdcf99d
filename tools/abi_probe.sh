#!/bin/sh
# abi_probe.sh — inspect the TARGET box's OFED peer_mem.h and ROCK
# amd_rdma.h and print the drift switches + make line to build the
# bridge against them (SURVEY.md §7 hard part #1: "peer_mem ABI drift
# 2016→now ... verify against the installed OFED before freezing").
#
# Usage: tools/abi_probe.sh [PEER_MEM_H] [AMD_RDMA_H]
#        (no args: search the standard locations)

find_first() {
    for f in "$@"; do
        [ -f "$f" ] && { echo "$f"; return 0; }
    done
    return 1
}

PEER=${1:-$(find_first \
    /usr/src/ofa_kernel/default/include/rdma/peer_mem.h \
    /usr/src/ofa_kernel/*/include/rdma/peer_mem.h \
    /var/lib/dkms/mlnx-ofed-kernel/*/build/include/rdma/peer_mem.h \
    /usr/include/rdma/peer_mem.h 2>/dev/null)}
AMDR=${2:-$(find_first \
    "/lib/modules/$(uname -r)/build/include/drm/amd_rdma.h" \
    /usr/src/amdgpu/include/drm/amd_rdma.h \
    /usr/src/amdgpu-*/include/drm/amd_rdma.h 2>/dev/null)}

rc=0
echo "== PeerDirect ABI (peer_mem.h) =="
if [ -z "$PEER" ]; then
    echo "  NOT FOUND — install MLNX_OFED with peer-memory support."
    echo "  Falling back to the vendored header (defaults below)."
    CORE_U64=1
    rc=1
else
    echo "  $PEER"
    # the registration cookie type in get_pages decides the big switch
    if grep -E 'get_pages' -A3 "$PEER" | grep -qE 'u64[[:space:]]+core_context'; then
        CORE_U64=1
        echo "  get_pages core_context: u64  (modern, >= OFED 4.x)"
    elif grep -E 'get_pages' -A3 "$PEER" | grep -qE 'void[[:space:]]*\*[[:space:]]*core_context'; then
        CORE_U64=0
        echo "  get_pages core_context: void* (legacy 2016 ABI)"
    else
        CORE_U64=1
        echo "  WARNING: could not classify core_context — defaulting to u64;"
        echo "  check $PEER manually."
        rc=1
    fi
    if grep -q 'peer_memory_client_ex' "$PEER"; then
        HAS_EX=1
        echo "  extended registration (peer_memory_client_ex): present"
    else
        HAS_EX=0
        echo "  extended registration: ABSENT — build with"
        echo "  ROCNR_PEER_MEM_HAS_EX=0 (bridge registers the plain client)"
    fi
fi
[ -z "$PEER" ] && HAS_EX=1

echo "== amdkfd RDMA ABI (amd_rdma.h) =="
if [ -z "$AMDR" ]; then
    echo "  NOT FOUND — install the ROCK/amdgpu DKMS headers."
    echo "  Falling back to the vendored header (defaults below)."
    HAS_DMADEV=1
    rc=1
else
    echo "  $AMDR"
    if grep -E 'get_pages' -A6 "$AMDR" | grep -qE 'struct[[:space:]]+device[[:space:]]*\*'; then
        HAS_DMADEV=1
        echo "  get_pages takes struct device* (modern ROCK)"
        echo "  NOTE: whether this KFD accepts dma_dev=NULL (bridge maps"
        echo "  per-HCA later) is NOT probeable from the header.  If pins"
        echo "  fail at registration with the bridge loaded, the bridge's"
        echo "  null_dev_fallback (default on) defers the pin to dma_map"
        echo "  and logs 'amd_rdma ABI drift fallback' — check dmesg."
    else
        HAS_DMADEV=0
        echo "  get_pages has NO dma_dev parameter (legacy KFD)"
    fi
fi

echo "== build line =="
LINE="make -C module/bridge KDIR=/lib/modules/\$(uname -r)/build"
[ -n "$PEER" ] && LINE="$LINE OFA_DIR=$(dirname "$(dirname "$(dirname "$PEER")")")"
[ -n "$AMDR" ] && LINE="$LINE AMD_RDMA=$(dirname "$AMDR")"
LINE="$LINE ROCNR_ABI_FLAGS=\"-DROCNR_PEER_MEM_CORE_CONTEXT_U64=$CORE_U64 -DROCNR_PEER_MEM_HAS_EX=$HAS_EX -DROCNR_AMD_RDMA_HAS_DMA_DEV=$HAS_DMADEV\""
echo "  $LINE"
exit $rc
