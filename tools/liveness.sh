#!/bin/sh
# Registration-liveness probe for the rocp2p bridge (docs/RUNBOOK.md).
# Exit 0: bridge loaded and dispatching; 1: loaded but never used;
# 2: not loaded.
P=/sys/module/rocp2p/parameters
if [ ! -d "$P" ]; then
    echo "rocp2p: NOT LOADED"
    exit 2
fi
a=$(cat "$P/active_regs")
b=$(cat "$P/pinned_bytes")
i=$(cat "$P/invalidations")
bar=$(cat "$P/bar_bytes" 2>/dev/null || echo "?")
echo "rocp2p: active_regs=$a pinned_bytes=$b invalidations=$i bar_bytes=$bar"
if [ "$bar" != "?" ] && [ "$bar" -lt 17179869184 ] 2>/dev/null; then
    echo "rocp2p: WARNING: GPU BAR0 aperture < 16 GiB — large BAR disabled?"
fi
if [ "$a" = "0" ] && [ "$b" = "0" ]; then
    echo "rocp2p: loaded but no registrations have dispatched here."
    echo "  -> run: harness/build/rocp2p_bw --transport verbs --mr peer"
    echo "  -> if still 0: peer-mem ABI drift (docs/RUNBOOK.md)"
    exit 1
fi
exit 0
