#!/bin/sh
# Full CPU-tier CI: build everything, run every no-GPU test.
set -e
python __graft_entry__.py build
python -m pytest tests -q -m "not gpu"
echo "CI (CPU tier) PASSED"
