#!/bin/bash
# Build + verify (reference: setup.sh installed conda deps; this image is
# offline, so setup = compile the gfx950 extension in-tree and run the
# CPU test suite).
set -e
cd "$(dirname "$0")/.."
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
python -m pytest tests -q -m "not gpu"
