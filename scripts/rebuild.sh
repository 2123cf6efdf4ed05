#!/bin/bash
# Torch's cpp_extension build emits no header depfiles for .hip sources, so
# header edits leave stale objects (ABI mismatch -> memory faults). Always
# clean-build.
set -e
cd "$(dirname "$0")/.."
rm -rf build
PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace "$@"
