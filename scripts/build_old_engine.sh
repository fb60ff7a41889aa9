#!/bin/bash
# Rebuild the pre-pipeline engine used as the same-box A/B control in
# BASELINE.md (bench.py --staged-push + DZ_ENGINE_SO=denormalized_amd/_dzengine_old.so).
# f05d493 is the last commit before the deferred-push pipeline work.
set -e
cd "$(dirname "$0")/.."
git show f05d493:denormalized_amd/csrc/window_op.cpp > /tmp/wop_old.cpp
hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared -Iinclude \
  -Idenormalized_amd/csrc denormalized_amd/csrc/kernels.hip /tmp/wop_old.cpp \
  -o denormalized_amd/_dzengine_old.so
echo "built denormalized_amd/_dzengine_old.so"
