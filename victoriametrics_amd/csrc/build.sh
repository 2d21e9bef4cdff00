#!/bin/sh
# Builds victoriametrics_amd/libvmgpu.so for gfx950.
# -ffp-contract=off: parity with the uncontracted reference semantics.
set -e
cd "$(dirname "$0")"
hipcc --offload-arch=gfx950 -O3 -std=c++17 -ffp-contract=off -fPIC -shared \
  vmgpu.hip decode.hip binop.hip transform.hip aggrcol.hip -o ../libvmgpu.so "$@"
