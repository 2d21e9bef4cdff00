#!/bin/sh
# build_abl.sh — A/B ablation builds of the pipe kernel (perf work).
# Produces ../libvmgpu_<name>.so variants selectable via VMGPU_LIB.
set -e
cd "$(dirname "$0")"
FLAGS="--offload-arch=gfx950 -O3 -std=c++17 -ffp-contract=off -fPIC"
mkdir -p .abl

# rollup path only (the bench exercises nothing else) — keeps the libs small
build_one() {
  name=$1
  shift
  hipcc $FLAGS "$@" -shared vmgpu.hip decode.hip binop.hip transform.hip aggrcol.hip -o ../libvmgpu_$name.so
}

build_one pipe_stage -DVMGPU_PIPE_ABL_STAGE &
build_one pipe_copy -DVMGPU_PIPE_ABL_STAGE -DVMGPU_PIPE_ABL_NO_SCAN &
build_one pipe_noeval -DVMGPU_PIPE_ABL_NO_EVAL &
build_one pipe_noscrape -DVMGPU_ABL_NO_SCRAPE &
wait
build_one pipe_mw8 -DVMGPU_PIPE_MINWAVES=8 &
build_one pipe_mw5 -DVMGPU_PIPE_MINWAVES=5 &
build_one pipe_mw6 -DVMGPU_PIPE_MINWAVES=6 &
build_one pipe_u2 -DVMGPU_PIPE_UNROLL=2 &
build_one pipe_scat -DVMGPU_PIPE_SCATTER &
wait
echo "ablation libs built"
