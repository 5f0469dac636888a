#!/bin/bash
# build + run the radix-sort tile-size A/B on a GPU box
set -e
for ipt in 8 16 32; do
  hipcc --offload-arch=gfx950 -O3 -DRS8_IPT=$ipt \
        csrc/tests/sort_ab.cpp csrc/hip/sortwin.hip \
        -o gpurun_out/sort_ab_$ipt 2>/dev/null || \
  hipcc --offload-arch=gfx950 -O3 -DRS8_IPT=$ipt \
        csrc/tests/sort_ab.cpp csrc/hip/sortwin.hip -o gpurun_out/sort_ab_$ipt
done
for ipt in 8 16 32; do
  ./gpurun_out/sort_ab_$ipt 4194304 15
  ./gpurun_out/sort_ab_$ipt 8388608 15
done
