#!/bin/bash
# Host-side ASAN/UBSAN pass: device code cannot be
# sanitized without xnack, but every HOST path of the native layer can —
# the extension's wrapper/sizing/cache logic and the microbench driver.
# Run on a GPU box:  bash tools/sanitize_check.sh
set -e
cd "$(dirname "$0")/.."
OUT=gpurun_out; mkdir -p $OUT

echo "== 1/2 microbench under ASan+UBSan (host) =="
hipcc --offload-arch=gfx950 -O1 -g -fsanitize=address,undefined \
    -fno-omit-frame-pointer tools/mfma_ab.hip -o $OUT/mfma_asan
ASAN_OPTIONS=detect_leaks=0 $OUT/mfma_asan 128 > $OUT/asan_mfma.log 2>&1 \
    && echo "   microbench: CLEAN" || { tail -20 $OUT/asan_mfma.log; exit 1; }

echo "== 2/2 extension host wrappers under UBSan =="
# (full ASan cannot coexist with the HIP runtime's fixed-address
# mappings in-process — the microbench above carries the ASan pass;
# UBSan has no allocator interposition and runs the real extension)
mkdir -p $OUT/asan_build
hipcc --offload-arch=gfx950 -O1 -g -fsanitize=undefined \
    -fno-omit-frame-pointer -shared -fPIC \
    -I"$(python -c 'import torch, os; print(os.path.join(os.path.dirname(torch.__file__), "include"))')" \
    -I"$(python -c 'import torch, os; print(os.path.join(os.path.dirname(torch.__file__), "include", "torch", "csrc", "api", "include"))')" \
    -I/usr/include/python3.10 \
    -D__HIP_PLATFORM_AMD__=1 -DUSE_ROCM=1 -DHIP_ENABLE_WARP_SYNC_BUILTINS=1 \
    -DTORCH_API_INCLUDE_EXTENSION_H -DTORCH_EXTENSION_NAME=_C_asan \
    -fno-gpu-rdc -std=c++17 csrc/ps_kernels.hip \
    -L"$(python -c 'import torch, os; print(os.path.join(os.path.dirname(torch.__file__), "lib"))')" \
    -L/opt/rocm/lib -lc10 -ltorch -ltorch_cpu -ltorch_python -lamdhip64 \
    -lc10_hip -ltorch_hip -o $OUT/asan_build/_C_asan.so
UBSAN_RT="$(hipcc -print-file-name=libclang_rt.ubsan_standalone-x86_64.so)"
LD_PRELOAD="$UBSAN_RT" \
UBSAN_OPTIONS=halt_on_error=1:print_stacktrace=1 \
python tools/sanitize_exercise.py > $OUT/asan_ext.log 2>&1 \
    && echo "   extension wrappers: CLEAN (UBSan)" \
    || { tail -30 $OUT/asan_ext.log; exit 1; }
