#!/bin/bash
# Sanitizer lane ("add a CI lane with compute-sanitizer-equivalent").
#
# Host lane: ASan + UBSan build of the header-only C++ surface exercised by
# the mdspan semantics test (catches OOB indexing / UB in the layout math).
# GPU lane (run on a GPU box): AMD_SERIALIZE_KERNEL=3 serializes every kernel
# launch and checks completion synchronously — the ROCm-native way to pin
# async kernel faults (page faults, OOB) to their launch site, standing in
# for CUDA's compute-sanitizer; HSA_XNACK=1 turns silent OOB reads into
# reported page faults where the address was never mapped.
set -e
cd "$(dirname "$0")/.."

echo "== host ASan+UBSan: mdspan/mdarray semantics =="
mkdir -p build
/opt/rocm/lib/llvm/bin/clang++ -std=c++17 -g -fsanitize=address,undefined \
  -fno-omit-frame-pointer -I include -D__HIP_PLATFORM_AMD__=1 \
  -I/opt/rocm/include tests/cpp/test_mdspan_host.cpp -L/opt/rocm/lib -lamdhip64 -o build/mdspan_host_asan
./build/mdspan_host_asan

if python -c 'import torch, sys; sys.exit(0 if torch.cuda.is_available() else 1)' 2>/dev/null; then
  echo "== GPU lane: serialized-kernel fault check over the kernel suite =="
  AMD_SERIALIZE_KERNEL=3 HSA_XNACK=1 \
    python -m pytest tests/test_gpu_kernels.py -q -x
else
  echo "(no GPU here: GPU serialized-kernel lane runs on the GPU box)"
fi
echo "SANITIZE OK"
