#!/bin/bash
# CI-style sanitizer pass over the HIP kernels.
# No compute-sanitizer ships in this image; the recipe is:
#   - AMD_SERIALIZE_KERNEL=3: every kernel launches synchronously and any
#     fault (incl. OOB reads that leave the mapped heap) aborts AT the
#     offending launch instead of corrupting later state;
#   - tests/test_gpu_sanitize.py: canary-halo (guard-band) tensors around
#     every kernel's inputs/outputs, verified bitwise after each op
#     (catches OOB WRITES that stay inside the heap);
#   - the full numerics suite re-run under serialization.
set -e
cd "$(dirname "$0")/.."
export AMD_SERIALIZE_KERNEL=3
export PYTHONUNBUFFERED=1
echo "== guard-band canary pass (AMD_SERIALIZE_KERNEL=3) =="
python -m pytest tests/test_gpu_sanitize.py -v -m gpu
echo "== full kernel numerics under serialized launches =="
python -m pytest tests/test_gpu_kernels.py -q -m gpu
echo "SANITIZE PASS CLEAN"
