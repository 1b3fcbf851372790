#!/bin/bash
# Race-detection-style CI job (the reference has no
# sanitizer story; ROCm ships no compute-sanitizer in this image, so this
# job serialises kernel launches and copies — races that depend on kernel
# overlap/ordering change behaviour under it — and runs the GPU suite twice
# (ordering-sensitive nondeterminism shows as cross-run diffs in the
# bit-exact tests).
set -e
cd "$(dirname "$0")/.."
echo "== pass 1: serialized kernels =="
AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 python -m pytest tests -q -m gpu -x
echo "== pass 2: default scheduling =="
python -m pytest tests -q -m gpu -x
echo "sanitize: OK"
