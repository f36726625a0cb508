#!/bin/bash
# Optional sanitizer / race lane: rerun the atomics- and
# race-sensitive GPU tests under serialized kernel execution
# (AMD_SERIALIZE_KERNEL=3 forces a sync after every kernel, surfacing
# ordering assumptions) plus repeated runs for the determinism claims.
set -e
cd "$(dirname "$0")/.."
echo "== serialized-kernel pass =="
AMD_SERIALIZE_KERNEL=3 python -m pytest tests -m gpu -q \
    -k "race or determinism or auto_grow or hipgraph or fused"
echo "== repeat determinism x3 =="
for i in 1 2 3; do
  python -m pytest tests/test_gpu.py -q -k "determinism or race"
done
echo "race lane OK"
