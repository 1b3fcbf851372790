#!/bin/bash
# 8-GPU flag A/B decision procedure (docs/round2_plan.md): run on an
# MI355X node; writes profiles/scale_readiness.json.
N=${NGPUS:-8}
torchrun --nnodes=1 --nproc-per-node "$N" --master-addr 127.0.0.1 \
  tools/scale_readiness.py --steps "${STEPS:-40}" "$@"
