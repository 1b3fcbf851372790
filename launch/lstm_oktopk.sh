#!/bin/bash
# DeepSpeech-style AN4 Ok-Topk training (reference /root/reference/LSTM/lstm_oktopk.sh)
set -e
NGPUS=${NGPUS:-8}
density=${density:-0.01}
torchrun --nnodes=1 --nproc-per-node "$NGPUS" --master-addr 127.0.0.1 \
  -m oktopk_amd.train \
  --dnn lstman4 --batch-size 2 --lr 1e-3 --optimizer sgd \
  --compressor oktopk --density "$density" --max-epochs 10
