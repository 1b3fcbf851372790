#!/bin/bash
# BERT-base pretraining with Ok-Topk (reference /root/reference/BERT/bert/bert_oktopk.sh:
#  seq 128, bs 8/rank, density 0.01, Adam lr 2e-4)
set -e
NGPUS=${NGPUS:-8}
density=${density:-0.01}
torchrun --nnodes=1 --nproc-per-node "$NGPUS" --master-addr 127.0.0.1 \
  -m oktopk_amd.train \
  --dnn bert_base --batch-size 8 --seq-len 128 --lr 2e-4 --optimizer adam \
  --compressor oktopk --density "$density" --max-epochs 1 --iters-per-epoch 1024
