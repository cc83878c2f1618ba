#!/bin/bash
# Launcher for examples/pretrain_bert (reference ships an sbatch/bash script per
# example, e.g. ziya_llama/finetune_with_tp.sh).  One process per GPU
# over RCCL; single node.  Tune NPROC to the node.
set -euo pipefail
cd "$(dirname "$0")"

NPROC=${NPROC:-$(python -c 'import torch;print(max(torch.cuda.device_count(),1))')}
export HSA_ENABLE_IPC_MODE_LEGACY=0
export MASTER_ADDR=127.0.0.1

if [ "$NPROC" -gt 1 ]; then
  exec python -m torch.distributed.run --nnodes=1 --nproc-per-node $NPROC \
    --master-addr 127.0.0.1 --master-port ${MASTER_PORT:-29531} \
    pretrain_bert_mmap.py --strategy zero2 --precision bf16 "$@"
else
  exec python pretrain_bert_mmap.py --strategy zero2 --precision bf16 "$@"
fi
