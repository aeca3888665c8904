#!/bin/bash
# Multi-node launcher for the chapter-5 trainer on 8x MI355X nodes.
#
# MI355X counterpart of /root/reference/05-training-llama-405b/launch.sh:1-37:
# ssh to every host in ./hosts, start a tmux session there, and run torchrun
# with this node's rank.  Env tuning is the RCCL/ROCm equivalent of the
# reference's NCCL lines (launch.sh:19-20) -- RCCL honors NCCL_* names:
#   NCCL_CROSS_NIC=1                  stripe inter-node traffic across NICs
#   TORCH_NCCL_AVOID_RECORD_STREAMS=1 fewer cached-stream allocations
#   HSA_ENABLE_IPC_MODE_LEGACY=0      dmabuf IPC (required on this driver)
#   OMP_NUM_THREADS                   CPU threads per rank (cpu-offload AdamW)
#
# Usage: bash launch.sh <experiment-name> [extra train_llm.py args...]
set -euo pipefail

EXPERIMENT=${1:?usage: launch.sh <experiment-name> [args...]}
shift || true
HOSTS_FILE=$(dirname "$0")/hosts
MASTER_ADDR=$(head -n 1 "$HOSTS_FILE")
MASTER_PORT=5001
NNODES=$(wc -l < "$HOSTS_FILE")
NPROC=8
REPO_DIR=$(cd "$(dirname "$0")/.." && pwd)

i=0
while read -r host; do
  echo "[launch] node $i: $host"
  ssh "$host" tmux new-session -d -s "dtga-$EXPERIMENT" \
    "cd $REPO_DIR && \
     HSA_ENABLE_IPC_MODE_LEGACY=0 \
     NCCL_CROSS_NIC=1 \
     TORCH_NCCL_AVOID_RECORD_STREAMS=1 \
     OMP_NUM_THREADS=\$(( \$(nproc) / $NPROC )) \
     python -m torch.distributed.run \
       --nnodes $NNODES --node-rank $i --nproc-per-node $NPROC \
       --master-addr $MASTER_ADDR --master-port $MASTER_PORT \
       --redirects 3 --log-dir ../logs/$EXPERIMENT \
       05-training-llama-405b/train_llm.py \
       -e $EXPERIMENT -m llama-3-405b -d synthetic \
       --cpu-offload --checkpoint-activations $*"
  i=$((i + 1))
done < "$HOSTS_FILE"
echo "[launch] started $NNODES nodes; monitor: python top-cluster.py $HOSTS_FILE"
