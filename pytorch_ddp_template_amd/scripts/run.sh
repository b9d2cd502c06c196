#!/bin/bash
# Single-node launcher — equivalent of reference run.sh:1-11, using the
# framework's own launcher instead of the deprecated torch.distributed.launch.
# One process per GPU over RCCL/xGMI.
set -e
GPUS=${GPUS:-0,1,2,3,4,5,6,7}
NGPU_PER_NODE=${NGPU_PER_NODE:-8}
NNODE=${NNODE:-1}
NODE_RANK=${NODE_RANK:-0}
MASTER_ADDR=${MASTER_ADDR:-127.0.0.1}
MASTER_PORT=${MASTER_PORT:-9315}
MAIN_SCRIPT=${MAIN_SCRIPT:-pytorch_ddp_template_amd.ddp}
SCRIPT_ARGS=${SCRIPT_ARGS:-}

HIP_VISIBLE_DEVICES=$GPUS HSA_ENABLE_IPC_MODE_LEGACY=0 \
python -m pytorch_ddp_template_amd.launch \
    --nproc_per_node "$NGPU_PER_NODE" --nnodes "$NNODE" --node_rank "$NODE_RANK" \
    --master_addr "$MASTER_ADDR" --master_port "$MASTER_PORT" \
    -m "$MAIN_SCRIPT" $SCRIPT_ARGS
