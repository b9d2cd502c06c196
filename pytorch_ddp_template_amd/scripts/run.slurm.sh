#!/bin/bash
# Per-node Slurm task — equivalent of reference run.slurm.sh:1-8.
set -e
NGPU_PER_NODE=${NGPU_PER_NODE:-8}
exec python -m pytorch_ddp_template_amd.launch \
    --nproc_per_node "$NGPU_PER_NODE" \
    --nnodes "$SLURM_JOB_NUM_NODES" --node_rank "$SLURM_NODEID" \
    --master_addr "$MASTER_ADDR" --master_port "$MASTER_PORT" \
    -m pytorch_ddp_template_amd.ddp $SCRIPT_ARGS
