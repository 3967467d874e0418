#!/bin/bash
# Multi-node disaggregated PPO: trainer ranks on the leading nodes, rollout
# ranks on the trailing ones, RCCL over xGMI within a node and over the
# inter-node fabric across nodes.
#
# Run ON EVERY NODE (node_rank 0..NNODES-1, rank order = node order):
#   MASTER=10.0.0.1 NNODES=2 NODE_RANK=<i> bash run_multinode_disagg.sh
#
# The rollout ranks' HTTP engine servers bind 0.0.0.0 and each advertises
# the interface that routes to MASTER (override per node with
# POLYRL_ADVERTISE_ADDR); the address map is all-gathered over the world
# group before the role branch (trainer/main_stream.py), so trainer ranks
# reach remote engines without any static host list.  The weight path stays
# collective (bucketed broadcast over the world group — RCCL picks
# xGMI / inter-node transports per pair).
set -euo pipefail

MASTER=${MASTER:?set MASTER to node 0's IP}
NNODES=${NNODES:?set NNODES}
NODE_RANK=${NODE_RANK:?set NODE_RANK}
GPUS_PER_NODE=${GPUS_PER_NODE:-8}
# trailing ranks serve rollout; e.g. the whole last node:
NUM_ROLLOUT=${NUM_ROLLOUT:-$GPUS_PER_NODE}

export HSA_ENABLE_IPC_MODE_LEGACY=0

python -m torch.distributed.run \
    --nnodes="$NNODES" --node-rank="$NODE_RANK" \
    --nproc-per-node="$GPUS_PER_NODE" \
    --master-addr="$MASTER" --master-port=29500 \
    -m polyrl_amd.trainer.main_stream \
    actor_rollout_ref.model.path=llama3-8b \
    actor_rollout_ref.rollout.num_rollout_ranks="$NUM_ROLLOUT" \
    actor_rollout_ref.rollout.sampling.n=8 \
    actor_rollout_ref.rollout.response_length=1024 \
    algorithm.adv_estimator=grpo \
    data.train_batch_size=128 \
    reward=random \
    max_steps=50
