#!/bin/bash
# Multi-node launcher for multinode_torchrun.py (parity with reference
# slurm/sbatch_run.sh, with its line-19 bug fixed: no trailing comment
# after a backslash continuation).

#SBATCH --job-name=mi355x-ddp-multinode
#SBATCH --nodes=4
#SBATCH --ntasks-per-node=1
#SBATCH --gpus-per-task=1
#SBATCH --cpus-per-task=8

# head-node IP discovery (reference slurm/sbatch_run.sh:9-12)
nodes=$(scontrol show hostnames "$SLURM_JOB_NODELIST")
nodes_array=($nodes)
head_node=${nodes_array[0]}
head_node_ip=$(srun --nodes=1 --ntasks=1 -w "$head_node" hostname --ip-address)

echo "Node IP: $head_node_ip"
export LOGLEVEL=INFO
# dmabuf IPC is the only mode the driver supports on these nodes
export HSA_ENABLE_IPC_MODE_LEGACY=0

srun torchrun \
  --nnodes 4 \
  --nproc_per_node 1 \
  --rdzv_id "$RANDOM" \
  --rdzv_backend c10d \
  --rdzv_endpoint "$head_node_ip:29500" \
  /shared/mi355x-ddp/multinode_torchrun.py 50 10
