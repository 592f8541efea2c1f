#!/bin/bash
# Launcher (reference parity: the mpirun wrapper scripts).
# MI355X-native launch model: one process per GPU via torchrun + RCCL.
#
#   dnn=resnet50 dataset=imagenet nworkers=8 density=0.001 ./dist.sh
#
dnn="${dnn:-resnet20}"
dataset="${dataset:-cifar10}"
nworkers="${nworkers:-1}"
density="${density:-0.001}"
compression="${compression:-gtopk}"
batch_size="${batch_size:-32}"
lr="${lr:-0.1}"
max_epochs="${max_epochs:-1}"
nsteps_update="${nsteps_update:-1}"
export HSA_ENABLE_IPC_MODE_LEGACY=0
cd "$(dirname "$0")/.."
if [ "$nworkers" -gt 1 ]; then
  exec python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node "$nworkers" --master-addr 127.0.0.1 \
    train.py --dnn "$dnn" --dataset "$dataset" --density "$density" \
    --compression "$compression" --batch-size "$batch_size" --lr "$lr" \
    --max-epochs "$max_epochs" --nsteps-update "$nsteps_update" "$@"
else
  exec python train.py --dnn "$dnn" --dataset "$dataset" \
    --density "$density" --compression "$compression" \
    --batch-size "$batch_size" --lr "$lr" --max-epochs "$max_epochs" \
    --nsteps-update "$nsteps_update" "$@"
fi
