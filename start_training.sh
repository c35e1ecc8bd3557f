#!/usr/bin/env bash
# Training launcher — KEY=VALUE args, per-dataset config pick, torchrun exec
# (ref start_training.sh:11-31,46-75). One process per GPU over RCCL.
#
#   bash start_training.sh DATASET=llff NUM_GPUS=8 WORKSPACE=/ws VERSION=v1 \
#       [MASTER_ADDR=127.0.0.1] [MASTER_PORT=29500] [NNODES=1] [NODE_RANK=0] \
#       [EXTRA_CONFIG='{"key": val}']
set -euo pipefail

MASTER_ADDR=127.0.0.1
MASTER_PORT=29500
NNODES=1
NODE_RANK=0
NUM_GPUS=1
DATASET=realestate10k
WORKSPACE=./workspace
VERSION=debug
EXTRA_CONFIG='{}'

for arg in "$@"; do
  key="${arg%%=*}"
  val="${arg#*=}"
  case "$key" in
    MASTER_ADDR|MASTER_PORT|NNODES|NODE_RANK|NUM_GPUS|DATASET|WORKSPACE|VERSION|EXTRA_CONFIG)
      printf -v "$key" '%s' "$val" ;;
    *) echo "unknown argument: $key" >&2; exit 1 ;;
  esac
done

case "$DATASET" in
  llff)      CONFIG=configs/params_llff.yaml ;;
  flowers)   CONFIG=configs/params_flowers.yaml ;;
  kitti_raw) CONFIG=configs/params_kitti_raw.yaml ;;
  dtu)       CONFIG=configs/params_dtu.yaml ;;
  *)         CONFIG=configs/params_realestate.yaml ;;
esac

export HSA_ENABLE_IPC_MODE_LEGACY=0

exec python3 -m torch.distributed.run \
  --nnodes "$NNODES" --node-rank "$NODE_RANK" \
  --nproc-per-node "$NUM_GPUS" \
  --master-addr "$MASTER_ADDR" --master-port "$MASTER_PORT" \
  train.py --config_path "$CONFIG" \
  --workspace "$WORKSPACE" --version "$VERSION" \
  --extra_config "$EXTRA_CONFIG"
