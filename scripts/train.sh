#!/usr/bin/env bash
# Training entry — the reference train.sh equivalent. Single GPU:
#   scripts/train.sh
# Data-parallel over N GPUs of one node (RCCL over xGMI):
#   NPROC=8 scripts/train.sh
set -euo pipefail

type=${TYPE:-java14m}
dataset_name=${DATASET_NAME:-my_dataset}
data_dir=${DATA_DIR:-data/${dataset_name}}
data=${data_dir}/${dataset_name}
test_data=${data_dir}/${dataset_name}.val.c2v
model_dir=${MODEL_DIR:-models/${type}}
NPROC=${NPROC:-1}

HERE="$(cd "$(dirname "$0")/.." && pwd)"
mkdir -p "${model_dir}"

if [ "${NPROC}" -gt 1 ]; then
  PYTHONPATH="${HERE}" python3 -m torch.distributed.run --nnodes=1 \
    --nproc-per-node "${NPROC}" --master-addr 127.0.0.1 --master-port 29517 \
    "${HERE}/code2vec.py" --data "${data}" --test "${test_data}" \
    --save "${model_dir}/saved_model" "$@"
else
  PYTHONPATH="${HERE}" python3 "${HERE}/code2vec.py" --data "${data}" \
    --test "${test_data}" --save "${model_dir}/saved_model" "$@"
fi
