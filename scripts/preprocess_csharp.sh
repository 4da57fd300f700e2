#!/usr/bin/env bash
###########################################################
# C# dataset preprocessing driver — the equivalent of the reference's
# preprocess_csharp.sh (extract -> histograms -> truncate/pad/pickle) using
# the native C++ C# extractor (extractor/c2v-extract-cs, Roslyn-kind trees)
# instead of `dotnet run`, and code2vec_amd.data.preprocess instead of the
# awk histogram pipeline.
#
# TRAIN_DIR / VAL_DIR / TEST_DIR: directories of .cs files (recursive).
###########################################################
set -euo pipefail

TRAIN_DIR=${TRAIN_DIR:-my_train_dir}
VAL_DIR=${VAL_DIR:-my_val_dir}
TEST_DIR=${TEST_DIR:-my_test_dir}
DATASET_NAME=${DATASET_NAME:-my_cs_dataset}
MAX_CONTEXTS=${MAX_CONTEXTS:-200}
MAX_EXTRACT_CONTEXTS=${MAX_EXTRACT_CONTEXTS:-30000}  # reservoir cap (Extractor.cs)
WORD_VOCAB_SIZE=${WORD_VOCAB_SIZE:-1301136}
PATH_VOCAB_SIZE=${PATH_VOCAB_SIZE:-911417}
TARGET_VOCAB_SIZE=${TARGET_VOCAB_SIZE:-261245}
NUM_THREADS=${NUM_THREADS:-64}
PYTHON=${PYTHON:-python3}

HERE="$(cd "$(dirname "$0")/.." && pwd)"
EXTRACTOR="${HERE}/extractor/c2v-extract-cs"
[ -x "${EXTRACTOR}" ] || make -C "${HERE}/extractor" c2v-extract-cs

TRAIN_DATA_FILE=${DATASET_NAME}.train.raw.txt
VAL_DATA_FILE=${DATASET_NAME}.val.raw.txt
TEST_DATA_FILE=${DATASET_NAME}.test.raw.txt

mkdir -p data/${DATASET_NAME}

echo "Extracting paths from validation set..."
"${EXTRACTOR}" --path "${VAL_DIR}" --max_length 8 --max_width 2 \
  --max_contexts "${MAX_EXTRACT_CONTEXTS}" --threads "${NUM_THREADS}" \
  --ofile_name "${VAL_DATA_FILE}"
echo "Extracting paths from test set..."
"${EXTRACTOR}" --path "${TEST_DIR}" --max_length 8 --max_width 2 \
  --max_contexts "${MAX_EXTRACT_CONTEXTS}" --threads "${NUM_THREADS}" \
  --ofile_name "${TEST_DATA_FILE}"
echo "Extracting paths from training set..."
"${EXTRACTOR}" --path "${TRAIN_DIR}" --max_length 8 --max_width 2 \
  --max_contexts "${MAX_EXTRACT_CONTEXTS}" --threads "${NUM_THREADS}" \
  --ofile_name "${TRAIN_DATA_FILE}.unshuf"
shuf "${TRAIN_DATA_FILE}.unshuf" > "${TRAIN_DATA_FILE}"
rm -f "${TRAIN_DATA_FILE}.unshuf"

echo "Preprocessing (histograms + truncate/pad + dictionaries)..."
PYTHONPATH="${HERE}" ${PYTHON} -m code2vec_amd.data.preprocess \
  --train_data "${TRAIN_DATA_FILE}" --test_data "${TEST_DATA_FILE}" \
  --val_data "${VAL_DATA_FILE}" --max_contexts "${MAX_CONTEXTS}" \
  --word_vocab_size "${WORD_VOCAB_SIZE}" --path_vocab_size "${PATH_VOCAB_SIZE}" \
  --target_vocab_size "${TARGET_VOCAB_SIZE}" \
  --output_name data/${DATASET_NAME}/${DATASET_NAME}

rm -f "${TRAIN_DATA_FILE}" "${VAL_DATA_FILE}" "${TEST_DATA_FILE}"
echo "Done: data/${DATASET_NAME}/${DATASET_NAME}.{train,val,test}.c2v + .dict.c2v"
