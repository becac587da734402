#!/usr/bin/env bash
# Regenerate tunables/<arch>.csv (TunableOp GEMM solution table) on the
# current GPU box. Run after a ROCm/hipBLASLt upgrade or to add a model's
# shapes; takes ~10-20 min of GPU time. The serve passes run the engine
# EAGER (graph capture cannot tune) with tiny step counts — only the GEMM
# shapes matter, not the throughput printed.
#
# Usage: bash scripts/tune_gemms.sh [out.csv]
set -euo pipefail
cd "$(dirname "$0")/.."

ARCH=$(python -c "import torch; print(torch.cuda.get_device_properties(0).gcnArchName.split(':')[0])")
OUT=${1:-substratus_amd/data/tunables/${ARCH}.csv}
WORK=$(mktemp -d)/tune.csv

export PYTORCH_TUNABLEOP_ENABLED=1
export PYTORCH_TUNABLEOP_FILENAME="$WORK"
# seed with the existing table so only new shapes tune
[ -f "$OUT" ] && cp "$OUT" "${WORK%.*}0.csv"

tune_serve() {  # model batch prompt prefill_chunk
    SUB_EAGER_DECODE=1 timeout 600 python bench.py --model "$1" \
        --batch "$2" --prompt-len "$3" --steps 3 --warmup 1 \
        ${4:+--prefill-chunk "$4"} > /dev/null 2>&1 \
        && echo "tuned $1 b=$2" || echo "WARN: $1 tuning pass failed"
}

tune_serve llama2-7b 512 64
tune_serve llama2-7b 256 64
tune_serve llama2-13b 256 64
tune_serve falcon-7b-instruct 256 64
tune_serve falcon-40b 256 64 16384
tune_serve llama2-70b 512 64 32768
# training shapes (fwd+bwd); generous timeout — big-M tuning is slow
timeout 1200 python bench.py --mode train --steps 2 --warmup 1 \
    > /dev/null 2>&1 && echo "tuned train" || echo "WARN: train pass partial"

# TunableOp appends a device ordinal before the extension
mkdir -p "$(dirname "$OUT")"
cp "${WORK%.*}0.csv" "$OUT"
echo "wrote $OUT ($(grep -c Gemm "$OUT" || true) entries)"
