#!/bin/bash
# Deterministic-loss CI check (reference:
# .buildkite/scripts/benchmark_master.sh:79-110): runs the synthetic
# benchmark with --deterministic for every algorithm and asserts the
# final loss matches the recorded golden exactly (async is
# approximate-checked). Run on an MI355X box:
#
#   bash scripts/ci_determinism.sh [nproc]
#
# Goldens recorded at 1 GPU (scripts/ci_determinism_goldens_1gpu.txt).
set -u
NPROC=${1:-1}
CKFILE="$(dirname "$0")/ci_determinism_goldens_${NPROC}gpu.txt"
OUT=$(mktemp -d)
rc=0

run_algo() {
    local algo=$1
    local extra=${2:-}
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
        --standalone --local-addr 127.0.0.1 \
        examples/benchmark/synthetic_benchmark.py \
        --algorithm "$algo" --deterministic --num-iters 2 \
        --num-batches-per-iter 5 --num-warmup-batches 2 $extra \
        > "$OUT/$algo.log" 2>&1
    grep "Final loss" "$OUT/$algo.log" | tail -1 | awk '{print $3}'
}

declare -A LOSSES
for algo in gradient_allreduce bytegrad decentralized \
            low_precision_decentralized qadam; do
    loss=$(run_algo "$algo")
    LOSSES[$algo]=$loss
    echo "algorithm=$algo final_loss=$loss"
done
# async is wall-clock dependent; only check it runs and is finite
aloss=$(run_algo async "--async-sync-interval-ms 100 --async-warmup-steps 5")
echo "algorithm=async final_loss=$aloss (approximate)"
python - "$aloss" <<'EOF'
import math, sys
v = float(sys.argv[1])
assert math.isfinite(v), "async loss not finite"
EOF
rc=$((rc + $?))

if [[ -f "$CKFILE" ]]; then
    while read -r algo golden; do
        got=${LOSSES[$algo]:-missing}
        if [[ "$got" != "$golden" ]]; then
            echo "DETERMINISM FAIL: $algo got $got expected $golden"
            rc=1
        else
            echo "determinism ok: $algo $got"
        fi
    done < "$CKFILE"
else
    echo "no goldens for ${NPROC} GPU(s); record with:"
    for algo in "${!LOSSES[@]}"; do
        echo "  $algo ${LOSSES[$algo]}"
    done
fi
exit $rc
