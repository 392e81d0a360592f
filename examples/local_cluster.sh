#!/usr/bin/env bash
# End-to-end walkthrough: two miners + validator + averager as separate OS
# processes over the file store (the reference's deployment shape with the
# HF hub replaced by a shared directory). Runs on CPU in under a minute;
# the same commands on an MI355X box run the full HIP kernel stack.
set -e
cd "$(dirname "$0")/.."
ROOT=$(mktemp -d /tmp/dta_cluster.XXXX)
COMMON=(--tiny --comm.root "$ROOT" --metrics-dir "$ROOT/metrics"
        --train.batch-size 4 --train.seq-len 32
        --validate.batch-size 4 --validate.seq-len 32
        --validate.n-eval-batches 3)

echo "== two miners train from the shared base and push weight deltas =="
python -m distributedtraining_amd.cli miner --hotkey m0 --steps 30 "${COMMON[@]}"
python -m distributedtraining_amd.cli miner --hotkey m1 --steps 30 "${COMMON[@]}"

echo "== validator scores each delta by held-out loss improvement =="
python -m distributedtraining_amd.cli validator --rounds 1 "${COMMON[@]}"

echo "== averager merges the deltas (meta-learned weights) into a new base =="
python -m distributedtraining_amd.cli averager --rounds 1 \
    --average.strategy mean "${COMMON[@]}"

echo "== miner m0 resumes and picks up the new base =="
python -m distributedtraining_amd.cli miner --hotkey m0 --steps 10 --resume "${COMMON[@]}"

echo "== artifacts =="
find "$ROOT" -type f | sed "s|$ROOT|.|"
echo "metrics sample:"; head -2 "$ROOT"/metrics/miner_m0.jsonl
rm -rf "$ROOT"
echo "done"
