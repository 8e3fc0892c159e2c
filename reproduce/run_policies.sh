#!/bin/bash
# Reproduce the headline policy comparison (reference: reproduce/tacc_32gpus.sh).
# Runs the 7 paper policies on the 120-job TACC-style trace over 32 simulated
# MI355X GPUs and aggregates makespan / JCT / finish-time fairness.
set -e
cd "$(dirname "$0")/.."

TRACE=${TRACE:-traces/tacc_like_120.trace}
NGPUS=${NGPUS:-32}
RESULTS=${RESULTS:-results/$(basename "$TRACE" .trace)_${NGPUS}gpus}
ROUND=${ROUND:-120}

for policy in shockwave min_total_duration finish_time_fairness \
              max_min_fairness allox max_sum_throughput_perf gandiva_fair; do
    echo "=== $policy ==="
    python scripts/simulate.py -t "$TRACE" -p "$policy" -n "$NGPUS" \
        --time_per_iteration "$ROUND" --results_dir "$RESULTS" \
        -c configs/mi355x_32gpus.json
done

python reproduce/aggregate_result.py --results_dir "$RESULTS"
python scripts/plotting.py --results_dir "$RESULTS"
