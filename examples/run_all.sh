#!/bin/bash
# Run every CPU example (parity: reference examples/run_all.sh).
set -e
cd "$(dirname "$0")"
for f in perf_llama3_8b_tp1_pp2.py perf_llama3_8b_tp8.py \
         perf_llama3_8b_vpp2.py perf_llama3_70b_tp2_pp2.py \
         perf_llama3_70b_l12_selective_recompute.py \
         perf_mixtral_8x7b_ep8.py perf_deepseekv2_ep8.py \
         perf_deepseekv2_ep4_pp2.py simulator_trace_snapshot.py \
         show_simu_available_modes.py; do
  echo "=== $f ==="
  python3 "$f"
done
echo "=== search (slow) ==="
python3 search_strategy_llama3_8b.py
