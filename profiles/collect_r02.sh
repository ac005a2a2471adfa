#!/bin/bash
# Round-2 evidence collection (run on an MI355X box via gpurun).
# Produces kernel-trace stats + PMC traffic for the headline workloads and
# a full bench battery into gpurun_out/prof_r02/.
set -u
cd /root/repo
export TMPDIR=/tmp
OUT=gpurun_out/prof_r02
mkdir -p "$OUT"

trace() {  # name, bench args...
  local name=$1; shift
  (cd /tmp && rocprofv3 --kernel-trace --stats --output-format csv \
      -d "/root/repo/$OUT/$name" -- \
      python /root/repo/bench.py "$@" --no-cpu-baseline) \
      > "$OUT/$name.log" 2>&1
}
pmc() {  # name, counter, bench args...
  local name=$1 ctr=$2; shift 2
  (cd /tmp && rocprofv3 --pmc "$ctr" --output-format csv \
      -d "/root/repo/$OUT/${name}_$ctr" -- \
      python /root/repo/bench.py "$@" --no-cpu-baseline) \
      > "$OUT/${name}_$ctr.log" 2>&1
}

trace q6_sf100 --workload tpch_q6_lineitem_sf100 --steps 3 --warmup 1
trace q1_sf100 --workload tpch_q1_lineitem_sf100 --steps 3 --warmup 1
trace sparse   --workload sparse_group_sf10 --steps 3 --warmup 1
trace star     --workload star_join_sf10 --steps 5 --warmup 2

pmc q6_sf100 FETCH_SIZE --workload tpch_q6_lineitem_sf100 --steps 2 --warmup 1
pmc q6_sf100 WRITE_SIZE --workload tpch_q6_lineitem_sf100 --steps 2 --warmup 1
pmc q1_sf100 FETCH_SIZE --workload tpch_q1_lineitem_sf100 --steps 2 --warmup 1
pmc q1_sf100 WRITE_SIZE --workload tpch_q1_lineitem_sf100 --steps 2 --warmup 1
pmc sparse   FETCH_SIZE --workload sparse_group_sf10 --steps 2 --warmup 1
pmc sparse   WRITE_SIZE --workload sparse_group_sf10 --steps 2 --warmup 1

# bench battery (one JSON line each, CPU baselines on)
B=$OUT/bench_r02_final.jsonl
: > "$B"
for W in tpch_q6_lineitem_sf100 tpch_q1_lineitem_sf100 tpch_q6_lineitem_sf10 \
         tpch_q1_lineitem_sf10 tpch_q6_mut_sf10 star_join_sf10 \
         config1_sum_where sparse_group_sf10; do
  timeout 900 python bench.py --workload "$W" --steps 10 --warmup 3 \
      >> "$B" 2> "$OUT/bench_$W.err"
done
echo DONE
