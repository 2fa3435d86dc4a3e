#!/usr/bin/env bash
# Reproduce every measured BASELINE.md configuration in one run (1 GPU).
# Each line of the output file is the bench.py JSON record for one config.
# Usage: bash benchmarks/reproduce_all.sh [outfile]
set -u
cd "$(dirname "$0")/.."
OUT="${1:-gpurun_out/reproduce_all.jsonl}"
mkdir -p "$(dirname "$OUT")"
: > "$OUT"

run_cfg() {
  local name="$1"; shift
  echo "== $name: python bench.py $* ==" >&2
  local line
  line=$(timeout 420 python bench.py "$@" 2>/dev/null | tail -1)
  if [ -n "$line" ]; then
    printf '{"config_name": "%s", "record": %s}\n' "$name" "$line" >> "$OUT"
  else
    printf '{"config_name": "%s", "record": null}\n' "$name" >> "$OUT"
  fi
}

E="--eps-iters 0"
run_cfg headline_d1e6            --steps 20 --warmup 5            # full metric incl. iters-to-eps tail
run_cfg csr_uniform_d1e7         --csr --rows 1000000 --dim 10000000 --steps 20 --warmup 3 $E
run_cfg csr_zipf_d1e7            --csr --rows 1000000 --dim 10000000 --steps 20 --warmup 3 --csr-dist zipf $E
run_cfg gram_headline            --solver gram --steps 100 --warmup 10 $E
run_cfg gram_multiclass_k16      --solver gram --classes 16 --rows 16384 --steps 100 --warmup 10 $E
run_cfg multinomial_k16          --classes 16 --dim 10240 --rows 200000 --steps 15 --warmup 3 $E
run_cfg multinomial_k1000        --classes 1000 --dim 2048 --rows 100000 --steps 15 --warmup 3 $E
run_cfg csr_multinomial_k16      --csr --classes 16 --rows 1000000 --dim 10000000 --steps 10 --warmup 2 $E
run_cfg mixed_dense_csr          --mixed --rows 32768 --steps 10 --warmup 2 $E
run_cfg hinge_smooth_l2          --loss hinge --reg 1e-4 --steps 10 --warmup 2 $E
run_cfg fp8_d1e6                 --dtype f8 --steps 20 --warmup 3 $E
run_cfg capacity_92gb            --rows 49152 --steps 5 --warmup 1 $E
run_cfg dense_n10m_d10k          --rows 10000000 --dim 10240 --steps 5 --warmup 1 $E
run_cfg lsq_186gb                --loss lsq --rows 1000000 --dim 100096 --steps 3 --warmup 1 $E
echo "wrote $OUT" >&2
