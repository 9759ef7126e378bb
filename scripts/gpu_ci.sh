#!/usr/bin/env bash
# GPU-side CI batch, run via: gpurun -- 'bash scripts/gpu_ci.sh [what]'
set -x
mkdir -p gpurun_out
WHAT="${1:-all}"

if [ "$WHAT" = "bf16" ] || [ "$WHAT" = "all" ]; then
  timeout 600 python -m pytest tests/test_bf16_gpu.py tests/test_bf16_conv_gpu.py -q \
      > gpurun_out/ci_bf16.log 2>&1
  echo "bf16 rc=$?" >> gpurun_out/ci_bf16.log
  tail -12 gpurun_out/ci_bf16.log
fi

if [ "$WHAT" = "suite" ] || [ "$WHAT" = "all" ]; then
  timeout 700 python -m pytest tests/ -m gpu -q > gpurun_out/ci_suite.log 2>&1
  echo "suite rc=$?" >> gpurun_out/ci_suite.log
  tail -6 gpurun_out/ci_suite.log
fi

if [ "$WHAT" = "bench" ] || [ "$WHAT" = "all" ]; then
  timeout 300 python bench.py --steps 5 --warmup 2 > gpurun_out/ci_bench.log 2>&1
  grep -o '"value": [0-9.]*' gpurun_out/ci_bench.log
fi

if [ "$WHAT" = "benchr" ] || [ "$WHAT" = "all" ]; then
  timeout 300 python bench.py --data cifar10 --model resnet18 --dtype bf16 \
      --steps 3 --warmup 1 > gpurun_out/ci_benchr.log 2>&1
  echo "benchr rc=$?"
  grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/ci_benchr.log
  tail -2 gpurun_out/ci_benchr.log | head -1
fi

if [ "$WHAT" = "sweep" ]; then
  for R in 2 4 6 8; do
    timeout 200 python bench.py --steps 4 --warmup 2 --agents_per_stream_override $R \
      > gpurun_out/sweep_$R.log 2>&1 || true
  done
  grep -H -o '"value": [0-9.]*' gpurun_out/sweep_*.log
fi

if [ "$WHAT" = "fedemnist" ]; then
  timeout 900 python -m rlr_amd.federated --data fedemnist --num_agents 3383 \
    --agent_frac 0.01 --num_corrupt 338 --poison_frac 0.5 \
    --robustLR_threshold 8 --local_ep 10 --bs 64 --rounds 5 --snap 5 \
    --pattern_type square --synthetic --no_tb > gpurun_out/ci_fed.log 2>&1
  echo "fedemnist rc=$?"
  tail -8 gpurun_out/ci_fed.log
fi

if [ "$WHAT" = "final" ]; then
  timeout 400 python bench.py --steps 20 --warmup 5 > gpurun_out/final_bench.log 2>&1
  grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/final_bench.log
  cd /tmp && export TMPDIR=/tmp
  timeout 400 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/proff -o bf \
    -- python $GRAFT_REPO_ROOT/bench.py --steps 3 --warmup 2 > /dev/null 2>&1
  echo "profile rc=$?"
fi

if [ "$WHAT" = "dba" ]; then
  # BASELINE config 4 semantics on ONE GPU (the driver scales it to 8):
  # CIFAR10, 40 agents, 4 corrupt with the DBA plus-pattern split, RLR 8
  timeout 600 python -m rlr_amd.federated --data cifar10 --num_agents 40 \
    --num_corrupt 4 --poison_frac 0.5 --pattern_type plus \
    --robustLR_threshold 8 --rounds 3 --snap 3 --synthetic --no_tb \
    > gpurun_out/ci_dba.log 2>&1
  echo "dba rc=$?"
  tail -5 gpurun_out/ci_dba.log
fi

if [ "$WHAT" = "profr" ]; then
  # kernel-time breakdown of the bf16 ResNet18 path (BASELINE config 3) —
  # evidence for the round-2 optimization roadmap
  cd /tmp && export TMPDIR=/tmp
  timeout 500 rocprofv3 --kernel-trace --stats -d $GRAFT_REPO_ROOT/gpurun_out/profr -o rn \
    -- python $GRAFT_REPO_ROOT/bench.py --data cifar10 --model resnet18 \
    --dtype bf16 --steps 2 --warmup 1 > $GRAFT_REPO_ROOT/gpurun_out/profr.log 2>&1
  echo "profr rc=$?"
  grep -o '"ms_per_step": [0-9.]*' $GRAFT_REPO_ROOT/gpurun_out/profr.log
fi

if [ "$WHAT" = "dbacurves" ]; then
  # BASELINE config 4 semantics at full reference scale (runner.sh:23-28):
  # CIFAR10, 40 agents, 4 corrupt, DBA plus-pattern, 200 rounds
  for CFG in "atk:--num_corrupt 4 --poison_frac 0.5" \
             "rlr:--num_corrupt 4 --poison_frac 0.5 --robustLR_threshold 8"; do
    TAG="${CFG%%:*}"; FLAGS="${CFG#*:}"
    timeout 700 python -m rlr_amd.federated --data cifar10 --num_agents 40 \
      --rounds 200 --snap 50 --pattern_type plus --synthetic --no_tb $FLAGS \
      > gpurun_out/dba_$TAG.log 2>&1
    echo "== $TAG rc=$?"
    grep -E "Val_Loss|Poison Loss" gpurun_out/dba_$TAG.log | tail -2
  done
fi

if [ "$WHAT" = "pmc" ]; then
  cd /tmp && export TMPDIR=/tmp
  timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT SQ_INSTS_VALU SQ_INSTS_MFMA SQ_INSTS_LDS \
    -d $GRAFT_REPO_ROOT/gpurun_out/pmc3 -o p3 --output-format csv \
    -- python $GRAFT_REPO_ROOT/tests/perf/conv_micro.py all 15 \
    > $GRAFT_REPO_ROOT/gpurun_out/pmc3.log 2>&1
  echo "pmc rc=$?"
fi

if [ "$WHAT" = "curves" ]; then
  # the reference's FMNIST triple at full scale (runner.sh:12-18):
  # no-attack / attack / attack+RLR(theta=4), 10 agents, 200 rounds
  for CFG in "na:--num_corrupt 0 --poison_frac 0" \
             "atk:--num_corrupt 1 --poison_frac 0.5" \
             "rlr:--num_corrupt 1 --poison_frac 0.5 --robustLR_threshold 4"; do
    TAG="${CFG%%:*}"; FLAGS="${CFG#*:}"
    timeout 500 python -m rlr_amd.federated --data fmnist --num_agents 10 \
      --rounds 200 --snap 50 --synthetic --no_tb $FLAGS \
      > gpurun_out/curve_$TAG.log 2>&1
    echo "== $TAG rc=$?"
    grep -E "Val_Loss|Poison Loss" gpurun_out/curve_$TAG.log | tail -4
  done
fi

if [ "$WHAT" = "san" ]; then
  # sanitizer-class pass: the full GPU kernel suite with
  # serialized kernel launches + blocking copies, so any async fault
  # (OOB, bad address) surfaces at the exact offending launch instead of
  # a later sync point.  Kept log: gpurun_out/ci_san.log
  AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3 HIP_LAUNCH_BLOCKING=1 \
    timeout 900 python -m pytest tests/test_kernels_gpu.py \
    tests/test_bf16_gpu.py tests/test_bf16_conv_gpu.py -q \
    > gpurun_out/ci_san.log 2>&1
  echo "san rc=$?" >> gpurun_out/ci_san.log
  tail -4 gpurun_out/ci_san.log
fi

if [ "$WHAT" = "defense" ]; then
  # the full-scale FMNIST defense curves as an
  # ASSERTED CI target with tight bands (attack >=0.95 backdoor without
  # defense, <=0.05 with RLR theta=4; val accs within 0.05), written to
  # gpurun_out/defense_ci.json for tracking under profiles/.
  timeout 1500 python scripts/defense_ci.py > gpurun_out/defense_ci.log 2>&1
  echo "defense rc=$?"
  tail -4 gpurun_out/defense_ci.log
fi
