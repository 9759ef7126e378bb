#!/usr/bin/env bash
# The reference's 9 canned experiments (reference src/runner.sh:12-38),
# adapted to the rlr_amd CLI.  Each triple: no-attack / attack / attack+RLR.
# --synthetic uses the deterministic on-box generator (no dataset downloads);
# drop it if real torchvision data is present under ../data.
set -e
PY="python -m rlr_amd.federated"
SYN="--synthetic"

# ---------------- FMNIST: 10 agents, bs 256, 2 local epochs, 200 rounds ----
$PY --data fmnist --num_agents 10 --rounds 200 --snap 5 $SYN
$PY --data fmnist --num_agents 10 --rounds 200 --snap 5 --num_corrupt 1 \
    --poison_frac 0.5 $SYN
$PY --data fmnist --num_agents 10 --rounds 200 --snap 5 --num_corrupt 1 \
    --poison_frac 0.5 --robustLR_threshold 4 $SYN

# ---------------- CIFAR10: 40 agents, 4 corrupt (DBA plus-pattern), θ=8 ----
$PY --data cifar10 --num_agents 40 --rounds 200 --snap 5 $SYN
$PY --data cifar10 --num_agents 40 --rounds 200 --snap 5 --num_corrupt 4 \
    --poison_frac 0.5 $SYN
$PY --data cifar10 --num_agents 40 --rounds 200 --snap 5 --num_corrupt 4 \
    --poison_frac 0.5 --robustLR_threshold 8 $SYN

# -------- Fed-EMNIST: 3383 writers, 1% sampled/round, 338 corrupt, θ=8 ----
$PY --data fedemnist --num_agents 3383 --agent_frac 0.01 --rounds 500 \
    --snap 5 --local_ep 10 --bs 64 $SYN
$PY --data fedemnist --num_agents 3383 --agent_frac 0.01 --rounds 500 \
    --snap 5 --local_ep 10 --bs 64 --num_corrupt 338 --poison_frac 0.5 $SYN
$PY --data fedemnist --num_agents 3383 --agent_frac 0.01 --rounds 500 \
    --snap 5 --local_ep 10 --bs 64 --num_corrupt 338 --poison_frac 0.5 \
    --robustLR_threshold 8 $SYN
