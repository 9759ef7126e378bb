"""rlr_amd — an MI355X-native federated-learning simulator with the Robust
Learning Rate (RLR) backdoor defense.

Re-implements, MI355X-first, the capabilities of the AAAI-2021 reference
`TinfoilHat0/Defending-Against-Backdoors-with-Robust-Learning-Rate`
(see SURVEY.md): federated training of K agents under pixel-pattern trojan
attacks, with FedAvg / coordinate-median / sign aggregation modulated by the
per-parameter robust-learning-rate sign vote.

Architecture (nothing here is a port):
  * one process per GPU; sampled agents are sharded contiguously across
    ranks; per-agent update vectors are all-gathered over RCCL/xGMI and the
    aggregation runs redundantly (deterministically) on every rank
    (reference: a single-process Python dict, federated.py:67-74).
  * model parameters / grads / momentum live in flat device buffers (module
    params are views), so pack/unpack (reference agent.py:35,63) is free and
    the clip+SGD step and PGD projection are fused HIP kernels.
  * the hot ops (conv/linear fwd+bwd, relu/pool/dropout, CE loss, the
    aggregation rules, trojan injection, eval reductions) are hand-written
    HIP/CDNA4 kernels in rlr_amd/ops/csrc, dispatched on CUDA(HIP) tensors;
    the plain-PyTorch path exists only for GPU-less CI.
"""

__version__ = "0.1.0"
