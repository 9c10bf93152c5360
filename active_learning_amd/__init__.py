"""active_learning_amd — MI355X-native pool-based active-learning trainer.

A from-scratch re-design of the capabilities of ``zeyademam/active_learning``
(the reference implementation of "Active Learning at the ImageNet Scale",
arXiv:2111.12880) for AMD Instinct MI355X (gfx950, CDNA4):

* PyTorch-ROCm orchestration, hand-written HIP/CDNA4 kernels for every hot op
  (implicit-GEMM convolution on MFMA, fused BatchNorm+ReLU, fused SGD update,
  softmax/margin scoring, pairwise-distance + greedy k-center), exposed as a
  torch extension (``active_learning_amd.ops``).
* RCCL over xGMI for all collectives: a first-party bucketed-all-reduce DDP
  (``active_learning_amd.parallel``) with buckets sized for the 7-link xGMI
  topology and overlapped with backward.
* The same capability surface as the reference: CLI flags, arg pools, 13 query
  strategies, checkpoint formats, resume, debug mode.

Layout (reference counterpart in parentheses, file:line cites in modules):
  cli.py          argparse surface          (src/utils/parser.py)
  arg_pools/      experiment config dicts   (src/arg_pools/*)
  data/           index-returning datasets  (src/data_utils/*)
  models/         native NHWC ResNet-18/50  (src/models/*)
  ops/            HIP kernels + autograd    (implicit cuDNN/ATen kernels)
  parallel/       RCCL DDP + SyncBN         (torch DDP usage in strategy.py)
  strategies/     Strategy base + samplers  (src/query_strategies/*)
  utils/          eval/ckpt/logging/pool    (src/utils/*)
"""

__version__ = "0.1.0"
