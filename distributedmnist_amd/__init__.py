"""distributedmnist_amd — MI355X-native synchronous data-parallel MNIST training.

A from-scratch re-expression of agnusmaximus/DistributedMNIST for a single
8xMI355X node: PyTorch-ROCm as the tensor substrate, hand-written HIP/CDNA4
kernels for every hot-path op (conv, GEMM, pool, softmax-CE, fused SGD), and
RCCL over xGMI for gradient aggregation (replacing the reference's TF gRPC
parameter-server runtime, /root/reference/src/distributed_train.py).

Layer map:
  ops/       HIP kernel bindings + CPU fp32 reference implementations
  models/    LeNet-5 CNN (reference src/mnist.py) and a 2-layer MLP
  data/      MNIST idx-gz loader, synthetic/fake data, per-rank sharding
  parallel/  sync engine: flat-bucket RCCL all-reduce, K-of-N, interval, CDF
  engine/    train loop, checkpoint supervisor, evaluator
  utils/     flag registry (reference-compatible), logging
"""

__version__ = "0.1.0"
