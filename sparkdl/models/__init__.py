"""sparkdl.models — model families for the framework's benchmark configs.

BASELINE.json configs: MNIST-shaped MLP (CPU plumbing), ResNet-50 bf16
(flagship images/sec), BERT-base bf16 seq-512 (sequences/sec).  All
models are defined here from scratch against torch.nn plus sparkdl.ops'
hand-written HIP kernels for the hot ops.
"""

from sparkdl.models.mlp import MnistMLP  # noqa: F401
