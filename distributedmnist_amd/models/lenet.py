"""LeNet-style MNIST CNN, re-expressed from /root/reference/src/mnist.py.

Architecture (mnist.py:76-147): conv5x5 SAME x32 + bias + ReLU + maxpool2x2
-> conv5x5 SAME x64 + bias + ReLU + maxpool2x2 -> flatten(3136) -> fc 512 +
ReLU (+ dropout 0.5 train-only) -> fc 10.  Init: truncated_normal stddev 0.1
seed 66478 (mnist.py:32,81-101); conv1_b = 0, other biases = 0.1.

Layout is NHWC with HWIO conv weights (the TF layout the reference uses and
the coalescing-friendly layout for the CDNA4 kernels: C innermost).  On GPU
the module keeps bf16 shadow weights for MFMA compute; fp32 masters own the
gradient (see parallel/flatten.py).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import functional as Fx

IMAGE_SIZE = 28
NUM_CHANNELS = 1
NUM_LABELS = 10
SEED = 66478  # mnist.py:32


def trunc_normal_(t: torch.Tensor, std: float, gen: torch.Generator):
    """TF truncated_normal: resample outside 2 std (torch trunc_normal_ with
    bounds +-2*std matches)."""
    nn.init.trunc_normal_(t, mean=0.0, std=std, a=-2 * std, b=2 * std,
                          generator=gen)
    return t


class LeNet5(nn.Module):
    PARAM_ORDER = ["conv1_w", "conv1_b", "conv2_w", "conv2_b",
                   "fc1_w", "fc1_b", "fc2_w", "fc2_b"]
    # forward-GEMM B operands kept as pre-transposed bf16 copies
    # (conv1 uses the direct VALU kernel -> no transpose needed)
    TRANSPOSED_WEIGHTS = ["conv2_w", "fc1_w", "fc2_w"]

    def __init__(self, seed: int = SEED, compute_dtype: torch.dtype = torch.float32):
        super().__init__()
        gen = torch.Generator(device="cpu")
        gen.manual_seed(seed)
        self.compute_dtype = compute_dtype
        self.conv1_w = nn.Parameter(trunc_normal_(torch.empty(5, 5, NUM_CHANNELS, 32), 0.1, gen))
        self.conv1_b = nn.Parameter(torch.zeros(32))
        self.conv2_w = nn.Parameter(trunc_normal_(torch.empty(5, 5, 32, 64), 0.1, gen))
        self.conv2_b = nn.Parameter(torch.full((64,), 0.1))
        self.fc1_w = nn.Parameter(trunc_normal_(torch.empty(7 * 7 * 64, 512), 0.1, gen))
        self.fc1_b = nn.Parameter(torch.full((512,), 0.1))
        self.fc2_w = nn.Parameter(trunc_normal_(torch.empty(512, NUM_LABELS), 0.1, gen))
        self.fc2_b = nn.Parameter(torch.full((NUM_LABELS,), 0.1))
        # bf16 shadow weights (set up by parallel.flatten.FlatParams on GPU);
        # on CPU/fp32 they alias the masters.
        self.shadows: dict[str, torch.Tensor] = {}
        self.shadows_T: dict[str, torch.Tensor] = {}
        self.dropout_seed = seed
        self._step = 0
        self.training_direct_grads = True
        self._step_dev = None  # device step counter (hipGraph capture)

    def set_step_dev(self, t):
        """Device-resident step counter: dropout offsets under hipGraph."""
        self._step_dev = t

    def set_step(self, step: int):
        """Dropout offset — keeps masks deterministic per (seed, step).
        Clears any device-side counter: the host step is authoritative on
        the eager path (e.g. after a failed hipGraph capture)."""
        self._step = int(step)
        self._step_dev = None

    def _comp(self, name: str) -> torch.Tensor:
        return self.shadows.get(name, getattr(self, name))

    def _compT(self, name: str):
        return self.shadows_T.get(name)

    def _gout(self, name: str):
        """Direct-grad bucket view (GPU bf16 path): backward kernels
        accumulate into flat_grad without autograd add glue."""
        if not self.shadows or not self.training_direct_grads:
            return None
        g = getattr(self, name).grad
        return g if (g is not None and g.is_cuda) else None

    def forward(self, x: torch.Tensor, train: bool = True) -> torch.Tensor:
        """x: [B,28,28,1] (compute dtype) -> logits [B,10]."""
        h = Fx.conv_pool(x, self.conv1_w, self.conv1_b,
                         self._comp("conv1_w"), self.conv1_b, need_dx=False,
                         dw_out=self._gout("conv1_w"),
                         db_out=self._gout("conv1_b"))
        h = Fx.conv_pool(h, self.conv2_w, self.conv2_b,
                         self._comp("conv2_w"), self.conv2_b, need_dx=True,
                         dw_out=self._gout("conv2_w"),
                         db_out=self._gout("conv2_b"),
                         w_t=self._compT("conv2_w"))
        h = h.reshape(h.shape[0], 7 * 7 * 64)
        p_keep = 0.5 if train else 1.0
        h = Fx.linear_act(h, self.fc1_w, self.fc1_b,
                          self._comp("fc1_w"), self.fc1_b,
                          relu=True, p_keep=p_keep,
                          seed=self.dropout_seed, offset=self._step,
                          dw_out=self._gout("fc1_w"),
                          db_out=self._gout("fc1_b"),
                          offset_dev=self._step_dev,
                          w_t=self._compT("fc1_w"))
        logits = Fx.linear_act(h, self.fc2_w, self.fc2_b,
                               self._comp("fc2_w"), self.fc2_b, relu=False,
                               dw_out=self._gout("fc2_w"),
                               db_out=self._gout("fc2_b"),
                               w_t=self._compT("fc2_w"))
        return logits

    def predictions(self, logits):
        """Softmax class probabilities (reference mnist.py:166-167)."""
        return torch.softmax(logits.float(), dim=1)

    def loss_and_accuracy(self, logits, labels):
        """(mean CE loss, mean top-1 accuracy) — mnist.py:149-164."""
        loss, correct = Fx.softmax_xent(logits, labels)
        return loss, correct / logits.shape[0]
