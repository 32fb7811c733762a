"""2-layer MLP for MNIST — BASELINE.json config 1 (CPU/gloo plumbing check).

784 -> 512 ReLU -> 10, built from the same fused linear_act/softmax_xent
primitives as the CNN so the whole engine path is exercised without conv.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import functional as Fx
from .lenet import trunc_normal_

NUM_LABELS = 10


class MLP(nn.Module):
    PARAM_ORDER = ["fc1_w", "fc1_b", "fc2_w", "fc2_b"]
    TRANSPOSED_WEIGHTS = ["fc1_w", "fc2_w"]

    def __init__(self, seed: int = 66478, hidden: int = 512,
                 compute_dtype: torch.dtype = torch.float32):
        super().__init__()
        gen = torch.Generator(device="cpu")
        gen.manual_seed(seed)
        self.compute_dtype = compute_dtype
        self.fc1_w = nn.Parameter(trunc_normal_(torch.empty(784, hidden), 0.1, gen))
        self.fc1_b = nn.Parameter(torch.full((hidden,), 0.1))
        self.fc2_w = nn.Parameter(trunc_normal_(torch.empty(hidden, NUM_LABELS), 0.1, gen))
        self.fc2_b = nn.Parameter(torch.full((NUM_LABELS,), 0.1))
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
        self._step = int(step)
        self._step_dev = None  # host step authoritative on the eager path

    def _comp(self, name: str) -> torch.Tensor:
        return self.shadows.get(name, getattr(self, name))

    def _compT(self, name: str):
        return self.shadows_T.get(name)

    def _gout(self, name: str):
        if not self.shadows or not self.training_direct_grads:
            return None
        g = getattr(self, name).grad
        return g if (g is not None and g.is_cuda) else None

    def forward(self, x: torch.Tensor, train: bool = True) -> torch.Tensor:
        h = x.reshape(x.shape[0], -1)
        h = Fx.linear_act(h, self.fc1_w, self.fc1_b,
                          self._comp("fc1_w"), self.fc1_b, relu=True,
                          dw_out=self._gout("fc1_w"),
                          db_out=self._gout("fc1_b"),
                          w_t=self._compT("fc1_w"))
        return Fx.linear_act(h, self.fc2_w, self.fc2_b,
                             self._comp("fc2_w"), self.fc2_b, relu=False,
                             dw_out=self._gout("fc2_w"),
                             db_out=self._gout("fc2_b"),
                             w_t=self._compT("fc2_w"))

    def predictions(self, logits):
        """Softmax class probabilities (reference mnist.py:166-167)."""
        return torch.softmax(logits.float(), dim=1)

    def loss_and_accuracy(self, logits, labels):
        loss, correct = Fx.softmax_xent(logits, labels)
        return loss, correct / logits.shape[0]
