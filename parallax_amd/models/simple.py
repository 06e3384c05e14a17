"""The 'simple' example: 2->1 linear regression with GradientDescent.

Reference: examples/simple/simple_driver.py (loss-decreases smoke test —
the reference's only end-to-end sanity example).
"""

import torch
import torch.nn as nn

from parallax_amd import optim
from parallax_amd.session import TrainGraph


class SimpleLinear(nn.Module):
    def __init__(self):
        super().__init__()
        self.w = nn.Parameter(torch.zeros(2, 1))
        self.b = nn.Parameter(torch.zeros(1))

    def forward(self, x):
        return x @ self.w + self.b


def build_simple_graph(lr=0.01):
    model = SimpleLinear()
    optimizer = optim.SGD(model.parameters(), lr=lr)

    def step_fn(m, batch):
        x, y = batch
        pred = m(x)
        loss = 0.5 * ((pred - y) ** 2).sum()
        return loss, {}

    return TrainGraph(model, optimizer, step_fn)
