"""Distributed PGPE + Adam on an MNIST30K-style convnet (~28k params) —
the reference's Training_MNIST30K notebook configuration (BASELINE.md
row 4), with synthetic MNIST-shaped data (no dataset downloads offline).

Single process:  python examples/mnist30k_distributed.py
Multi-GPU:       python -m torch.distributed.run --nnodes=1 \
                     --nproc-per-node 4 --master-addr 127.0.0.1 \
                     examples/mnist30k_distributed.py
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import argparse
import os

import torch
from torch import nn

from evotorch_amd.algorithms import PGPE
from evotorch_amd.logging import StdOutLogger
from evotorch_amd.neuroevolution import SupervisedNE
from evotorch_amd.parallel import init_comm


class MNIST30K(nn.Module):
    """~28k-parameter convnet (the paper's MNIST30K architecture shape)."""

    def __init__(self):
        super().__init__()
        self.net = nn.Sequential(
            nn.Conv2d(1, 16, kernel_size=5, stride=2, padding=2),  # 28->14
            nn.Tanh(),
            nn.Conv2d(16, 16, kernel_size=5, stride=2, padding=2),  # 14->7
            nn.Tanh(),
            nn.Flatten(),
            nn.Linear(16 * 7 * 7, 32),
            nn.Tanh(),
            nn.Linear(32, 10),
        )

    def forward(self, x):
        return self.net(x)


def make_synthetic_mnist(n=8192, seed=0):
    g = torch.Generator().manual_seed(seed)
    # class-conditional blobs in image space: learnable but offline
    prototypes = torch.randn(10, 1, 28, 28, generator=g)
    labels = torch.randint(0, 10, (n,), generator=g)
    images = prototypes[labels] + 0.7 * torch.randn(n, 1, 28, 28, generator=g)
    return torch.utils.data.TensorDataset(images, labels)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--generations", type=int, default=50)
    p.add_argument("--popsize", type=int, default=3200)
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    comm = init_comm() if world > 1 else None
    device = comm.device if comm is not None else ("cuda:0" if torch.cuda.is_available() else "cpu")

    problem = SupervisedNE(
        make_synthetic_mnist(),
        MNIST30K(),
        nn.CrossEntropyLoss(),
        minibatch_size=1024,
        common_minibatch=True,
        subbatch_size=50,       # reference MNIST30K config (Training_MNIST30K.ipynb)
        device=device,
        seed=1 + (comm.rank if comm else 0),
    )
    if comm is not None:
        problem.use_comm(comm)
    searcher = PGPE(
        problem,
        popsize=args.popsize,
        radius_init=2.25,
        center_learning_rate=1e-2,
        stdev_learning_rate=0.1,
        optimizer="adam",
        ranking_method=None,
        distributed=True,
    )
    if comm is None or comm.is_main:
        StdOutLogger(searcher, interval=10)
    searcher.run(args.generations)


if __name__ == "__main__":
    main()
