#!/usr/bin/env python
"""MNIST30K-style supervised-NE benchmark.

The reference config (/root/reference/examples/notebooks/
Training_MNIST30K.ipynb cells 2, 6, 8, 11): distributed PGPE + Adam over
a ~28k-parameter convnet (conv5x5x16 → pool → conv5x5x32 → pool →
layernorm → linear(1568, 10)), popsize 3200, center_lr 1e-2, stdev_lr
0.1, radius_init 2.25, ranking_method None (raw), minibatch 1024,
common_minibatch with subbatch_size 50. No dataset ships offline, so the
data is a synthetic MNIST-shaped classification task (random class
prototypes + pixel noise — random-init weights, synthetic data per the
benchmark contract); the measured quantity is generations/second and the
sanity signal is the training loss decreasing.

Single GPU:  python scripts/bench_supervised.py --steps 10 --warmup 2
Multi GPU:   python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                 --master-addr 127.0.0.1 scripts/bench_supervised.py ...
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch import nn
from torch.utils.data import TensorDataset


class MNIST30KNet(nn.Module):
    """The reference notebook's ~28k-parameter MNIST architecture."""

    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 16, kernel_size=5, stride=1, padding=2)
        self.pool1 = nn.MaxPool2d(kernel_size=2)
        self.conv2 = nn.Conv2d(16, 32, kernel_size=5, stride=1, padding=2)
        self.pool2 = nn.MaxPool2d(kernel_size=2)
        self.norm = nn.LayerNorm(1568, elementwise_affine=False)
        self.out = nn.Linear(1568, 10)
        self.act = nn.ReLU()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.pool1(self.act(self.conv1(x)))
        x = self.pool2(self.act(self.conv2(x)))
        # flatten the trailing (C, H, W) dims only — vmap-friendly
        x = self.norm(x.flatten(start_dim=-3))
        return self.out(x)

    def population_forward(self, params: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
        """SupervisedNE population_forward protocol: the whole member
        subbatch in ONE channel-grouped pass. Member weights become
        channel groups (conv1: one plain conv with G·16 output channels
        over the SHARED minibatch; conv2: groups=G), so the data stays in
        a single MIOpen-friendly NCHW layout — no vmap grouped/naive conv
        lowering, no per-op member-dim reshapes. Exactly the same math as
        vmapping `forward` over members (same parameter flattening order:
        conv1.w, conv1.b, conv2.w, conv2.b, out.w, out.b)."""
        import torch.nn.functional as F

        G, N = params.shape[0], x.shape[0]
        i = 0

        def take(k):
            nonlocal i
            v = params[:, i : i + k]
            i += k
            return v

        w1 = take(400).reshape(G * 16, 1, 5, 5)
        b1 = take(16).reshape(G * 16)
        w2 = take(12800).reshape(G * 32, 16, 5, 5)
        b2 = take(32).reshape(G * 32)
        w3 = take(15680).reshape(G, 10, 1568)
        b3 = take(10)
        h = F.max_pool2d(F.relu(F.conv2d(x, w1, b1, padding=2)), 2)
        h = F.max_pool2d(F.relu(F.conv2d(h, w2, b2, padding=2, groups=G)), 2)
        f = F.layer_norm(h.reshape(N, G, 1568), (1568,))
        return torch.baddbmm(b3.unsqueeze(1), f.transpose(0, 1), w3.transpose(1, 2))


def synthetic_mnist(n: int, seed: int = 0):
    """MNIST-shaped synthetic classification: 10 random 28×28 prototypes +
    noise; labels = prototype index."""
    g = torch.Generator().manual_seed(seed)
    protos = torch.randn(10, 1, 28, 28, generator=g)
    labels = torch.randint(0, 10, (n,), generator=g)
    x = protos[labels] + 0.7 * torch.randn(n, 1, 28, 28, generator=g)
    return TensorDataset(x, labels)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--popsize", type=int, default=3200)
    p.add_argument("--minibatch", type=int, default=1024)
    p.add_argument("--subbatch", type=int, default=50)
    p.add_argument("--data-size", type=int, default=8192)
    args = p.parse_args()

    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import SupervisedNE
    from evotorch_amd.parallel import init_comm

    comm = init_comm()
    device = comm.device
    world = comm.world_size
    rank = comm.rank

    net = MNIST30KNet()

    problem = SupervisedNE(
        synthetic_mnist(args.data_size, seed=123),
        lambda: MNIST30KNet(),
        loss_func=lambda y_hat, y: torch.nn.functional.cross_entropy(y_hat, y),
        minibatch_size=args.minibatch,
        common_minibatch=True,
        subbatch_size=args.subbatch,
        device=device,
        seed=1000 + rank,
    )
    problem.use_comm(comm)

    searcher = PGPE(
        problem,
        popsize=args.popsize,
        center_learning_rate=1e-2,
        stdev_learning_rate=0.1,
        radius_init=2.25,
        optimizer="adam",
        ranking_method=None,
        distributed=True,
    )

    def sync():
        comm.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        searcher.step()
    first_loss = float(searcher.status["mean_eval"])

    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        searcher.step()
    sync()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if torch.cuda.is_available() else "cpu")
        comm.all_reduce_(t, op="max")
        elapsed = float(t)

    if rank == 0:
        n_params = sum(x.numel() for x in net.parameters())
        print(json.dumps({
            "metric": "gens/sec, distributed PGPE+Adam, MNIST30K convnet",
            "value": args.steps / elapsed,
            "unit": "gens/sec",
            "n_gpus": world if torch.cuda.is_available() else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "dtype": "fp32",
            "data": "synthetic (MNIST-shaped prototype classification; no dataset offline)",
            "config": {
                "model": f"MNIST30K convnet ({n_params} params), PGPE+Adam, popsize {args.popsize}, "
                         f"minibatch {args.minibatch}, common_minibatch subbatch {args.subbatch}",
                "parallelism": f"dp{world}",
                "first_mean_loss": first_loss,
                "final_mean_loss": float(searcher.status["mean_eval"]),
            },
        }))


if __name__ == "__main__":
    main()
