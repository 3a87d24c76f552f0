#!/usr/bin/env python
"""Flagship benchmark: PGPE + ClipUp over a population of linear policies
(Humanoid-v4 geometry: obs 376, act 17) with whole-population batched
rollouts — the BASELINE.json headline metric, measured as solutions/sec.

Single GPU:   python bench.py --gpus 1 --steps 20 --warmup 3
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Weak scaling: each rank owns a fixed per-GPU sub-population (default 4000)
sampled with counter-based philox streams; the centered ranking is global
(one all-gather of N floats) and the (mu, sigma) gradient merge is one
all-reduce. The rollout itself is ONE fused HIP kernel
launch per generation per rank (evotorch_amd/ops/hip/rollout.hip): policy
weights and env matrices live in LDS for the whole episode; compute is
bf16 operands with fp32 accumulation.

Config mirrors the reference's brax-humanoid PGPE setup
(/root/reference/examples/notebooks/Brax_Experiments_with_PGPE.ipynb
cells 5, 11: popsize 4000, radius_init 2.25, max_speed radius/15,
center_lr 0.75*max_speed, stdev_lr 0.1, obs-norm on) on synthetic data
(no simulator offline — see evotorch_amd/neuroevolution/synthetic_env.py).
"""

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    # defaults: on a GPU, enough steps that the timed region spans >=30 s
    # (strong SMI/telemetry evidence) while still finishing in ~a minute;
    # tiny on CPU where a generation costs seconds
    on_gpu_default = torch.cuda.is_available()
    p.add_argument("--steps", type=int, default=10000 if on_gpu_default else 4)
    p.add_argument("--warmup", type=int, default=100 if on_gpu_default else 1)
    p.add_argument("--popsize-per-gpu", type=int, default=4000)
    p.add_argument("--episode-length", type=int, default=1000,
                   help="episode steps per rollout; 1000 matches the reference flagship config (BASELINE.md row 3)")
    p.add_argument("--seed", type=int, default=1)
    p.add_argument("--policy", choices=["linear", "mlp64"], default="linear",
                   help="linear = Humanoid-v4 linear policy (the headline metric); mlp64 = the paper's MLP-64-tanh brax config")
    return p.parse_args()


def main():
    args = parse_args()
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem
    from evotorch_amd.parallel import init_comm

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    have_gpu = torch.cuda.is_available()

    # one Comm for every world size: N=1 runs the same SPMD code path
    # (counter-addressed sampling, global ranking, fused all-reduce with
    # no-op collectives) as the 8-GPU scaling run
    comm = init_comm()
    device = comm.device

    total_popsize = args.popsize_per_gpu * world  # weak scaling
    radius_init = 2.25
    max_speed = radius_init / 15.0
    center_lr = 0.75 * max_speed

    problem = SyntheticRolloutProblem(
        device=device,
        seed=args.seed + rank,
        episode_length=args.episode_length,
        observation_normalization=True,
        policy_hidden=64 if args.policy == "mlp64" else 0,
    )
    problem.use_comm(comm)

    searcher = PGPE(
        problem,
        popsize=total_popsize,
        radius_init=radius_init,
        center_learning_rate=center_lr,
        stdev_learning_rate=0.1,
        optimizer="clipup",
        optimizer_config={"max_speed": max_speed},
        ranking_method="centered",
        distributed=True,
    )

    def sync():
        comm.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        searcher.step()

    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        searcher.step()
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if have_gpu else "cpu")
        comm.all_reduce_(t, op="max")
        elapsed = float(t)

    solutions_per_sec = (total_popsize * args.steps) / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        result = {
            "metric": ("solutions/sec, PGPE Humanoid-v4 linear policy" if args.policy == "linear"
                       else "solutions/sec, PGPE Humanoid-v4 MLP-64 policy"),
            "value": solutions_per_sec,
            "unit": "solutions/sec",
            "n_gpus": world if have_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic (offline low-rank neural dynamics, Humanoid-v4 obs/act geometry; no simulator available offline)",
            "config": {
                "model": ("linear policy obs376->act17 (6409 params), PGPE+ClipUp, obs-norm on" if args.policy == "linear" else "MLP-64-tanh policy obs376->64->act17 (25233 params), PGPE+ClipUp, obs-norm on"),
                "global_batch": total_popsize,
                "seq_len": args.episode_length,
                "parallelism": f"dp{world}",
                "mean_eval": searcher.status["mean_eval"],
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
