"""VecEnvNE with a ~1M-parameter MLP policy (the BASELINE 'PGPE 1M-param
MLP popsize=100k sharded across 8 GPUs' config, per-GPU shard): the
general (non-fused) rollout path — vmapped population forward through
rocBLAS batched GEMMs, obs-norm on device, PGPE distributed mode.

Run (1 GPU):  python scripts/bench_mlp_rollout.py
Multi-GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 scripts/bench_mlp_rollout.py
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--popsize-per-gpu", type=int, default=2048)
    p.add_argument("--hidden", type=int, default=2048)   # 376*2048 + 2048*17 + biases ≈ 0.81M
    p.add_argument("--episode-length", type=int, default=64)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=1)
    args = p.parse_args()

    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE
    from evotorch_amd.parallel import init_comm

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    have_gpu = torch.cuda.is_available()
    comm = init_comm() if world > 1 else None
    device = comm.device if comm is not None else ("cuda:0" if have_gpu else "cpu")

    def env_factory(num_envs):
        return SyntheticTorchEnv(num_envs=num_envs, episode_length=args.episode_length, device=device)

    net = f"Linear(obs_length, {args.hidden}) >> Tanh() >> Linear({args.hidden}, act_length)"
    problem = VecEnvNE(env_factory, net, device=device, seed=1 + rank,
                       max_num_steps=args.episode_length)
    if comm is not None:
        problem.use_comm(comm)
    total_pop = args.popsize_per_gpu * world
    searcher = PGPE(problem, popsize=total_pop, radius_init=2.25,
                    center_learning_rate=0.1, stdev_learning_rate=0.1,
                    optimizer="clipup", distributed=True)

    def sync():
        if comm is not None:
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
    dt = time.perf_counter() - t0
    if comm is not None:
        t = torch.tensor([dt], dtype=torch.float64, device=device if have_gpu else "cpu")
        comm.all_reduce_(t, op="max")
        dt = float(t)
    if rank == 0:
        print(json.dumps({
            "metric": "MLP-policy rollout solutions/sec",
            "params_per_solution": problem.solution_length,
            "global_popsize": total_pop,
            "episode_length": args.episode_length,
            "n_gpus": world,
            "solutions_per_sec": total_pop * args.steps / dt,
            "ms_per_gen": dt / args.steps * 1000,
        }))


if __name__ == "__main__":
    main()
