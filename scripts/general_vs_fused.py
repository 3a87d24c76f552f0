"""K10 evidence: the GENERAL batched-policy rollout path (VecEnvNE +
vmapped Policy forward per env step + torch env math — what any
user-supplied torch env/net gets) vs the fused whole-episode kernel
(SyntheticRolloutProblem), same dynamics and geometry."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticRolloutProblem, SyntheticTorchEnv, VecEnvNE


def time_searcher(prob, steps=10, warmup=2):
    r = 2.25
    s = PGPE(prob, popsize=2048, radius_init=r, center_learning_rate=0.75 * r / 15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r / 15},
             distributed=True)
    for _ in range(warmup):
        s.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        s.step()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1000


def main():
    T = 200
    fused = SyntheticRolloutProblem(device="cuda:0", seed=1, episode_length=T)
    ms_fused = time_searcher(fused)
    print(f"fused kernel path:      {ms_fused:8.2f} ms/gen  ({2048/ms_fused*1000:,.0f} sol/s)")

    general = VecEnvNE(
        lambda n: SyntheticTorchEnv(num_envs=n, episode_length=T, device="cuda:0"),
        "Linear(obs_length, act_length)",
        device="cuda:0",
        seed=1,
        max_num_steps=T,
    )
    ms_gen = time_searcher(general, steps=5, warmup=1)
    print(f"general vmapped path:   {ms_gen:8.2f} ms/gen  ({2048/ms_gen*1000:,.0f} sol/s)")

    graphed = VecEnvNE(
        lambda n: SyntheticTorchEnv(num_envs=n, episode_length=T, device="cuda:0"),
        "Linear(obs_length, act_length)",
        device="cuda:0",
        seed=1,
        max_num_steps=T,
        use_hip_graph=True,
    )
    ms_graph = time_searcher(graphed, steps=10, warmup=2)
    print(f"general + hipGraph:     {ms_graph:8.2f} ms/gen  ({2048/ms_graph*1000:,.0f} sol/s)")
    print(f"fused vs general: {ms_gen/ms_fused:.1f}x   graph vs general: {ms_gen/ms_graph:.1f}x   fused vs graph: {ms_graph/ms_fused:.1f}x")


main()
