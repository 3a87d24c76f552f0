"""Whole-generation hipGraph on the flagship rollout problem: PGPE +
fused rollout kernel + ranking + gradients + ClipUp + obs-norm merge
replayed as ONE graph per generation (device-side episode/sampling seed
chains). Eager SPMD numbers for comparison come from bench.py."""
import os, sys, time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from evotorch_amd.algorithms import PGPE, GraphedSearch
from evotorch_amd.neuroevolution import SyntheticRolloutProblem

for T in (200, 1000):
    prob = SyntheticRolloutProblem(device="cuda:0", seed=1, episode_length=T)
    r = 2.25
    s = PGPE(prob, popsize=4000, radius_init=r, center_learning_rate=0.75 * r / 15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r / 15})
    g = GraphedSearch(s, generations_per_capture=10)
    g.capture()
    g.run(20)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    steps = 200 if T == 200 else 60
    g.run(steps)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / steps * 1000
    print(f"T={T:5d}: {ms:6.3f} ms/gen graphed  ({4000/ms*1000:,.0f} sol/s)  mean_eval={float(g.mean_eval):.1f}")

# latency-bound region: small populations gain the most from whole-
# generation replay
for pop in (256, 1024):
    prob = SyntheticRolloutProblem(device="cuda:0", seed=2, episode_length=200)
    r = 2.25
    s = PGPE(prob, popsize=pop, radius_init=r, center_learning_rate=0.75 * r / 15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r / 15})
    g = GraphedSearch(s, generations_per_capture=20)
    g.capture()
    g.run(40)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    g.run(400)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 400 * 1000
    e = PGPE(SyntheticRolloutProblem(device="cuda:0", seed=2, episode_length=200), popsize=pop,
             radius_init=r, center_learning_rate=0.75 * r / 15, stdev_learning_rate=0.1,
             optimizer="clipup", optimizer_config={"max_speed": r / 15}, distributed=True)
    for _ in range(10):
        e.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(100):
        e.step()
    torch.cuda.synchronize()
    ems = (time.perf_counter() - t0) / 100 * 1000
    print(f"pop={pop:5d} T=200: eager {ems:.3f} ms/gen vs graphed {ms:.3f} ms/gen ({ems/ms:.2f}x)")
