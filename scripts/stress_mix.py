"""Cross-feature stress: create, run, and DROP many searchers of every
family in one process (a long notebook session) — exercises allocator
reuse, graph-pool lifetime, and GC interplay."""
import gc, math, os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from evotorch_amd import Problem
from evotorch_amd.algorithms import CEM, CMAES, PGPE, SNES, XNES, GeneticAlgorithm, GraphedSearch, MAPElites, make_feature_grid
from evotorch_amd.decorators import vectorized
from evotorch_amd.neuroevolution import SyntheticRolloutProblem, SyntheticTorchEnv, VecEnvNE
from evotorch_amd.operators import GaussianMutation, PolynomialMutation, SimulatedBinaryCrossOver

@vectorized
def sphere(x): return (x**2).sum(-1)

@vectorized
def multi(x):
    return torch.stack([(x**2).sum(-1), ((x - 1) ** 2).sum(-1)], dim=-1)

D = "cuda:0"
gc.set_threshold(50, 5, 5)  # aggressive cyclic GC to provoke lifetime bugs
for cycle in range(6):
    prob = SyntheticRolloutProblem(device=D, seed=cycle, episode_length=30)
    s = PGPE(prob, popsize=256, center_learning_rate=0.1, stdev_learning_rate=0.1, radius_init=1.0)
    g = GraphedSearch(s, generations_per_capture=4)
    g.capture(); g.run(20)

    pv = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=20, device=D),
                  "Linear(obs_length, act_length)", device=D, seed=cycle, max_num_steps=20, use_hip_graph=True)
    vp = PGPE(pv, popsize=64 + 32 * (cycle % 2), radius_init=1.0, center_learning_rate=0.1,
              stdev_learning_rate=0.1, distributed=True)
    vp.run(6)

    p2 = Problem("min", sphere, solution_length=300 + cycle, initial_bounds=(-1, 1), seed=cycle, device=D)
    CMAES(p2, stdev_init=1.0, popsize=32).run(30)
    SNES(p2, popsize=64, stdev_init=1.0).run(30)
    XNES(Problem("min", sphere, solution_length=48, initial_bounds=(-1, 1), seed=cycle, device=D),
         popsize=32, stdev_init=1.0).run(20)
    CEM(p2, popsize=128, stdev_init=1.0, parenthood_ratio=0.3).run(20)

    p3 = Problem(["min", "min"], multi, solution_length=16, initial_bounds=(0, 1), bounds=(0.0, 1.0),
                 seed=cycle, device=D)
    GeneticAlgorithm(p3, popsize=512, operators=[
        SimulatedBinaryCrossOver(p3, tournament_size=2, eta=10),
        PolynomialMutation(p3, eta=20)]).run(10)

    del prob, s, g, pv, vp, p2, p3
    gc.collect()
    print(f"cycle {cycle} ok, mem {torch.cuda.memory_allocated()/2**20:.0f} MiB")
torch.cuda.synchronize()
print("stress mix ok")
