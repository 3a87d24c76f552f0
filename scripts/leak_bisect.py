import gc, os, sys, torch
sys.path.insert(0, "/root/repo")
from evotorch_amd import Problem
from evotorch_amd.algorithms import CEM, CMAES, PGPE, SNES, XNES, GeneticAlgorithm, GraphedSearch
from evotorch_amd.decorators import vectorized
from evotorch_amd.neuroevolution import SyntheticRolloutProblem, SyntheticTorchEnv, VecEnvNE
from evotorch_amd.operators import PolynomialMutation, SimulatedBinaryCrossOver

@vectorized
def sphere(x): return (x**2).sum(-1)
@vectorized
def multi(x): return torch.stack([(x**2).sum(-1), ((x-1)**2).sum(-1)], dim=-1)
D = "cuda:0"

def cyc_graphed(i):
    prob = SyntheticRolloutProblem(device=D, seed=i, episode_length=30)
    s = PGPE(prob, popsize=256, center_learning_rate=0.1, stdev_learning_rate=0.1, radius_init=1.0)
    g = GraphedSearch(s, generations_per_capture=4); g.capture(); g.run(20)

def cyc_vecenv(i):
    pv = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=20, device=D),
                  "Linear(obs_length, act_length)", device=D, seed=i, max_num_steps=20, use_hip_graph=True)
    PGPE(pv, popsize=64, radius_init=1.0, center_learning_rate=0.1, stdev_learning_rate=0.1, distributed=True).run(6)

def cyc_es(i):
    p2 = Problem("min", sphere, solution_length=300+i, initial_bounds=(-1,1), seed=i, device=D)
    CMAES(p2, stdev_init=1.0, popsize=32).run(30)
    SNES(p2, popsize=64, stdev_init=1.0).run(30)
    CEM(p2, popsize=128, stdev_init=1.0, parenthood_ratio=0.3).run(20)

def cyc_xnes(i):
    XNES(Problem("min", sphere, solution_length=48, initial_bounds=(-1,1), seed=i, device=D),
         popsize=32, stdev_init=1.0).run(20)

def cyc_ga(i):
    p3 = Problem(["min","min"], multi, solution_length=16, initial_bounds=(0,1), bounds=(0.0,1.0), seed=i, device=D)
    GeneticAlgorithm(p3, popsize=512, operators=[
        SimulatedBinaryCrossOver(p3, tournament_size=2, eta=10), PolynomialMutation(p3, eta=20)]).run(10)

for name, fn in (("graphed", cyc_graphed), ("vecenv", cyc_vecenv), ("es", cyc_es), ("xnes", cyc_xnes), ("ga", cyc_ga)):
    gc.collect(); torch.cuda.synchronize()
    m0 = torch.cuda.memory_allocated()
    for i in range(4):
        fn(i)
        gc.collect()
    torch.cuda.synchronize()
    print(f"{name:8s} growth {(torch.cuda.memory_allocated()-m0)/2**20:7.1f} MiB over 4 cycles")
