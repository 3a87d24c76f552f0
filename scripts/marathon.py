"""Round-2 end marathon: long sustained runs of every major path on one
box — throughput stability + convergence sanity under load."""
import math, os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from evotorch_amd import Problem
from evotorch_amd.algorithms import CMAES, PGPE, GeneticAlgorithm, GraphedSearch
from evotorch_amd.decorators import vectorized
from evotorch_amd.neuroevolution import SyntheticRolloutProblem, SyntheticTorchEnv, VecEnvNE
from evotorch_amd.operators import PolynomialMutation, SimulatedBinaryCrossOver
from evotorch_amd.parallel import init_comm

def stamp(name, gens, t0, extra=""):
    el = time.perf_counter() - t0
    print(f"{name:<42} {gens:>6} gens {el:8.1f}s  {gens/el:9.1f} gens/s  {extra}")

# 1. flagship SPMD 30k generations at T=200
prob = SyntheticRolloutProblem(device="cuda:0", seed=1, episode_length=200)
prob.use_comm(init_comm())
r = 2.25
s = PGPE(prob, popsize=4000, radius_init=r, center_learning_rate=0.75*r/15, stdev_learning_rate=0.1,
         optimizer="clipup", optimizer_config={"max_speed": r/15}, distributed=True)
for _ in range(20): s.step()
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(30000): s.step()
torch.cuda.synchronize()
stamp("flagship SPMD T=200", 30000, t0, f"mean_eval={float(s.status['mean_eval']):.1f} ({4000*30000/(time.perf_counter()-t0)/1e6:.2f}M sol/s)")

# 2. CMA-ES d=4096, 10k generations
@vectorized
def sphere(x): return (x**2).sum(-1)
p2 = Problem("min", sphere, solution_length=4096, initial_bounds=(-1, 1), seed=2, device="cuda:0")
c = CMAES(p2, stdev_init=1.0, popsize=64)
c.run(50)
torch.cuda.synchronize(); t0 = time.perf_counter()
c.run(10000)
torch.cuda.synchronize()
stamp("CMA-ES d=4096 full-cov", 10000, t0, f"best={float(c.status['pop_best_eval']):.2e}")

# 3. graphed VecEnvNE, 2000 generations
pv = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=200, device="cuda:0"),
              "Linear(obs_length, act_length)", device="cuda:0", seed=3, max_num_steps=200, use_hip_graph=True)
g = PGPE(pv, popsize=2048, radius_init=r, center_learning_rate=0.75*r/15, stdev_learning_rate=0.1,
         optimizer="clipup", optimizer_config={"max_speed": r/15}, distributed=True)
for _ in range(5): g.step()
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(2000): g.step()
torch.cuda.synchronize()
stamp("graphed VecEnvNE T=200", 2000, t0, f"mean_eval={float(g.status['mean_eval']):.1f}")

# 4. NSGA-II popsize 8192, 2000 generations
@vectorized
def multi(x):
    f1 = (x**2).sum(-1); f2 = ((x - 2.0)**2).sum(-1)
    return torch.stack([f1, f2], dim=-1)
p4 = Problem(["min", "min"], multi, solution_length=64, initial_bounds=(0, 1), bounds=(0.0, 1.0), seed=4, device="cuda:0")
ga = GeneticAlgorithm(p4, popsize=8192, operators=[
    SimulatedBinaryCrossOver(p4, tournament_size=4, cross_over_rate=1.0, eta=8),
    PolynomialMutation(p4, eta=20, mutation_probability=0.2)])
ga.step()
torch.cuda.synchronize(); t0 = time.perf_counter()
ga.run(2000)
torch.cuda.synchronize()
stamp("NSGA-II pop=8192", 2000, t0)

print(f"peak mem {torch.cuda.max_memory_allocated()/2**30:.1f} GiB")
print("marathon ok")
