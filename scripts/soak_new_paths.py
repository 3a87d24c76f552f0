"""Stability/convergence soak of the round-2 late additions: panel-
Cholesky CMA-ES and hipGraph-replayed VecEnvNE rollouts."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from evotorch_amd import Problem
from evotorch_amd.algorithms import CMAES, PGPE
from evotorch_amd.decorators import vectorized
from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE

@vectorized
def sphere(x):
    return (x**2).sum(-1)

prob = Problem("min", sphere, solution_length=4096, initial_bounds=(-1, 1), seed=1, device="cuda:0")
s = CMAES(prob, stdev_init=1.0, popsize=64)
t0 = time.perf_counter(); s.run(2000); torch.cuda.synchronize()
el = time.perf_counter() - t0
print(f"CMA d=4096: 2000 gens in {el:.1f}s ({2000/el:.0f} gens/s), mean_eval {float(s.status['mean_eval']):.4f} (start ~4096)")
assert float(s.status["mean_eval"]) < 500.0

T = 200
p = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=T, device="cuda:0"),
             "Linear(obs_length, act_length)", device="cuda:0", seed=1, max_num_steps=T, use_hip_graph=True)
r = 2.25
g = PGPE(p, popsize=2048, radius_init=r, center_learning_rate=0.75*r/15, stdev_learning_rate=0.1,
         optimizer="clipup", optimizer_config={"max_speed": r/15}, distributed=True)
first = None
t0 = time.perf_counter()
for i in range(300):
    g.step()
    if i == 0:
        first = float(g.status["mean_eval"])
torch.cuda.synchronize()
el = time.perf_counter() - t0
last = float(g.status["mean_eval"])
print(f"graphed VecEnvNE: 300 gens in {el:.1f}s ({el/300*1000:.1f} ms/gen), mean_eval {first:.2f} -> {last:.2f}")
assert last > first + 1.0
print(f"mem allocated {torch.cuda.memory_allocated()/2**20:.0f} MiB")
print("soak ok")

# extended flagship soak: 10k generations of the v7 SPMD path at T=200
from evotorch_amd.neuroevolution import SyntheticRolloutProblem
from evotorch_amd.parallel import init_comm
prob = SyntheticRolloutProblem(device="cuda:0", seed=2, episode_length=200)
comm = init_comm()
prob.use_comm(comm)
r = 2.25
s2 = PGPE(prob, popsize=4000, radius_init=r, center_learning_rate=0.75*r/15, stdev_learning_rate=0.1,
          optimizer="clipup", optimizer_config={"max_speed": r/15}, distributed=True)
for _ in range(50):
    s2.step()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10000):
    s2.step()
torch.cuda.synchronize()
el = time.perf_counter() - t0
me = float(s2.status["mean_eval"])
print(f"flagship 10k-gen soak: {el:.1f}s ({10000/el:.0f} gens/s, {4000*10000/el/1e6:.2f}M sol/s), mean_eval {me:.1f}")
assert me > 400.0 and el == el
print("extended soak ok")
