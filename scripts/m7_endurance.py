import sys, time, torch
sys.path.insert(0, "/root/repo")
from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticRolloutProblem
from evotorch_amd.parallel import init_comm

prob = SyntheticRolloutProblem(device="cuda:0", seed=2, episode_length=1000, policy_hidden=64)
prob.use_comm(init_comm())
r = 2.25
s = PGPE(prob, popsize=4000, radius_init=r, center_learning_rate=0.75 * r / 15, stdev_learning_rate=0.1,
         optimizer="clipup", optimizer_config={"max_speed": r / 15}, distributed=True)
for _ in range(5):
    s.step()
torch.cuda.synchronize()
t0 = time.perf_counter()
n = 0
while time.perf_counter() - t0 < 330.0:
    s.step()
    n += 1
torch.cuda.synchronize()
el = time.perf_counter() - t0
me = float(s.status["mean_eval"])
mem = torch.cuda.memory_allocated() / 2**20
print(f"m7 endurance: {n} gens in {el:.1f}s ({4000*n/el/1000:.1f}k sol/s sustained), mean_eval {me:.1f}, mem {mem:.0f} MiB")
