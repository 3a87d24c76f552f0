import sys, time, torch
sys.path.insert(0, "/root/repo")
from evotorch_amd import ops
from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticRolloutProblem
from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec
from evotorch_amd.parallel import init_comm

# m7 popsize sweep (kernel-only)
mod = ops.hip_required()
spec = SyntheticEnvSpec(episode_length=1000, device="cuda", policy_hidden=64)
mean = torch.zeros(spec.obs_dim, device="cuda"); std = torch.ones(spec.obs_dim, device="cuda")
blob = spec.env_blob(mean, std, device="cuda")
for pop in (512, 1024, 2048, 4096, 8192, 16384):
    params = 0.1 * torch.randn(pop, spec.solution_length, device="cuda")
    os_ = torch.zeros(2 * spec.obs_dim, device="cuda")
    for _ in range(2):
        mod.rollout_linear(params, blob, os_, spec.obs_dim, spec.act_dim, spec.rank, 1000,
                           spec.alive_bonus, spec.act_cost, 7, 0, 64)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for i in range(4):
        mod.rollout_linear(params, blob, os_, spec.obs_dim, spec.act_dim, spec.rank, 1000,
                           spec.alive_bonus, spec.act_cost, 7 + i, 0, 64)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 4 * 1000
    print(f"m7 pop={pop:6d}: {ms:7.2f} ms  ({pop/ms*1000:,.0f} sol/s)")

# endurance: 100k generations flagship T=200
prob = SyntheticRolloutProblem(device="cuda:0", seed=1, episode_length=200)
prob.use_comm(init_comm())
r = 2.25
s = PGPE(prob, popsize=4000, radius_init=r, center_learning_rate=0.75*r/15, stdev_learning_rate=0.1,
         optimizer="clipup", optimizer_config={"max_speed": r/15}, distributed=True)
for _ in range(20): s.step()
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(100_000): s.step()
torch.cuda.synchronize()
el = time.perf_counter() - t0
print(f"flagship endurance: 100,000 gens in {el:.1f}s ({4000*100000/el/1e6:.2f}M sol/s sustained), "
      f"mean_eval {float(s.status['mean_eval']):.1f}, mem {torch.cuda.memory_allocated()/2**20:.0f} MiB")
