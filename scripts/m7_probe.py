import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from evotorch_amd import ops
from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec, rollout_eager

mod = ops.hip_required()
spec = SyntheticEnvSpec(episode_length=10, device="cuda", policy_hidden=64)
torch.manual_seed(5)
n = 35  # off-multiple of 4
params = 0.1 * torch.randn(n, spec.solution_length, device="cuda")
mean = torch.zeros(spec.obs_dim, device="cuda")
std = torch.ones(spec.obs_dim, device="cuda")
blob = spec.env_blob(mean, std, device="cuda")
os_ = torch.zeros(2 * spec.obs_dim, device="cuda")
fit = mod.rollout_linear(params, blob, os_, spec.obs_dim, spec.act_dim, spec.rank,
                         spec.episode_length, spec.alive_bonus, spec.act_cost, 77, 0, spec.policy_hidden)
efit, (cnt, esum, esumsq) = rollout_eager(spec, params, mean, std, init_seed=77)
ok_f = torch.allclose(fit, efit, rtol=2e-2, atol=2e-2)
ok_s = torch.allclose(os_[:spec.obs_dim], esum, rtol=5e-2, atol=5e-1) and torch.allclose(os_[spec.obs_dim:], esumsq, rtol=5e-2, atol=5e-1)
print("numerics fit:", bool(ok_f), " stats:", bool(ok_s), " maxerr:", float((fit-efit).abs().max()))

spec = SyntheticEnvSpec(episode_length=1000, device="cuda", policy_hidden=64)
params = 0.1 * torch.randn(4000, spec.solution_length, device="cuda")
blob = spec.env_blob(mean, std, device="cuda")
os_ = torch.zeros(2 * spec.obs_dim, device="cuda")
for _ in range(2):
    mod.rollout_linear(params, blob, os_, spec.obs_dim, spec.act_dim, spec.rank, 1000,
                       spec.alive_bonus, spec.act_cost, 7, 0, spec.policy_hidden)
torch.cuda.synchronize()
t0 = time.perf_counter()
for i in range(6):
    mod.rollout_linear(params, blob, os_, spec.obs_dim, spec.act_dim, spec.rank, 1000,
                       spec.alive_bonus, spec.act_cost, 7 + i, 0, spec.policy_hidden)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / 6 * 1000
which = "v6" if os.environ.get("EVOTORCH_AMD_ROLLOUT_V6") else "m7"
print(f"{which} MLP-64 rollout T=1000 popsize 4000: {ms:.2f} ms ({4000/ms*1000:,.0f} sol/s kernel-only)")
