"""Kernel-only timing + eager numerics check for the flagship rollout."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from evotorch_amd import ops
from evotorch_amd.neuroevolution.synthetic_env import SyntheticEnvSpec, rollout_eager

mod = ops.hip_required()
spec = SyntheticEnvSpec(episode_length=10, device="cuda")
torch.manual_seed(5)
params = 0.1 * torch.randn(33, spec.solution_length, device="cuda")
mean = torch.zeros(spec.obs_dim, device="cuda")
std = torch.ones(spec.obs_dim, device="cuda")
blob = spec.env_blob(mean, std, device="cuda")
os_ = torch.zeros(2 * spec.obs_dim, device="cuda")
fit = mod.rollout_linear(params, blob, os_, spec.obs_dim, spec.act_dim, spec.rank,
                         spec.episode_length, spec.alive_bonus, spec.act_cost, 77, 0)
efit, _ = rollout_eager(spec, params, mean, std, init_seed=77)
print("numerics:", bool(torch.allclose(fit, efit, rtol=2e-2, atol=2e-2)), float((fit - efit).abs().max()))

spec = SyntheticEnvSpec(episode_length=1000, device="cuda")
params = 0.1 * torch.randn(4000, spec.solution_length, device="cuda")
blob = spec.env_blob(mean, std, device="cuda")
for _ in range(3):
    mod.rollout_linear(params, blob, os_, spec.obs_dim, spec.act_dim, spec.rank, 1000,
                       spec.alive_bonus, spec.act_cost, 7, 0)
torch.cuda.synchronize()
t0 = time.perf_counter()
for i in range(20):
    mod.rollout_linear(params, blob, os_, spec.obs_dim, spec.act_dim, spec.rank, 1000,
                       spec.alive_bonus, spec.act_cost, 7 + i, 0)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / 20 * 1000
print(f"rollout kernel T=1000 popsize 4000: {ms:.3f} ms ({4000/ms*1000:,.0f} sol/s kernel-only)")
