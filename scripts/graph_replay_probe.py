"""Isolate hipGraph replay cost for the captured rollout: replay-only
wall time vs T (node count scales with T)."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE
from evotorch_amd.core import SolutionBatch

for T in (25, 50, 100, 200):
    prob = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=T, device="cuda:0"),
                    "Linear(obs_length, act_length)", device="cuda:0", seed=1,
                    max_num_steps=T, use_hip_graph=True, observation_normalization=False)
    n = 2048
    batch = SolutionBatch(prob, popsize=n, device="cuda:0")
    batch.access_values()[:] = 0.01 * torch.randn(n, prob.solution_length, device="cuda:0")
    prob.evaluate(batch)          # capture happens here
    g = prob._graph_state["graph"]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        g.replay()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 10 * 1000
    print(f"T={T:4d}: replay {ms:7.2f} ms  ({ms/T*1000:6.1f} us/step)")
