import sys, time
sys.path.insert(0, ".")
import torch
from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticRolloutProblem

def run(policy_hidden, distributed, steps=30, warmup=5):
    prob = SyntheticRolloutProblem(device="cuda:0", seed=1, episode_length=200, policy_hidden=policy_hidden)
    radius = 2.25
    s = PGPE(prob, popsize=4000, radius_init=radius, center_learning_rate=0.75*radius/15,
             stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": radius/15},
             distributed=distributed)
    for _ in range(warmup): s.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps): s.step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    return dt * 1000

for ph in (0, 64):
    for dist in (True, False):
        ms = run(ph, dist)
        print(f"policy_hidden={ph} distributed={dist}: {ms:.2f} ms/gen ({4000/ms*1000:,.0f} sol/s)")
