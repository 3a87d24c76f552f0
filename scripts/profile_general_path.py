import sys, os, time, torch
sys.path.insert(0, "/root/repo")
from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE
T = 200
prob = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=T, device="cuda:0"),
                "Linear(obs_length, act_length)", device="cuda:0", seed=1,
                max_num_steps=T, use_hip_graph=True)
r = 2.25
s = PGPE(prob, popsize=2048, radius_init=r, center_learning_rate=0.75*r/15,
         stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r/15},
         distributed=True)
for _ in range(3): s.step()
torch.cuda.synchronize()
for _ in range(5): s.step()
torch.cuda.synchronize()
print("done")
