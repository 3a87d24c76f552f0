import gc, sys, torch
sys.path.insert(0, "/root/repo")
torch.cuda.memory._record_memory_history(max_entries=200000)
from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE
D = "cuda:0"
for i in range(2):
    pv = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=20, device=D),
                  "Linear(obs_length, act_length)", device=D, seed=i, max_num_steps=20,
                  use_hip_graph=False)
    PGPE(pv, popsize=64, radius_init=1.0, center_learning_rate=0.1, stdev_learning_rate=0.1,
         distributed=True).run(6)
    del pv
gc.collect(); torch.cuda.synchronize()
_ = torch.zeros(4, device=D)  # force allocator event processing
gc.collect(); torch.cuda.synchronize()
print("allocated", torch.cuda.memory_allocated()/2**20, "MiB")
snap = torch.cuda.memory_snapshot()
from collections import Counter
sites = Counter()
for seg in snap:
    for blk in seg.get("blocks", []):
        if blk.get("state") == "active_allocated":
            frames = blk.get("frames") or []
            key = " <- ".join(f"{f['filename'].split('/')[-1]}:{f['line']}" for f in frames[:5]) or "no-frames"
            sites[key] += blk["size"]
for k, v in sites.most_common(6):
    print(f"{v/2**20:8.1f} MiB  {k[:200]}")
