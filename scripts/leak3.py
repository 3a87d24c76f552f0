import gc, sys, torch
sys.path.insert(0, "/root/repo")
from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE
D = "cuda:0"
for i in range(3):
    pv = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=20, device=D),
                  "Linear(obs_length, act_length)", device=D, seed=i, max_num_steps=20,
                  use_hip_graph=False)
    PGPE(pv, popsize=64, radius_init=1.0, center_learning_rate=0.1, stdev_learning_rate=0.1,
         distributed=True).run(6)
    del pv
gc.collect()
torch.cuda.synchronize()
print("allocated", torch.cuda.memory_allocated()/2**20, "MiB")
tensors = [o for o in gc.get_objects() if isinstance(o, torch.Tensor) and o.is_cuda]
tensors.sort(key=lambda t: -t.numel() * t.element_size())
total = sum(t.numel()*t.element_size() for t in tensors)
print(f"live cuda tensors: {len(tensors)}, total {total/2**20:.1f} MiB")
import itertools
seen_chains = 0
for t in tensors[:6]:
    print("tensor", tuple(t.shape), t.dtype, f"{t.numel()*t.element_size()/2**20:.2f} MiB")
    chain = t
    for depth in range(4):
        refs = [r for r in gc.get_referrers(chain) if not isinstance(r, (list, tuple)) or True][:3]
        named = []
        for r in refs:
            if isinstance(r, dict):
                keys = [k for k, v in r.items() if v is chain]
                named.append(f"dict(keys={keys[:3]})")
            else:
                named.append(type(r).__name__)
        print("   referrers:", named)
        if refs and isinstance(refs[0], (dict, list)):
            parents = gc.get_referrers(refs[0])[:2]
            print("   parent of first:", [type(p).__name__ for p in parents])
        break
