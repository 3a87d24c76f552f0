import gc, sys, torch
sys.path.insert(0, "/root/repo")
from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticTorchEnv, VecEnvNE
D = "cuda:0"

def cyc(i, graph, obsnorm, dist):
    pv = VecEnvNE(lambda n: SyntheticTorchEnv(num_envs=n, episode_length=20, device=D),
                  "Linear(obs_length, act_length)", device=D, seed=i, max_num_steps=20,
                  use_hip_graph=graph, observation_normalization=obsnorm)
    PGPE(pv, popsize=64, radius_init=1.0, center_learning_rate=0.1, stdev_learning_rate=0.1,
         distributed=dist).run(6)

for name, kw in (("eager", dict(graph=False, obsnorm=True, dist=True)),
                 ("graph+obsnorm", dict(graph=True, obsnorm=True, dist=True)),
                 ("graph-no-obsnorm", dict(graph=True, obsnorm=False, dist=True)),
                 ("graph-nondist", dict(graph=True, obsnorm=True, dist=False))):
    gc.collect(); torch.cuda.synchronize()
    m0 = torch.cuda.memory_allocated()
    for i in range(3):
        cyc(i, **kw)
        gc.collect()
    torch.cuda.synchronize()
    print(f"{name:18s} growth {(torch.cuda.memory_allocated()-m0)/2**20:7.1f} MiB / 3 cycles")
# referrer hunt: find live CUDAGraphs after everything dropped
gc.collect()
graphs = [o for o in gc.get_objects() if isinstance(o, torch.cuda.CUDAGraph)]
print("live CUDAGraph objects:", len(graphs))
if graphs:
    import types
    refs = gc.get_referrers(graphs[0])
    for r in refs[:5]:
        print("  referrer:", type(r), str(r)[:120])
