"""5-billion-parameter separable-ES generation on ONE MI355X via the
streaming gradient path (grad_chunk_rows): the population is never
materialized. Memory: center + sigma + 2 gradient rows + 1 chunk row
of 3e9 fp32 each of the 288 GB HBM3E."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from evotorch_amd import Problem
from evotorch_amd.algorithms import PGPE
from evotorch_amd.decorators import vectorized

L = 5_000_000_000

@vectorized
def head_sphere(x):
    return (x[:, :4096] ** 2).sum(-1)

prob = Problem("min", head_sphere, solution_length=L, initial_bounds=(-0.1, 0.1), seed=5, device="cuda:0")
s = PGPE(prob, popsize=8, center_learning_rate=0.05, stdev_learning_rate=0.05,
         stdev_init=0.1, distributed=True, grad_chunk_rows=1)
s.step()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(3):
    s.step()
torch.cuda.synchronize()
el = (time.perf_counter() - t0) / 3
peak = torch.cuda.max_memory_allocated() / 2**30
print(f"5e9-param PGPE generation: {el:.2f} s/gen, peak {peak:.0f} GiB, mean_eval {float(s.status['mean_eval']):.3f}")
