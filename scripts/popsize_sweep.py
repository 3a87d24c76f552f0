"""Flagship throughput vs popsize: how the fused rollout fills the chip."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticRolloutProblem


def main():
    for popsize in (512, 1024, 2048, 4000, 8192, 16384, 32768):
        prob = SyntheticRolloutProblem(device="cuda:0", seed=1, episode_length=200)
        r = 2.25
        s = PGPE(prob, popsize=popsize, radius_init=r, center_learning_rate=0.75 * r / 15,
                 stdev_learning_rate=0.1, optimizer="clipup", optimizer_config={"max_speed": r / 15},
                 distributed=True)
        for _ in range(3):
            s.step()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(15):
            s.step()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 15
        print(f"popsize {popsize:6d}: {dt*1000:7.2f} ms/gen  {popsize/dt:12,.0f} sol/s")


if __name__ == "__main__":
    main()
