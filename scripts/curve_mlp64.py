"""500-generation PGPE learning curve on the MLP-64 synthetic humanoid
(reference notebook config: popsize 4000, radius 2.25, ClipUp). Writes the
per-generation mean/best rewards to gpurun_out/curve_mlp64.json."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd.algorithms import PGPE
from evotorch_amd.neuroevolution import SyntheticRolloutProblem


def main():
    gens = int(sys.argv[1]) if len(sys.argv) > 1 else 500
    prob = SyntheticRolloutProblem(device="cuda:0", seed=42, episode_length=200, policy_hidden=64)
    radius = 2.25
    searcher = PGPE(
        prob, popsize=4000, radius_init=radius,
        center_learning_rate=0.75 * radius / 15, stdev_learning_rate=0.1,
        optimizer="clipup", optimizer_config={"max_speed": radius / 15},
        distributed=True,
    )
    curve = []
    t0 = time.perf_counter()
    for g in range(gens):
        searcher.step()
        curve.append(float(searcher.status["mean_eval"]))
        if (g + 1) % 100 == 0:
            print(f"gen {g+1}: mean_eval={curve[-1]:.1f}", flush=True)
    dt = time.perf_counter() - t0
    out = {
        "gens": gens, "seconds": round(dt, 2), "gens_per_sec": round(gens / dt, 2),
        "solutions_per_sec": round(gens * 4000 / dt, 1),
        "first": curve[0], "last": curve[-1],
        "curve_every_10": [round(c, 2) for c in curve[::10]],
    }
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/curve_mlp64.json", "w") as f:
        json.dump(out, f, indent=1)
    print(json.dumps({k: v for k, v in out.items() if k != "curve_every_10"}))


if __name__ == "__main__":
    main()
