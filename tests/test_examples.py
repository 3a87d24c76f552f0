"""Run each example end-to-end with tiny budgets (mirrors the reference's
tests/test_examples.py integration strategy)."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

CASES = [
    ("examples/rastrigin_snes.py", ["--generations", "3"]),
    ("examples/cartpole_gymne.py", ["--generations", "3"]),
    ("examples/parallel_cpu_actors.py", ["--generations", "2", "--popsize", "8"]),
    ("examples/multiobjective_nsga2.py", []),
    ("examples/mapelites_illumination.py", []),
    ("examples/functional_api_batched.py", []),
    ("examples/synthetic_humanoid_pgpe.py", ["--generations", "2", "--popsize", "16", "--device", "cpu"]),
    ("examples/mnist30k_distributed.py", ["--generations", "1", "--popsize", "8"]),
    ("examples/genetic_programming.py", ["--generations", "5", "--popsize", "64"]),
    ("examples/checkpoint_resume.py", []),
    ("examples/lennard_jones_cmaes.py", ["--generations", "30"]),
    ("examples/mpc_cem_pendulum.py", ["--steps", "90"]),
    ("examples/vqe_snes.py", ["--generations", "250"]),
]


@pytest.mark.parametrize("script,args", CASES, ids=[c[0] for c in CASES])
def test_example_runs(script, args, tmp_path):
    result = subprocess.run(
        [sys.executable, os.path.join(REPO, script), *args],
        cwd=tmp_path,  # checkpoints etc. land in a temp dir
        capture_output=True,
        text=True,
        timeout=420,
    )
    assert result.returncode == 0, f"{script} failed:\n{result.stderr[-2000:]}"


def test_bench_json_contract():
    """bench.py must print ONE JSON line with the driver-contract fields
    (metric/value/unit/n_gpus/steps/warmup/ms_per_step/higher_is_better/
    scaling/vs_baseline/dtype/data/config) — run tiny on CPU."""
    import json

    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "1", "--warmup", "0",
         "--popsize", "8", "--episode-length", "5"],
        capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
                "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"):
        assert key in d, f"missing {key}"
    assert d["steps"] == 1 and d["warmup"] == 0
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["scaling"] in ("weak", "strong")
    assert isinstance(d["config"], dict) and "model" in d["config"]
