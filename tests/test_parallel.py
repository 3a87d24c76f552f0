"""SPMD comm-layer tests over gloo with world_size 2 — the in-process
seam replacing the reference's `ray local_mode` trick
(tests/conftest.py:28-40 of the reference)."""

import os

import pytest
import torch
import torch.multiprocessing as mp

PORT = 29611


def _worker(rank: int, world: int, fn_name: str, port: int, q):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.manual_seed(100 + rank)
    import evotorch_amd.parallel.comm as comm_mod

    comm_mod._global_comm = None
    comm = comm_mod.Comm(backend="gloo", device=torch.device("cpu"))
    try:
        result = globals()[fn_name](comm, rank, world)
        q.put((rank, "ok", result))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, "error", traceback.format_exc()))
    finally:
        torch.distributed.destroy_process_group()


_port_counter = [PORT]


def _run_world(fn_name: str, world: int = 2):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    _port_counter[0] += 1
    port = _port_counter[0]
    procs = [ctx.Process(target=_worker, args=(r, world, fn_name, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, payload = q.get(timeout=180)
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=60)
    return results


# -- worker bodies -----------------------------------------------------------


def _body_collectives(comm, rank, world):
    t = torch.full((4,), float(rank + 1))
    comm.all_reduce_(t)
    assert torch.allclose(t, torch.full((4,), 3.0))

    v = torch.arange(3, dtype=torch.float32) + rank * 10
    gathered = comm.all_gather_vector(v)
    assert gathered.shape == (6,)
    assert torch.allclose(gathered[:3], torch.arange(3, dtype=torch.float32))
    assert torch.allclose(gathered[3:], torch.arange(3, dtype=torch.float32) + 10)

    grads = {"mu": torch.full((5,), float(rank)), "sigma": torch.full((2,), 1.0)}
    comm.all_reduce_container(grads)
    assert torch.allclose(grads["mu"], torch.full((5,), 1.0))
    assert torch.allclose(grads["sigma"], torch.full((2,), 2.0))

    full = torch.zeros(5, 2)
    ranges = [(0, 3), (3, 5)]
    a, b = ranges[rank]
    full[a:b] = rank + 1.0
    comm.all_gather_rows(full, ranges)
    assert torch.allclose(full[:3], torch.full((3, 2), 1.0))
    assert torch.allclose(full[3:], torch.full((2, 2), 2.0))
    return True


def _body_sharded_evaluate(comm, rank, world):
    from evotorch_amd import Problem
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=6, initial_bounds=(-1, 1), seed=5)
    prob.use_comm(comm)
    batch = prob.generate_batch(7)  # same values on both ranks (same seed)
    prob.evaluate(batch)
    evals = batch.unsafe_evals[:, 0]
    expected = (batch.unsafe_values**2).sum(-1)
    assert torch.allclose(evals, expected, atol=1e-5)
    return evals.tolist()


def _body_distributed_pgpe(comm, rank, world):
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=8, initial_bounds=(-1, 1), seed=50 + rank)
    prob.use_comm(comm)
    searcher = PGPE(prob, popsize=40, center_learning_rate=0.2, stdev_learning_rate=0.1,
                    stdev_init=1.0, distributed=True, center_init=torch.ones(8))
    for _ in range(10):
        searcher.step()
    center = searcher.status["center"]
    # all ranks hold identical distributions after all-reduced updates
    center_t = torch.Tensor.as_subclass(center, torch.Tensor).clone()
    ref = center_t.clone()
    comm.broadcast_(ref, src=0)
    assert torch.allclose(center_t, ref, atol=1e-6), "ranks diverged"
    assert searcher.status["mean_eval"] < 8.0  # descended from ~8 (center ones)
    return center_t.tolist()


def _body_obs_norm_allreduce(comm, rank, world):
    from evotorch_amd.neuroevolution import SyntheticRolloutProblem

    prob = SyntheticRolloutProblem(seed=60 + rank, episode_length=5)
    prob.use_comm(comm)
    result = prob.sample_and_compute_gradients(_make_dist(prob), 8, ranking_method="centered")
    assert result["num_solutions"] == 8
    # both ranks merged both shards' stats: 8 members * 5 steps = 40 obs
    assert prob.obs_norm.count == 40.0
    return prob.obs_norm.count


def _body_streamed_pgpe(comm, rank, world):
    """Sharded + streamed gradients: ranks stay in lockstep and descend."""
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import PGPE
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=8, initial_bounds=(-1, 1), seed=70 + rank)
    prob.use_comm(comm)
    searcher = PGPE(prob, popsize=40, center_learning_rate=0.2, stdev_learning_rate=0.1,
                    stdev_init=1.0, distributed=True, center_init=torch.ones(8),
                    grad_chunk_rows=4)
    for _ in range(10):
        searcher.step()
    center_t = torch.Tensor.as_subclass(searcher.status["center"], torch.Tensor).clone()
    ref = center_t.clone()
    comm.broadcast_(ref, src=0)
    assert torch.allclose(center_t, ref, atol=1e-6), "ranks diverged"
    assert searcher.status["mean_eval"] < 8.0
    return center_t.tolist()


def _body_streamed_cem(comm, rank, world):
    """Sharded + streamed CEM: global elite set, (Σx, Σx²) all-reduce."""
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import CEM
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    prob = Problem("min", sphere, solution_length=8, initial_bounds=(-1, 1), seed=80 + rank)
    prob.use_comm(comm)
    searcher = CEM(prob, popsize=48, parenthood_ratio=0.25, stdev_init=1.5,
                   center_init=torch.ones(8) * 2, distributed=True, grad_chunk_rows=6)
    for _ in range(15):
        searcher.step()
    center_t = torch.Tensor.as_subclass(searcher.status["center"], torch.Tensor).clone()
    ref = center_t.clone()
    comm.broadcast_(ref, src=0)
    assert torch.allclose(center_t, ref, atol=1e-6), "ranks diverged"
    assert searcher.status["mean_eval"] < 8 * 4
    return center_t.tolist()


def _make_dist(prob):
    from evotorch_amd.distributions import SymmetricSeparableGaussian

    L = prob.solution_length
    return SymmetricSeparableGaussian({"mu": torch.zeros(L), "sigma": torch.ones(L) * 0.1})


def _trajectory(comm, rank, algo: str, *, ranking=None, steps=6, chunk_rows=None, num_interactions=None):
    """Run a short distributed search; rank-0's problem seed fixes the
    shared SPMD seed chain, so the trajectory must be IDENTICAL for any
    world size (counter-addressed global population + global weight
    normalization)."""
    from evotorch_amd import Problem
    from evotorch_amd.algorithms import CEM, PGPE, SNES, XNES
    from evotorch_amd.decorators import vectorized

    @vectorized
    def sphere(x):
        return (x**2).sum(-1)

    # deliberately IDENTICAL per-rank seeds: the sharded path owns disjoint
    # direction slices of one global virtual population, so equal seeding
    # must NOT collapse the shards (round-1 ADVICE low finding)
    prob = Problem("min", sphere, solution_length=7, initial_bounds=(-1, 1), seed=1234)
    if num_interactions is not None:
        # report a fake per-eval interaction count so the adaptive loop runs
        prob.last_eval_interaction_count = 0
        orig = prob._evaluate_batch

        def counting_eval(batch):
            orig(batch)
            prob.last_eval_interaction_count = len(batch)

        prob._evaluate_batch = counting_eval
    prob.use_comm(comm)
    kwargs = dict(distributed=True)
    if chunk_rows is not None:
        kwargs["grad_chunk_rows"] = chunk_rows
    if num_interactions is not None:
        kwargs["num_interactions"] = num_interactions
        kwargs["popsize_max"] = 200
    if algo == "pgpe":
        searcher = PGPE(prob, popsize=40, center_learning_rate=0.3, stdev_learning_rate=0.1,
                        stdev_init=1.0, center_init=torch.ones(7), ranking_method=ranking or "centered", **kwargs)
    elif algo == "snes":
        searcher = SNES(prob, popsize=40, stdev_init=1.0, center_init=torch.ones(7),
                        ranking_method=ranking or "nes", **kwargs)
    elif algo == "cem":
        searcher = CEM(prob, popsize=40, parenthood_ratio=0.25, stdev_init=1.0,
                       center_init=torch.ones(7) * 2, **kwargs)
    elif algo == "xnes":
        searcher = XNES(prob, popsize=40, stdev_init=1.0, center_init=torch.ones(7), **kwargs)
    else:
        raise ValueError(algo)
    for _ in range(steps):
        searcher.step()
    center = torch.Tensor.as_subclass(searcher.status["center"], torch.Tensor)
    return center.tolist()


def _body_traj_pgpe(comm, rank, world):
    return _trajectory(comm, rank, "pgpe")


def _body_traj_pgpe_raw(comm, rank, world):
    # 'raw' ranking exercises the global zero-centering (round-1 ADVICE
    # medium: per-shard centering changed the merged gradient)
    return _trajectory(comm, rank, "pgpe", ranking="raw")


def _body_traj_snes(comm, rank, world):
    return _trajectory(comm, rank, "snes")


def _body_traj_snes_centered(comm, rank, world):
    # SNES with non-'nes' ranking divides by the GLOBAL |w| sum
    return _trajectory(comm, rank, "snes", ranking="centered")


def _body_traj_cem(comm, rank, world):
    return _trajectory(comm, rank, "cem")


def _body_traj_xnes(comm, rank, world):
    return _trajectory(comm, rank, "xnes")


def _body_traj_pgpe_streamed(comm, rank, world):
    return _trajectory(comm, rank, "pgpe", chunk_rows=4)


def _body_traj_pgpe_interactions(comm, rank, world):
    # num_interactions above one round's worth: the SPMD adaptive loop must
    # take the same (multi-)round schedule at every world size
    return _trajectory(comm, rank, "pgpe", num_interactions=100, steps=4)


# -- tests -------------------------------------------------------------------


@pytest.mark.parametrize(
    "body",
    ["_body_collectives", "_body_sharded_evaluate", "_body_distributed_pgpe", "_body_obs_norm_allreduce", "_body_streamed_pgpe", "_body_streamed_cem"],
)
def test_world2(body):
    results = _run_world(body, world=2)
    assert len(results) == 2
    if body == "_body_sharded_evaluate":
        assert results[0] == results[1]  # both ranks hold the full eval vector
    if body in ("_body_distributed_pgpe", "_body_streamed_pgpe", "_body_streamed_cem"):
        assert results[0] == results[1]


@pytest.mark.parametrize(
    "body",
    [
        "_body_traj_pgpe",
        "_body_traj_pgpe_raw",
        "_body_traj_snes",
        "_body_traj_snes_centered",
        "_body_traj_cem",
        "_body_traj_xnes",
        "_body_traj_pgpe_streamed",
        "_body_traj_pgpe_interactions",
    ],
)
def test_world_size_invariant_trajectory(body):
    """The SPMD path samples ONE counter-addressed global population and
    normalizes weights globally, so the search trajectory is identical for
    any world size dividing the popsize — world 1 vs world 2 must agree to
    float tolerance (this is the single-process-equivalence guarantee the
    round-1 sharded path lacked)."""
    r1 = _run_world(body, world=1)
    r2 = _run_world(body, world=2)
    assert r2[0] == r2[1], "ranks diverged"
    c1 = torch.tensor(r1[0], dtype=torch.float64)
    c2 = torch.tensor(r2[0], dtype=torch.float64)
    assert torch.allclose(c1, c2, atol=1e-5, rtol=1e-5), f"world-1 vs world-2 trajectories differ:\n{c1}\n{c2}"


def _body_nsga2_sharded(comm, rank, world):
    """GA-family sharding contract: identical seeds on every rank, the
    searcher's own RNG reproduces the same populations, and evaluate()
    shards fitness rows across ranks (P1). The full NSGA-II trajectory
    must agree across ranks AND with a single-process run."""
    import torch as _t

    from evotorch_amd import Problem
    from evotorch_amd.algorithms import GeneticAlgorithm
    from evotorch_amd.decorators import vectorized
    from evotorch_amd.operators import PolynomialMutation, SimulatedBinaryCrossOver

    @vectorized
    def zdt1ish(x):
        f1 = x[..., 0]
        g = 1.0 + 9.0 * x[..., 1:].mean(-1)
        f2 = g * (1.0 - _t.sqrt(_t.clamp(f1 / g, min=0.0)))
        return _t.stack([f1, f2], dim=-1)

    prob = Problem(["min", "min"], zdt1ish, solution_length=6, initial_bounds=(0.0, 1.0),
                   bounds=(0.0, 1.0), seed=99)  # SAME seed on all ranks (GA contract)
    prob.use_comm(comm)
    ga = GeneticAlgorithm(
        prob, popsize=24,
        operators=[
            SimulatedBinaryCrossOver(prob, tournament_size=3, cross_over_rate=1.0, eta=8),
            PolynomialMutation(prob, eta=20, mutation_probability=0.25),
        ],
    )
    for _ in range(6):
        ga.step()
    vals = _t.Tensor.as_subclass(ga.population.values, _t.Tensor).clone()
    ref = vals.clone()
    comm.broadcast_(ref.reshape(-1), src=0)
    assert _t.allclose(vals, ref, atol=1e-6), "ranks diverged"
    return vals.reshape(-1).tolist()


def test_nsga2_sharded_evaluate_matches_single_process():
    r1 = _run_world("_body_nsga2_sharded", world=1)
    r2 = _run_world("_body_nsga2_sharded", world=2)
    v1 = torch.tensor(r1[0], dtype=torch.float64)
    v2 = torch.tensor(r2[0], dtype=torch.float64)
    assert torch.allclose(v1, v2, atol=1e-6), "sharded NSGA-II diverged from single-process"


def test_world4_matches_world1():
    """Same invariance at world size 4 (the popsize-40 shards stay even,
    mirroring a 4-GPU slice of the driver's scaling ladder)."""
    r1 = _run_world("_body_traj_pgpe", world=1)
    r4 = _run_world("_body_traj_pgpe", world=4)
    assert all(r4[0] == r4[i] for i in range(4)), "ranks diverged"
    c1 = torch.tensor(r1[0], dtype=torch.float64)
    c4 = torch.tensor(r4[0], dtype=torch.float64)
    assert torch.allclose(c1, c4, atol=1e-5, rtol=1e-5)


def test_world5_matches_world1():
    """Non-power-of-two world size (5 ranks, 8-direction shards of the
    40-member virtual population): catches any hidden power-of-two
    assumptions in the counter-addressed sharding."""
    r1 = _run_world("_body_traj_pgpe", world=1)
    r5 = _run_world("_body_traj_pgpe", world=5)
    assert all(r5[0] == r5[i] for i in range(5)), "ranks diverged"
    c1 = torch.tensor(r1[0], dtype=torch.float64)
    c5 = torch.tensor(r5[0], dtype=torch.float64)
    assert torch.allclose(c1, c5, atol=1e-5, rtol=1e-5)


def test_bench_entry_torchrun_world2(tmp_path):
    """The driver launches bench.py via torch.distributed.run; validate that
    exact entry path (world 2, gloo on CPU) end to end."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    result = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29877",
            os.path.join(repo, "bench.py"),
            "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--popsize-per-gpu", "16", "--episode-length", "5",
        ],
        cwd=repo,
        capture_output=True,
        text=True,
        timeout=300,
    )
    assert result.returncode == 0, result.stderr[-3000:]
    line = [l for l in result.stdout.splitlines() if l.startswith("{")][-1]
    payload = json.loads(line)
    assert payload["config"]["global_batch"] == 32
    assert payload["config"]["parallelism"] == "dp2"
    assert payload["value"] > 0


def test_bench_json_contract_single_process(tmp_path):
    """bench.py's output line carries every field the driver contract
    requires, with sane types (single-process CPU run)."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    result = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--steps", "2", "--warmup", "1",
         "--popsize-per-gpu", "16", "--episode-length", "4"],
        capture_output=True, text=True, timeout=300,
    )
    assert result.returncode == 0, result.stderr[-2000:]
    line = result.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key, typ in [("metric", str), ("value", float), ("unit", str), ("n_gpus", int),
                     ("steps", int), ("warmup", int), ("ms_per_step", float),
                     ("higher_is_better", bool), ("scaling", str), ("dtype", str),
                     ("data", str), ("config", dict)]:
        assert key in d and isinstance(d[key], typ), (key, type(d.get(key)))
    assert d["vs_baseline"] is None  # BASELINE.json publishes no numbers
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["scaling"] == "weak"
    assert "Humanoid-v4 linear policy" in d["metric"]
    assert d["config"]["global_batch"] == 16
