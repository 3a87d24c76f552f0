"""Numerical cross-validation against the reference implementation
(read-only at /root/reference): feed IDENTICAL populations/fitnesses into
both frameworks' ranking functions, distribution gradients
(Separable/Symmetric/ExpSeparable), and ClipUp/Adam optimizers, and report
max elementwise differences. Expected: <=1e-6 everywhere (the reference
casts rankings to float32 internally, so fp64 inputs bottom out at fp32
epsilon). Output recorded in profiles/crosscheck_reference.log.

Requires the reference tree; ray/gymnasium are stubbed just enough to
import its distributions/optimizers modules (no reference code executes
beyond the functions under test)."""

import os
import sys
import types

REFERENCE = "/root/reference/src"


def _stub(name, **attrs):
    m = types.ModuleType(name)
    for k, v in attrs.items():
        setattr(m, k, v)
    sys.modules[name] = m
    return m


def main():
    if not os.path.isdir(REFERENCE):
        print("reference tree not available; nothing to check")
        return

    class _T:  # generic placeholder type
        pass

    _stub("ray", remote=lambda *a, **k: (lambda f: f), is_initialized=lambda: False, ObjectRef=object)
    _stub("ray.util", ActorPool=object)
    _stub("ray.actor", ActorHandle=object)
    gym = _stub("gymnasium", Env=object, Wrapper=object, make=lambda *a, **k: None, __version__="1.0.0")
    spaces = _stub("gymnasium.spaces", Box=_T, Discrete=_T, MultiDiscrete=_T, Space=_T)
    gym.spaces = spaces
    vecutils = _stub("gymnasium.vector.utils", batch_space=lambda *a, **k: None)
    vec = _stub("gymnasium.vector", VectorEnv=_T, SyncVectorEnv=_T, utils=vecutils)
    gym.vector = vec
    _stub("gymnasium.wrappers")

    sys.path.insert(0, REFERENCE)
    import torch

    from evotorch.distributions import ExpSeparableGaussian as RefExp
    from evotorch.distributions import SeparableGaussian as RefSep
    from evotorch.distributions import SymmetricSeparableGaussian as RefSym
    from evotorch.optimizers import Adam as RefAdam
    from evotorch.optimizers import ClipUp as RefClipUp
    from evotorch.tools import ranking as ref_ranking

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from evotorch_amd.distributions import ExpSeparableGaussian as MyExp
    from evotorch_amd.distributions import SeparableGaussian as MySep
    from evotorch_amd.distributions import SymmetricSeparableGaussian as MySym
    from evotorch_amd.optimizers import Adam as MyAdam
    from evotorch_amd.optimizers import ClipUp as MyClipUp
    from evotorch_amd.utils import ranking as my_ranking

    torch.manual_seed(0)
    L, N = 13, 24
    report = []

    fit = torch.randn(N, dtype=torch.float64)
    for method in ["centered", "linear", "nes", "normalized", "raw"]:
        for hib in (True, False):
            r_ref = ref_ranking.rank(fit, ranking_method=method, higher_is_better=hib)
            r_my = my_ranking.rank(fit, method, higher_is_better=hib)
            report.append((f"rank/{method}/hib={int(hib)}", (r_ref.double() - r_my.double()).abs().max().item()))

    mu = torch.randn(L, dtype=torch.float64)
    sigma = torch.rand(L, dtype=torch.float64) + 0.5
    samples = mu + sigma * torch.randn(N, L, dtype=torch.float64)
    fitn = torch.randn(N, dtype=torch.float64)

    for div in [None, "num_solutions", "total_weight"]:
        params = {"mu": mu.clone(), "sigma": sigma.clone()}
        if div:
            params.update(divide_mu_grad_by=div, divide_sigma_grad_by=div)
        g_ref = RefSep(dict(params)).compute_gradients(samples, fitn, objective_sense="max", ranking_method="centered")
        g_my = MySep(dict(params)).compute_gradients(samples, fitn, objective_sense="max", ranking_method="centered")
        for k in sorted(g_ref):
            report.append((f"Sep/div={div}/{k}", (g_ref[k] - g_my[k]).abs().max().item()))

    # symmetric: reference interleaves (+,-) pairs, this framework uses the
    # halves layout — remap before comparing
    d_half = N // 2
    base = torch.randn(d_half, L, dtype=torch.float64)
    plus, minus = mu + sigma * base, mu - sigma * base
    ref_samples = torch.empty(N, L, dtype=torch.float64)
    ref_samples[0::2], ref_samples[1::2] = plus, minus
    my_samples = torch.cat([plus, minus])
    f_plus, f_minus = torch.randn(d_half, dtype=torch.float64), torch.randn(d_half, dtype=torch.float64)
    ref_fit = torch.empty(N, dtype=torch.float64)
    ref_fit[0::2], ref_fit[1::2] = f_plus, f_minus
    my_fit = torch.cat([f_plus, f_minus])
    params = {"mu": mu.clone(), "sigma": sigma.clone(),
              "divide_mu_grad_by": "num_directions", "divide_sigma_grad_by": "num_directions"}
    g_ref = RefSym(dict(params)).compute_gradients(ref_samples, ref_fit, objective_sense="max", ranking_method="centered")
    g_my = MySym(dict(params)).compute_gradients(my_samples, my_fit, objective_sense="max", ranking_method="centered")
    for k in sorted(g_ref):
        report.append((f"Sym/{k}", (g_ref[k] - g_my[k]).abs().max().item()))

    g_ref = RefExp({"mu": mu.clone(), "sigma": sigma.clone()}).compute_gradients(samples, fitn, objective_sense="min", ranking_method="nes")
    g_my = MyExp({"mu": mu.clone(), "sigma": sigma.clone()}).compute_gradients(samples, fitn, objective_sense="min", ranking_method="nes")
    for k in sorted(g_ref):
        report.append((f"ExpSep-nes/{k}", (g_ref[k] - g_my[k]).abs().max().item()))

    # stdev-control clamping (modify_tensor drives PGPE's stdev_max_change)
    from evotorch.tools.misc import modify_tensor as ref_modify
    from evotorch.tools.misc import stdev_from_radius as ref_sfr

    from evotorch_amd.utils import modify_tensor as my_modify
    from evotorch_amd.utils import stdev_from_radius as my_sfr

    orig = torch.rand(40, dtype=torch.float64) + 0.5
    target = orig * torch.empty(40, dtype=torch.float64).uniform_(0.3, 3.0)
    for kwargs in (dict(lb=0.6, ub=2.0), dict(max_change=0.2), dict(lb=0.7, ub=1.8, max_change=0.3)):
        a = ref_modify(orig, target, **kwargs)
        b = my_modify(orig, target, **kwargs)
        report.append((f"modify_tensor/{sorted(kwargs)}", (a - b).abs().max().item()))
    report.append(("stdev_from_radius", abs(ref_sfr(2.25, 6409) - my_sfr(2.25, 6409))))

    # constraint penalization helpers
    from evotorch.tools import constraints as ref_constraints

    from evotorch_amd.utils import constraints as my_constraints

    cx = torch.randn(64, dtype=torch.float64) * 3
    for op in ("<=", ">="):
        a = ref_constraints.violation(lhs=cx, comparison=op, rhs=1.0)
        b = my_constraints.violation(lhs=cx, comparison=op, rhs=1.0)
        report.append((f"violation/{op}", (a - b).abs().max().item()))
        a = ref_constraints.log_barrier(lhs=cx, comparison=op, rhs=1.0, penalty_sign="-")
        b = my_constraints.log_barrier(lhs=cx, comparison=op, rhs=1.0, penalty_sign="-")
        finite = torch.isfinite(a)
        assert torch.equal(torch.isfinite(b), finite)  # same -inf pattern
        report.append((f"log_barrier/{op}", (a[finite] - b[finite]).abs().max().item() if finite.any() else 0.0))
    a = ref_constraints.penalty(lhs=cx, comparison="<=", rhs=1.0, penalty_sign="-", linear=2.0, step=5.0, exp=2.0, exp_inf=100.0)
    b = my_constraints.penalty(lhs=cx, comparison="<=", rhs=1.0, penalty_sign="-", linear=2.0, step=5.0, exp=2.0, exp_inf=100.0)
    report.append(("penalty", (a - b).abs().max().item()))

    g_seq = [torch.randn(L, dtype=torch.float64) for _ in range(5)]
    pairs = [
        ("ClipUp/5steps",
         RefClipUp(solution_length=L, dtype=torch.float64, stepsize=0.1, max_speed=0.2),
         MyClipUp(solution_length=L, dtype=torch.float64, stepsize=0.1, max_speed=0.2)),
        ("Adam/5steps",
         RefAdam(solution_length=L, dtype=torch.float64, stepsize=0.01),
         MyAdam(solution_length=L, dtype=torch.float64, stepsize=0.01)),
    ]
    for name, ref_o, my_o in pairs:
        worst = 0.0
        for g in g_seq:
            worst = max(worst, (ref_o.ascent(g) - my_o.ascent(g)).abs().max().item())
        report.append((name, worst))

    failures = 0
    for n, d in report:
        status = "OK " if d <= 1e-6 else "FAIL"
        if d > 1e-6:
            failures += 1
        print(f"{status} {n:28s} maxdiff {d:.3e}")
    print("RESULT:", "all within 1e-6 of the reference" if failures == 0 else f"{failures} mismatches")
    sys.exit(1 if failures else 0)


if __name__ == "__main__":
    main()
