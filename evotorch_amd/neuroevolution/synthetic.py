"""SyntheticRolloutProblem: the VecGymNE-equivalent over the offline
synthetic environment — the flagship benchmark problem.

Reference parity: the role of VecGymNE
(/root/reference/src/evotorch/neuroevolution/vecgymne.py:95-1073) — a
whole-population batched rollout with on-device observation normalization —
with the env replaced by `SyntheticEnvSpec` (no simulator exists offline)
and the rollout executed by the fused gfx950 kernel
(evotorch_amd/ops/hip/rollout.hip) in ONE launch per generation.
"""

from typing import Optional

import torch

from ..core import Problem, SolutionBatch
from .runningnorm import ObsNormLayer, RunningNorm
from .synthetic_env import SyntheticEnvSpec, rollout_eager

__all__ = ["SyntheticRolloutProblem"]


class SyntheticRolloutProblem(Problem):
    def __init__(
        self,
        *,
        spec: Optional[SyntheticEnvSpec] = None,
        device="cpu",
        seed: Optional[int] = None,
        observation_normalization: bool = True,
        decrease_rewards_by: float = 0.0,
        **spec_kwargs,
    ):
        if spec is None:
            spec = SyntheticEnvSpec(device=device, **spec_kwargs)
        self._spec = spec
        super().__init__(
            objective_sense="max",
            solution_length=spec.solution_length,
            initial_bounds=(-0.01, 0.01),
            dtype=torch.float32,
            device=device,
            seed=seed,
            store_solution_stats=False,
        )
        self._obs_norm_enabled = bool(observation_normalization)
        self._obs_norm = RunningNorm(shape=spec.obs_dim, device=device)
        self._decrease_rewards_by = float(decrease_rewards_by)
        self._pending_stats = None
        self.last_eval_interaction_count = 0
        self._total_interactions = 0
        self._episode_count = 0
        self.after_eval_hook.append(self._after_eval_status_getter)

    # -- properties ----------------------------------------------------------

    @property
    def spec(self) -> SyntheticEnvSpec:
        return self._spec

    @property
    def obs_norm(self) -> RunningNorm:
        return self._obs_norm

    def _norm_mean_std(self):
        if self._obs_norm_enabled and self._obs_norm.has_data:
            return self._obs_norm.mean, self._obs_norm.stdev
        O = self._spec.obs_dim
        return (
            torch.zeros(O, dtype=torch.float32, device=self._device),
            torch.ones(O, dtype=torch.float32, device=self._device),
        )

    def observation_normalization_data(self):
        mean, std = self._norm_mean_std()
        return {"mean": mean.cpu(), "stdev": std.cpu(), "count": self._obs_norm.count}

    # -- evaluation ----------------------------------------------------------

    def enable_graph_mode(self) -> None:
        """Switch episode seeding to a DEVICE splitmix64 chain so that
        `_evaluate_batch` is hipGraph-capturable (a host-derived seed
        would be frozen into the capture and every replay would rerun the
        same episodes). Called by GraphedSearch; the chain starts from
        the problem's own generator."""
        if self._device.type != "cuda":
            return
        if getattr(self, "_graph_seed_buf", None) is None:
            from ..ops.dispatch import _seed_from_generator

            seed0 = _seed_from_generator(self._generator, self._device)
            self._graph_seed_buf = torch.tensor([seed0], dtype=torch.int64, device=self._device)

    def _evaluate_batch(self, batch: SolutionBatch):
        values = batch.access_values(keep_evals=True)
        n = len(batch)
        spec = self._spec
        from ..ops.dispatch import _seed_from_generator

        seed_buf = getattr(self, "_graph_seed_buf", None)
        init_seed = 0 if seed_buf is not None else _seed_from_generator(self._generator, self._device) & 0x7FFFFFFF
        member_offset = 0
        comm = self._comm
        if comm is not None and comm.world_size > 1:
            # distinct episode-init streams per rank-sharded member; the
            # per-rank init_seed difference is environment stochasticity
            member_offset = comm.rank * n
        if values.device.type == "cuda":
            from .. import ops

            mod = ops.hip_required()
            mean, std = self._norm_mean_std()
            blob = spec.env_blob(mean, std, device=values.device)
            obs_stats = torch.zeros(2 * spec.obs_dim, dtype=torch.float32, device=values.device)
            if seed_buf is not None:
                mod.bump_seed(seed_buf)  # fresh episode seed per replay, on device
            fitness = mod.rollout_linear(
                values.contiguous(),
                blob,
                obs_stats,
                spec.obs_dim,
                spec.act_dim,
                spec.rank,
                spec.episode_length,
                spec.alive_bonus,
                spec.act_cost,
                init_seed,
                member_offset,
                spec.policy_hidden,
                seed_buf,
            )
            triple = (float(n) * spec.episode_length, obs_stats[: spec.obs_dim], obs_stats[spec.obs_dim :])
        else:
            mean, std = self._norm_mean_std()
            fitness, triple = rollout_eager(
                spec, values.to(torch.float32), mean, std, init_seed=init_seed, member_offset=member_offset
            )
        if self._decrease_rewards_by != 0.0:
            fitness = fitness - self._decrease_rewards_by * spec.episode_length
        batch.set_evals(fitness.to(self._eval_dtype))
        self._pending_stats = triple
        self.last_eval_interaction_count = n * spec.episode_length
        self._total_interactions += n * spec.episode_length
        self._episode_count += n

    def _after_eval_status_getter(self, batch) -> dict:
        self._merge_pending_stats()
        return {
            "total_interaction_count": self._total_interactions,
            "total_episode_count": self._episode_count,
        }

    def _merge_pending_stats(self):
        """Fold the rollout's (count, Σ, Σ²) into the running norm — with an
        attached Comm this is ONE all-reduce across ranks (P5), and during a
        sharded gradient generation it is FUSED into the gradient all-reduce
        (Problem.request_fused_reduce) so a generation costs exactly two
        collectives: fitness all-gather + fused all-reduce."""
        if self._pending_stats is None or not self._obs_norm_enabled:
            self._pending_stats = None
            return
        count, s, ss = self._pending_stats
        self._pending_stats = None
        comm = self._comm
        if comm is not None and comm.world_size > 1:
            packed = torch.cat([torch.tensor([count], dtype=torch.float32, device=s.device), s.reshape(-1), ss.reshape(-1)])
            O = self._spec.obs_dim

            def merge(reduced: torch.Tensor):
                # count stays a device tensor: no host sync
                self._obs_norm.update((reduced[0], reduced[1 : 1 + O], reduced[1 + O :]))

            self.request_fused_reduce(packed, merge)
        else:
            self._obs_norm.update((count, s, ss))

    # -- policy export -------------------------------------------------------

    def to_policy(self, x: torch.Tensor) -> torch.nn.Module:
        spec = self._spec
        O, A, H = spec.obs_dim, spec.act_dim, spec.policy_hidden
        x = torch.as_tensor(x, dtype=torch.float32).detach().cpu().reshape(-1)
        mean, std = self._norm_mean_std()
        layers = []
        if self._obs_norm_enabled:
            layers.append(ObsNormLayer(mean.cpu(), std.cpu()))
        with torch.no_grad():
            if H > 0:
                l1 = torch.nn.Linear(O, H)
                l2 = torch.nn.Linear(H, A)
                off = 0
                l1.weight.copy_(x[off : off + H * O].reshape(H, O)); off += H * O
                l1.bias.copy_(x[off : off + H]); off += H
                l2.weight.copy_(x[off : off + A * H].reshape(A, H)); off += A * H
                l2.bias.copy_(x[off:])
                layers += [l1, torch.nn.Tanh(), l2, torch.nn.Hardtanh()]
            else:
                linear = torch.nn.Linear(O, A)
                linear.weight.copy_(x[: A * O].reshape(A, O))
                linear.bias.copy_(x[A * O :])
                layers += [linear, torch.nn.Hardtanh()]
        policy = torch.nn.Sequential(*layers)
        policy.requires_grad_(False)  # inference artifact
        return policy
