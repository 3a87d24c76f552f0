"""GymNE: classic per-solution episode rollouts in (gymnasium) envs.

Reference parity: /root/reference/src/evotorch/neuroevolution/
gymne.py:64-730 — one env per worker, per-solution rollout loop, CPU-side
observation normalization (RunningNorm here plays the RunningStat role),
alive-bonus removal via `decrease_rewards_by`, episode/interaction
counters, and `to_policy` exporting ObsNorm + ActClip wrapped modules.

Environments come from real gymnasium when installed, otherwise from the
vendored classic-control registry (`gym_compat`) — the rollout loop runs
real environment dynamics either way.
"""

from typing import Callable, Optional, Union

import torch
from torch import nn

from ..core import Solution, SolutionBatch
from ..models import ensure_stateful
from .neproblem import NEProblem
from .runningnorm import RunningNorm

__all__ = [
    "ActClipLayer",
    "ActClipWrapperModule",
    "AliveBonusScheduleWrapper",
    "GymNE",
    "ObsNormWrapperModule",
    "ensure_space_types",
    "reset_env",
    "take_step_in_env",
]


class ActClipLayer(nn.Module):
    """Clips actions into the env's action-space box (reference
    net/rl.py:130)."""

    def __init__(self, lb, ub):
        super().__init__()
        self.register_buffer("lb", torch.as_tensor(lb, dtype=torch.float32))
        self.register_buffer("ub", torch.as_tensor(ub, dtype=torch.float32))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return torch.clamp(x, self.lb, self.ub)


# reference net/rl.py exposes these under different names; aliases keep
# imports from the reference working unchanged
from .runningnorm import ObsNormLayer as ObsNormWrapperModule  # noqa: E402

ActClipWrapperModule = ActClipLayer


def ensure_space_types(env) -> None:
    """Require Box observation and action spaces (reference
    gymne.py: ensure_space_types). Accepts both real gymnasium Box spaces
    and the vendored gym_compat.Box."""
    from . import gym_compat

    box_types = [gym_compat.Box]
    try:
        import gymnasium as gym

        box_types.append(gym.spaces.Box)
    except ImportError:
        pass
    box_types = tuple(box_types)
    if not isinstance(env.observation_space, box_types):
        raise TypeError(f"Unsupported observation space {env.observation_space!r}: expected a Box space")
    if not isinstance(env.action_space, box_types):
        raise TypeError(f"Unsupported action space {env.action_space!r}: expected a Box space")


def reset_env(env):
    """Version-tolerant env reset: returns the observation alone whether the
    env follows the old (obs) or new (obs, info) API (reference
    net/rl.py:63)."""
    result = env.reset()
    if isinstance(result, tuple) and len(result) == 2:
        return result[0]
    return result


def take_step_in_env(env, action):
    """Version-tolerant env step: normalizes 4-tuple (obs, reward, done,
    info) and 5-tuple (obs, reward, terminated, truncated, info) step APIs
    to (obs, reward, done, info) (reference net/rl.py:83)."""
    result = env.step(action)
    if len(result) == 5:
        obs, reward, terminated, truncated, info = result
        return obs, reward, bool(terminated) or bool(truncated), info
    return result


class AliveBonusScheduleWrapper:
    """Env wrapper subtracting the constant alive bonus early in training
    and restoring it on a schedule (t0, t1[, bonus]) — by step t0 nothing
    is given back, between t0 and t1 the bonus ramps linearly, after t1 the
    full bonus is added (reference net/rl.py:199)."""

    def __init__(self, env, alive_bonus_schedule):
        self.env = env
        if len(alive_bonus_schedule) == 3:
            self._t0, self._t1, self._bonus = alive_bonus_schedule
        else:
            self._t0, self._t1 = alive_bonus_schedule
            self._bonus = 1.0
        self._t = 0

    def __getattr__(self, name):
        return getattr(self.env, name)

    def reset(self, *args, **kwargs):
        self._t = 0
        return self.env.reset(*args, **kwargs)

    def step(self, action):
        result = self.env.step(action)
        self._t += 1
        if self._t < self._t0:
            scale = 0.0
        elif self._t >= self._t1:
            scale = 1.0
        else:
            scale = (self._t - self._t0) / (self._t1 - self._t0)
        reward = result[1] + scale * self._bonus
        return (result[0], reward) + tuple(result[2:])


class GymNE(NEProblem):
    def __init__(
        self,
        env: Optional[Union[str, Callable]] = None,
        network: Optional[Union[str, nn.Module, Callable]] = None,
        *,
        env_name: Optional[str] = None,
        network_args: Optional[dict] = None,
        env_config: Optional[dict] = None,
        observation_normalization: bool = False,
        num_episodes: int = 1,
        episode_length: Optional[int] = None,
        decrease_rewards_by: Optional[float] = None,
        alive_bonus_schedule: Optional[tuple] = None,
        action_noise_stdev: Optional[float] = None,
        initial_bounds=(-0.00001, 0.00001),
        seed: Optional[int] = None,
        num_actors=None,
        actor_config=None,
        num_subbatches=None,
        subbatch_size=None,
    ):
        if env is None:
            env = env_name
        if env is None:
            raise ValueError("Provide env (or env_name)")
        self._env_def = env
        self._env_config = dict(env_config or {})
        self._env = None
        probe = self._make_env()
        self._obs_dim = int(probe.observation_space.shape[0])
        self._act_dim = int(probe.action_space.shape[0]) if hasattr(probe.action_space, "shape") and probe.action_space.shape else int(probe.action_space.n)
        self._act_space = probe.action_space
        self._env = probe
        self._num_episodes = int(num_episodes)
        self._episode_length = episode_length
        self._decrease_rewards_by = 0.0 if decrease_rewards_by is None else float(decrease_rewards_by)
        self._alive_bonus_schedule = alive_bonus_schedule
        self._action_noise_stdev = action_noise_stdev
        self._obs_norm_enabled = bool(observation_normalization)
        super().__init__("max", network, network_args=network_args, initial_bounds=initial_bounds, seed=seed, store_solution_stats=False,
                         num_actors=num_actors, actor_config=actor_config, num_subbatches=num_subbatches, subbatch_size=subbatch_size)
        self._obs_norm = RunningNorm(shape=self._obs_dim, device="cpu")
        self.last_eval_interaction_count = 0
        self._total_interactions = 0
        self._episode_count = 0
        self._pending_stats: Optional[RunningNorm] = None
        self.after_eval_hook.append(
            lambda b: {"total_interaction_count": self._total_interactions, "total_episode_count": self._episode_count}
        )

    def _network_constants(self) -> dict:
        return {"obs_length": self._obs_dim, "act_length": self._act_dim, "obs_space": getattr(self._env, "observation_space", None), "act_space": self._act_space}

    def _make_env(self):
        # real gymnasium when installed, the vendored classic-control
        # registry (gym_compat) otherwise — the rollout loop runs real
        # environment dynamics either way
        from . import gym_compat

        if isinstance(self._env_def, str):
            return gym_compat.make(self._env_def, **self._env_config)
        return self._env_def(**self._env_config)

    @property
    def obs_norm(self) -> RunningNorm:
        return self._obs_norm

    def observation_normalization_data(self):
        return {"mean": self._obs_norm.mean, "stdev": self._obs_norm.stdev, "count": self._obs_norm.count}

    # -- actor sync protocol (reference gymne.py / core.py:2239-2340) ---------

    def get_observation_stats(self) -> RunningNorm:
        return self._obs_norm

    def set_observation_stats(self, rn) -> None:
        """Replace this problem's stats (main → worker sync)."""
        self._obs_norm.reset()
        self._obs_norm.update(rn if not isinstance(rn, RunningNorm) else rn.stats_triple())

    def update_observation_stats(self, rn) -> RunningNorm:
        """Merge another RunningNorm (or (count, sum, sumsq) triple) into
        this problem's stats (worker → main sync)."""
        self._obs_norm.update(rn if not isinstance(rn, RunningNorm) else rn.stats_triple())
        return self._obs_norm

    def pop_observation_stats(self) -> Optional[RunningNorm]:
        """Take (and clear) the stats collected since the last pop — what a
        worker ships back to the main process."""
        pending = self._pending_stats
        self._pending_stats = None
        return pending

    def _alive_bonus(self, t: int) -> float:
        if self._alive_bonus_schedule is None:
            return 0.0
        t0, t1, bonus = self._alive_bonus_schedule
        if t < t0:
            return 0.0
        if t >= t1:
            return float(bonus)
        return float(bonus) * (t - t0) / max(1, (t1 - t0))

    def _rollout(self, policy: nn.Module) -> float:
        import numpy as np

        env = self._env
        total = 0.0
        obs, _ = env.reset(seed=None)
        t = 0
        policy = ensure_stateful(policy)
        policy.reset()
        while True:
            obs_t = torch.as_tensor(np.asarray(obs), dtype=torch.float32)
            if self._obs_norm_enabled:
                if self._pending_stats is None:
                    self._pending_stats = RunningNorm(shape=self._obs_dim, device="cpu")
                self._pending_stats.update(obs_t)  # delta since last pop (actor sync protocol)
                obs_in = self._obs_norm.update_and_normalize(obs_t)
            else:
                obs_in = obs_t
            with torch.no_grad():
                act = policy(obs_in)
            if self._action_noise_stdev is not None:
                act = act + torch.randn_like(act) * self._action_noise_stdev
            if hasattr(self._act_space, "low"):
                act = torch.clamp(act, torch.as_tensor(self._act_space.low), torch.as_tensor(self._act_space.high))
                action = np.asarray(act)
            else:
                action = int(act.argmax())
            obs, reward, terminated, truncated, _ = env.step(action)
            total += float(reward) - self._decrease_rewards_by + self._alive_bonus(t)
            t += 1
            self._total_interactions += 1
            self.last_eval_interaction_count += 1
            if terminated or truncated:
                break
            if self._episode_length is not None and t >= self._episode_length:
                break
        self._episode_count += 1
        return total

    def _evaluate_network(self, network: nn.Module) -> float:
        total = 0.0
        for _ in range(self._num_episodes):
            total += self._rollout(network)
        return total / self._num_episodes

    def _evaluate_batch(self, batch: SolutionBatch):
        self.last_eval_interaction_count = 0
        super()._evaluate_batch(batch)

    def run(self, solution, *, render: bool = False) -> float:
        """Roll out one solution (optionally rendering)."""
        net = self.parameterize_net(torch.as_tensor(solution if not isinstance(solution, Solution) else torch.Tensor.as_subclass(solution.values, torch.Tensor)))
        return self._rollout(net)

    def to_policy(self, x, *, clip_actions: bool = True) -> nn.Module:
        module = self.make_net(x)
        layers = []
        if self._obs_norm_enabled and self._obs_norm.has_data:
            layers.append(self._obs_norm.to_layer())
        layers.append(module)
        if clip_actions and hasattr(self._act_space, "low"):
            layers.append(ActClipLayer(self._act_space.low, self._act_space.high))
        policy = nn.Sequential(*layers)
        policy.requires_grad_(False)  # inference artifact
        return policy
