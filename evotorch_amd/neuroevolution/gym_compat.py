"""Minimal gymnasium-compatible layer with vendored classic-control envs.

The MI355X image ships no simulator, so the reference's real-environment
RL paths (GymNE per-solution rollouts, `gymne.py:361-414`; the vectorized
SyncVectorEnv with per-env autoreset and info batching,
`net/vecrl.py:1541-1664`) could otherwise never EXECUTE in CI. This module
closes that gap:

* When the real `gymnasium` package is installed it is used unchanged
  (`make`, `spaces` are re-exports).
* Otherwise a small vendored implementation provides the same API surface:
  `spaces.Box` / `spaces.Discrete`, `make(name)` over a registry of
  classic-control environments written here from their textbook dynamics
  (cart-pole per Barto, Sutton & Anderson 1983; torque-limited pendulum
  swing-up; mountain car per Moore 1990, discrete and continuous; acrobot
  per Sutton 1996 with RK4), and `SyncVectorEnv` with SAME-STEP AUTORESET: when a sub-env
  ends, the batched step returns the freshly reset observation while
  `info["final_observation"][i]` carries the terminal one — the contract
  of the reference's own vector env.

Everything speaks numpy at the boundary (like gymnasium) so the
torch-native adapters (`vecenv.GymVectorEnvAdapter`, `TorchWrapper`)
exercise their real conversion paths.
"""

import math
from typing import Callable, List, Optional, Sequence

import numpy as np

__all__ = [
    "AcrobotEnv",
    "Box",
    "CartPoleEnv",
    "Discrete",
    "MountainCarContinuousEnv",
    "MountainCarEnv",
    "PendulumEnv",
    "SyncVectorEnv",
    "have_real_gymnasium",
    "make",
    "make_vec",
    "register",
    "spaces",
]

try:
    import gymnasium as _real_gym
except ImportError:
    _real_gym = None


def have_real_gymnasium() -> bool:
    return _real_gym is not None


# ---------------------------------------------------------------------------
# spaces
# ---------------------------------------------------------------------------


class Box:
    """Continuous box space (gymnasium.spaces.Box surface subset)."""

    def __init__(self, low, high, shape=None, dtype=np.float32):
        if shape is None:
            shape = np.broadcast(np.asarray(low), np.asarray(high)).shape
        self.shape = tuple(shape)
        self.low = np.broadcast_to(np.asarray(low, dtype=dtype), self.shape).copy()
        self.high = np.broadcast_to(np.asarray(high, dtype=dtype), self.shape).copy()
        self.dtype = np.dtype(dtype)

    def sample(self, rng: Optional[np.random.Generator] = None):
        rng = rng or np.random.default_rng()
        lo = np.where(np.isfinite(self.low), self.low, -1.0)
        hi = np.where(np.isfinite(self.high), self.high, 1.0)
        return rng.uniform(lo, hi).astype(self.dtype)

    def contains(self, x) -> bool:
        x = np.asarray(x)
        return x.shape == self.shape and bool(np.all(x >= self.low - 1e-6) and np.all(x <= self.high + 1e-6))

    def __repr__(self):
        return f"Box(shape={self.shape}, dtype={self.dtype})"


class Discrete:
    """Discrete space (gymnasium.spaces.Discrete surface subset)."""

    def __init__(self, n: int):
        self.n = int(n)
        self.shape = ()
        self.dtype = np.dtype(np.int64)

    def sample(self, rng: Optional[np.random.Generator] = None):
        rng = rng or np.random.default_rng()
        return int(rng.integers(self.n))

    def contains(self, x) -> bool:
        return 0 <= int(x) < self.n

    def __repr__(self):
        return f"Discrete({self.n})"


class _SpacesModule:
    Box = Box
    Discrete = Discrete


spaces = _real_gym.spaces if _real_gym is not None else _SpacesModule()


# ---------------------------------------------------------------------------
# vendored classic-control environments (textbook dynamics)
# ---------------------------------------------------------------------------


class _EnvBase:
    metadata: dict = {}

    def __init__(self):
        self._rng = np.random.default_rng()

    def _seed(self, seed: Optional[int]):
        if seed is not None:
            self._rng = np.random.default_rng(int(seed))

    def close(self):
        pass


class CartPoleEnv(_EnvBase):
    """Cart-pole balancing (Barto, Sutton & Anderson 1983 dynamics).

    Observation: [x, ẋ, θ, θ̇]; action: Discrete(2) push left/right;
    reward 1 per step; terminates when |x| > 2.4 or |θ| > 12°; truncates
    at `max_episode_steps` (500, the -v1 budget)."""

    GRAVITY = 9.8
    MASS_CART = 1.0
    MASS_POLE = 0.1
    HALF_POLE_LEN = 0.5
    FORCE_MAG = 10.0
    TAU = 0.02
    X_LIMIT = 2.4
    THETA_LIMIT = 12.0 * math.pi / 180.0

    def __init__(self, max_episode_steps: int = 500):
        super().__init__()
        hi = np.array([self.X_LIMIT * 2, np.inf, self.THETA_LIMIT * 2, np.inf], dtype=np.float32)
        self.observation_space = Box(-hi, hi)
        self.action_space = Discrete(2)
        self._max_steps = int(max_episode_steps)
        self._state = np.zeros(4, dtype=np.float64)
        self._t = 0

    def reset(self, *, seed: Optional[int] = None, options=None):
        self._seed(seed)
        self._state = self._rng.uniform(-0.05, 0.05, size=4)
        self._t = 0
        return self._state.astype(np.float32), {}

    def step(self, action):
        x, x_dot, theta, theta_dot = self._state
        force = self.FORCE_MAG if int(action) == 1 else -self.FORCE_MAG
        total_mass = self.MASS_CART + self.MASS_POLE
        pole_ml = self.MASS_POLE * self.HALF_POLE_LEN
        cos_t, sin_t = math.cos(theta), math.sin(theta)
        tmp = (force + pole_ml * theta_dot**2 * sin_t) / total_mass
        theta_acc = (self.GRAVITY * sin_t - cos_t * tmp) / (
            self.HALF_POLE_LEN * (4.0 / 3.0 - self.MASS_POLE * cos_t**2 / total_mass)
        )
        x_acc = tmp - pole_ml * theta_acc * cos_t / total_mass
        # explicit Euler on the current derivatives (the classic integrator)
        x = x + self.TAU * x_dot
        x_dot = x_dot + self.TAU * x_acc
        theta = theta + self.TAU * theta_dot
        theta_dot = theta_dot + self.TAU * theta_acc
        self._state = np.array([x, x_dot, theta, theta_dot])
        self._t += 1
        terminated = bool(abs(x) > self.X_LIMIT or abs(theta) > self.THETA_LIMIT)
        truncated = bool(self._t >= self._max_steps)
        return self._state.astype(np.float32), 1.0, terminated, truncated, {}


class PendulumEnv(_EnvBase):
    """Torque-limited pendulum swing-up.

    Observation: [cos θ, sin θ, θ̇]; action: Box(-2, 2) torque; reward
    −(θ̄² + 0.1 θ̇² + 0.001 u²) with θ̄ the angle wrapped to [−π, π];
    never terminates, truncates at 200 steps."""

    G = 10.0
    M = 1.0
    L = 1.0
    DT = 0.05
    MAX_TORQUE = 2.0
    MAX_SPEED = 8.0

    def __init__(self, max_episode_steps: int = 200):
        super().__init__()
        hi = np.array([1.0, 1.0, self.MAX_SPEED], dtype=np.float32)
        self.observation_space = Box(-hi, hi)
        self.action_space = Box(np.array([-self.MAX_TORQUE], dtype=np.float32), np.array([self.MAX_TORQUE], dtype=np.float32))
        self._max_steps = int(max_episode_steps)
        self._theta = 0.0
        self._theta_dot = 0.0
        self._t = 0

    def _obs(self):
        return np.array([math.cos(self._theta), math.sin(self._theta), self._theta_dot], dtype=np.float32)

    def reset(self, *, seed: Optional[int] = None, options=None):
        self._seed(seed)
        self._theta = float(self._rng.uniform(-math.pi, math.pi))
        self._theta_dot = float(self._rng.uniform(-1.0, 1.0))
        self._t = 0
        return self._obs(), {}

    def step(self, action):
        u = float(np.clip(np.asarray(action).reshape(-1)[0], -self.MAX_TORQUE, self.MAX_TORQUE))
        th_wrapped = ((self._theta + math.pi) % (2 * math.pi)) - math.pi
        cost = th_wrapped**2 + 0.1 * self._theta_dot**2 + 0.001 * u**2
        theta_acc = 3.0 * self.G / (2.0 * self.L) * math.sin(self._theta) + 3.0 / (self.M * self.L**2) * u
        self._theta_dot = float(np.clip(self._theta_dot + theta_acc * self.DT, -self.MAX_SPEED, self.MAX_SPEED))
        self._theta = self._theta + self._theta_dot * self.DT
        self._t += 1
        truncated = bool(self._t >= self._max_steps)
        return self._obs(), -cost, False, truncated, {}


class MountainCarEnv(_EnvBase):
    """Under-powered car in a valley (Moore 1990 dynamics).

    Observation: [position, velocity]; action: Discrete(3) push
    left/none/right; reward −1 per step; terminates at the right hilltop
    (position ≥ 0.5); truncates at 200 steps."""

    MIN_POS, MAX_POS = -1.2, 0.6
    MAX_SPEED = 0.07
    GOAL_POS = 0.5
    FORCE = 0.001
    GRAVITY = 0.0025

    def __init__(self, max_episode_steps: int = 200):
        super().__init__()
        lo = np.array([self.MIN_POS, -self.MAX_SPEED], dtype=np.float32)
        hi = np.array([self.MAX_POS, self.MAX_SPEED], dtype=np.float32)
        self.observation_space = Box(lo, hi)
        self.action_space = Discrete(3)
        self._max_steps = int(max_episode_steps)
        self._pos = 0.0
        self._vel = 0.0
        self._t = 0

    def _obs(self):
        return np.array([self._pos, self._vel], dtype=np.float32)

    def reset(self, *, seed: Optional[int] = None, options=None):
        self._seed(seed)
        self._pos = float(self._rng.uniform(-0.6, -0.4))
        self._vel = 0.0
        self._t = 0
        return self._obs(), {}

    def step(self, action):
        self._vel += (int(action) - 1) * self.FORCE - self.GRAVITY * math.cos(3.0 * self._pos)
        self._vel = float(np.clip(self._vel, -self.MAX_SPEED, self.MAX_SPEED))
        self._pos = float(np.clip(self._pos + self._vel, self.MIN_POS, self.MAX_POS))
        if self._pos <= self.MIN_POS and self._vel < 0.0:
            self._vel = 0.0  # inelastic left wall
        self._t += 1
        terminated = bool(self._pos >= self.GOAL_POS)
        truncated = bool(self._t >= self._max_steps)
        return self._obs(), -1.0, terminated, truncated, {}


class MountainCarContinuousEnv(_EnvBase):
    """Continuous-torque mountain car.

    Observation: [position, velocity]; action: Box(−1, 1) engine force;
    reward +100 on reaching the goal minus 0.1·u² per step; terminates at
    position ≥ 0.45; truncates at 999 steps."""

    MIN_POS, MAX_POS = -1.2, 0.6
    MAX_SPEED = 0.07
    GOAL_POS = 0.45
    POWER = 0.0015
    GRAVITY = 0.0025

    def __init__(self, max_episode_steps: int = 999):
        super().__init__()
        lo = np.array([self.MIN_POS, -self.MAX_SPEED], dtype=np.float32)
        hi = np.array([self.MAX_POS, self.MAX_SPEED], dtype=np.float32)
        self.observation_space = Box(lo, hi)
        self.action_space = Box(np.array([-1.0], dtype=np.float32), np.array([1.0], dtype=np.float32))
        self._max_steps = int(max_episode_steps)
        self._pos = 0.0
        self._vel = 0.0
        self._t = 0

    def _obs(self):
        return np.array([self._pos, self._vel], dtype=np.float32)

    def reset(self, *, seed: Optional[int] = None, options=None):
        self._seed(seed)
        self._pos = float(self._rng.uniform(-0.6, -0.4))
        self._vel = 0.0
        self._t = 0
        return self._obs(), {}

    def step(self, action):
        u = float(np.clip(np.asarray(action).reshape(-1)[0], -1.0, 1.0))
        self._vel += u * self.POWER - self.GRAVITY * math.cos(3.0 * self._pos)
        self._vel = float(np.clip(self._vel, -self.MAX_SPEED, self.MAX_SPEED))
        self._pos = float(np.clip(self._pos + self._vel, self.MIN_POS, self.MAX_POS))
        if self._pos <= self.MIN_POS and self._vel < 0.0:
            self._vel = 0.0
        self._t += 1
        terminated = bool(self._pos >= self.GOAL_POS)
        truncated = bool(self._t >= self._max_steps)
        reward = (100.0 if terminated else 0.0) - 0.1 * u * u
        return self._obs(), reward, terminated, truncated, {}


class AcrobotEnv(_EnvBase):
    """Two-link underactuated pendulum swing-up (Sutton 1996 dynamics,
    book variant, RK4 integration).

    Observation: [cos θ1, sin θ1, cos θ2, sin θ2, θ̇1, θ̇2]; action:
    Discrete(3) torque {−1, 0, +1} on the SECOND joint; reward −1 per
    step; terminates when the tip rises above one link length
    (−cos θ1 − cos(θ1+θ2) > 1); truncates at 500 steps."""

    DT = 0.2
    M1 = M2 = 1.0
    L1 = 1.0
    LC1 = LC2 = 0.5
    I1 = I2 = 1.0
    G = 9.8
    MAX_VEL1 = 4.0 * math.pi
    MAX_VEL2 = 9.0 * math.pi

    def __init__(self, max_episode_steps: int = 500):
        super().__init__()
        hi = np.array([1.0, 1.0, 1.0, 1.0, self.MAX_VEL1, self.MAX_VEL2], dtype=np.float32)
        self.observation_space = Box(-hi, hi)
        self.action_space = Discrete(3)
        self._max_steps = int(max_episode_steps)
        self._s = np.zeros(4)
        self._t = 0

    def _obs(self):
        t1, t2, d1, d2 = self._s
        return np.array([math.cos(t1), math.sin(t1), math.cos(t2), math.sin(t2), d1, d2], dtype=np.float32)

    def reset(self, *, seed: Optional[int] = None, options=None):
        self._seed(seed)
        self._s = self._rng.uniform(-0.1, 0.1, size=4)
        self._t = 0
        return self._obs(), {}

    def _dsdt(self, s, tau):
        m1, m2, l1, lc1, lc2, i1, i2, g = self.M1, self.M2, self.L1, self.LC1, self.LC2, self.I1, self.I2, self.G
        t1, t2, dt1, dt2 = s
        d1 = m1 * lc1**2 + m2 * (l1**2 + lc2**2 + 2 * l1 * lc2 * math.cos(t2)) + i1 + i2
        d2 = m2 * (lc2**2 + l1 * lc2 * math.cos(t2)) + i2
        phi2 = m2 * lc2 * g * math.cos(t1 + t2 - math.pi / 2.0)
        phi1 = (
            -m2 * l1 * lc2 * dt2**2 * math.sin(t2)
            - 2 * m2 * l1 * lc2 * dt2 * dt1 * math.sin(t2)
            + (m1 * lc1 + m2 * l1) * g * math.cos(t1 - math.pi / 2.0)
            + phi2
        )
        # the "book" formulation (Sutton & Barto): solve the second joint
        # acceleration first, then back-substitute
        ddt2 = (tau + d2 / d1 * phi1 - m2 * l1 * lc2 * dt1**2 * math.sin(t2) - phi2) / (
            m2 * lc2**2 + i2 - d2**2 / d1
        )
        ddt1 = -(d2 * ddt2 + phi1) / d1
        return np.array([dt1, dt2, ddt1, ddt2])

    @staticmethod
    def _wrap(x, lo, hi):
        rng = hi - lo
        while x > hi:
            x -= rng
        while x < lo:
            x += rng
        return x

    def step(self, action):
        tau = float(int(action) - 1)
        s = self._s
        # one RK4 step of the autonomous dynamics with constant torque
        k1 = self._dsdt(s, tau)
        k2 = self._dsdt(s + 0.5 * self.DT * k1, tau)
        k3 = self._dsdt(s + 0.5 * self.DT * k2, tau)
        k4 = self._dsdt(s + self.DT * k3, tau)
        ns = s + self.DT / 6.0 * (k1 + 2 * k2 + 2 * k3 + k4)
        ns[0] = self._wrap(ns[0], -math.pi, math.pi)
        ns[1] = self._wrap(ns[1], -math.pi, math.pi)
        ns[2] = float(np.clip(ns[2], -self.MAX_VEL1, self.MAX_VEL1))
        ns[3] = float(np.clip(ns[3], -self.MAX_VEL2, self.MAX_VEL2))
        self._s = ns
        self._t += 1
        terminated = bool(-math.cos(ns[0]) - math.cos(ns[1] + ns[0]) > 1.0)
        truncated = bool(self._t >= self._max_steps)
        return self._obs(), -1.0, terminated, truncated, {}


# ---------------------------------------------------------------------------
# registry + make
# ---------------------------------------------------------------------------

_REGISTRY = {
    "CartPole-v1": lambda **kw: CartPoleEnv(**kw),
    "Pendulum-v1": lambda **kw: PendulumEnv(**kw),
    "MountainCar-v0": lambda **kw: MountainCarEnv(**kw),
    "MountainCarContinuous-v0": lambda **kw: MountainCarContinuousEnv(**kw),
    "Acrobot-v1": lambda **kw: AcrobotEnv(**kw),
}


def register(name: str, factory: Callable):
    """Register an additional vendored environment factory."""
    _REGISTRY[name] = factory


def make(name: str, **kwargs):
    """gymnasium.gym.make replacement: real gymnasium if installed, the
    vendored registry otherwise."""
    if _real_gym is not None:
        try:
            return _real_gym.make(name, **kwargs)
        except Exception:
            pass  # fall through to the vendored registry
    if name not in _REGISTRY:
        raise KeyError(f"Unknown environment {name!r}; vendored registry has {sorted(_REGISTRY)} and gymnasium is "
                       f"{'installed' if _real_gym is not None else 'not installed'}")
    return _REGISTRY[name](**kwargs)


# ---------------------------------------------------------------------------
# SyncVectorEnv with same-step autoreset + info batching
# ---------------------------------------------------------------------------


class SyncVectorEnv:
    """Sequential vector env with per-env autoreset (the contract of the
    reference's own SyncVectorEnv, net/vecrl.py:1541-1664): when sub-env i
    ends, step() returns the freshly reset observation in row i while
    ``info["final_observation"][i]`` holds the terminal observation and
    ``info["_final_observation"][i]`` is True. Scalar per-env info values
    are batched into arrays under their key."""

    def __init__(self, env_fns: Sequence[Callable[[], object]]):
        self.envs: List = [fn() for fn in env_fns]
        if not self.envs:
            raise ValueError("SyncVectorEnv needs at least one environment")
        self.num_envs = len(self.envs)
        self.single_observation_space = self.envs[0].observation_space
        self.single_action_space = self.envs[0].action_space
        self.observation_space = self.single_observation_space
        self.action_space = self.single_action_space

    def reset(self, *, seed: Optional[int] = None, options=None):
        obs_rows = []
        infos = {}
        for i, env in enumerate(self.envs):
            env_seed = None if seed is None else int(seed) + i
            obs, info = env.reset(seed=env_seed, options=options)
            obs_rows.append(np.asarray(obs))
            self._merge_info(infos, info, i)
        return np.stack(obs_rows), infos

    def step(self, actions):
        obs_rows, rewards, terms, truncs = [], [], [], []
        infos = {}
        final_obs = [None] * self.num_envs
        final_mask = np.zeros(self.num_envs, dtype=bool)
        for i, env in enumerate(self.envs):
            action = actions[i]
            obs, reward, terminated, truncated, info = env.step(action)
            if terminated or truncated:
                final_obs[i] = np.asarray(obs)
                final_mask[i] = True
                obs, reset_info = env.reset()
                self._merge_info(infos, reset_info, i)
            obs_rows.append(np.asarray(obs))
            rewards.append(float(reward))
            terms.append(bool(terminated))
            truncs.append(bool(truncated))
            self._merge_info(infos, info, i)
        if final_mask.any():
            infos["final_observation"] = np.array(final_obs, dtype=object)
            infos["_final_observation"] = final_mask
        return (
            np.stack(obs_rows),
            np.asarray(rewards, dtype=np.float32),
            np.asarray(terms),
            np.asarray(truncs),
            infos,
        )

    def _merge_info(self, infos: dict, info: dict, index: int):
        for k, v in (info or {}).items():
            if k not in infos:
                infos[k] = np.zeros(self.num_envs, dtype=np.asarray(v).dtype if not isinstance(v, str) else object)
                infos["_" + k] = np.zeros(self.num_envs, dtype=bool)
            infos[k][index] = v
            infos["_" + k][index] = True

    def close(self):
        for env in self.envs:
            env.close()


def make_vec(name: str, *, num_envs: int, **kwargs):
    """gymnasium.make_vec replacement over the vendored SyncVectorEnv."""
    if _real_gym is not None:
        try:
            return _real_gym.make_vec(name, num_envs=num_envs, **kwargs)
        except Exception:
            pass
    return SyncVectorEnv([(lambda: make(name, **kwargs)) for _ in range(num_envs)])
