"""Vectorized torch-native environments + the VecEnvNE problem (the
VecGymNE-equivalent).

Reference parity: /root/reference/src/evotorch/neuroevolution/
vecgymne.py:95-1073 and net/vecrl.py (TorchWrapper :362, SyncVectorEnv
:1541). Since no simulator ships offline, the environment contract here is
torch-native from the start:

    env.reset(seed) -> obs (num_envs, obs_dim)
    env.step(actions) -> (obs, reward, done)          all torch tensors

`SyntheticTorchEnv` implements the contract over `SyntheticEnvSpec`;
`GymVectorEnvAdapter` (import-guarded) wraps a gymnasium vector env. The
VecEnvNE hot loop is SURVEY.md §3.4: one vmapped population forward per
simulator step, masked reward accumulation, masked recurrent-state resets,
on-device RunningNorm.
"""

from typing import Callable, Optional, Union

import torch
from torch import nn

from ..core import SolutionBatch
from ..models import Policy
from .neproblem import NEProblem
from .runningnorm import RunningNorm
from .synthetic_env import SyntheticEnvSpec

__all__ = [
    "GymVectorEnvAdapter",
    "SyntheticTorchEnv",
    "TorchWrapper",
    "VecEnvNE",
    "VecGymNE",
    "array_type",
    "convert_from_torch",
    "convert_to_torch",
    "convert_to_torch_bool",
    "is_brax_env",
    "jax_to_torch",
    "make_brax_env",
    "make_gym_env",
    "make_vector_env",
    "torch_to_jax",
]


class SyntheticTorchEnv:
    """Batched torch env over the synthetic low-rank dynamics."""

    def __init__(self, spec: Optional[SyntheticEnvSpec] = None, *, num_envs: int, device="cpu", **spec_kwargs):
        self.spec = spec if spec is not None else SyntheticEnvSpec(**spec_kwargs)
        self.num_envs = int(num_envs)
        self.device = torch.device(device)
        self.obs_dim = self.spec.obs_dim
        self.act_dim = self.spec.act_dim
        s = self.spec
        self._V = s.V.to(self.device)
        self._U_T = s.U_T.to(self.device)
        self._D2_T = s.D2_T.to(self.device)
        self._c = s.c.to(self.device)
        self._wr = s.wr.to(self.device)
        self._obs: Optional[torch.Tensor] = None
        self._t = 0

    # Cross-step device state for hipGraph capture (VecEnvNE
    # use_hip_graph=True): the capture machinery re-homes these tensors
    # into static buffers. The host counter self._t is restored by
    # reset(), which runs before every capture/replay.
    graph_state_attrs = ("_obs",)

    def reset(self, seed: int = 0) -> torch.Tensor:
        self._obs = self.spec.initial_obs(self.num_envs, 0, int(seed), device=self.device)
        self._t = 0
        return self._obs

    def step(self, actions: torch.Tensor):
        s = self.spec
        act = torch.clamp(actions.to(self.device), -1.0, 1.0)
        h = self._obs @ self._V.T
        o_new = torch.tanh(h @ self._U_T + act @ self._D2_T + self._c)
        reward = o_new @ self._wr + s.alive_bonus - s.act_cost * (act**2).sum(-1) / s.act_dim
        self._obs = o_new
        self._t += 1
        done = torch.full((self.num_envs,), self._t >= s.episode_length, dtype=torch.bool, device=self.device)
        return o_new, reward, done


class GymVectorEnvAdapter:
    """Wraps a vector env with the gymnasium 5-tuple step API (real
    gymnasium's, or gym_compat.SyncVectorEnv's with per-env autoreset +
    info batching) into the torch contract (reference net/vecrl.py:362
    TorchWrapper)."""

    def __init__(self, env, *, device="cpu"):
        import numpy as np  # noqa: F401

        self._env = env
        self.device = torch.device(device)
        self.num_envs = env.num_envs
        self.obs_dim = int(env.single_observation_space.shape[0])
        self.act_dim = int(env.single_action_space.shape[0])

    def reset(self, seed: int = 0) -> torch.Tensor:
        obs, _ = self._env.reset(seed=int(seed))
        return torch.as_tensor(obs, dtype=torch.float32, device=self.device)

    def step(self, actions: torch.Tensor):
        import numpy as np

        obs, reward, terminated, truncated, _ = self._env.step(np.asarray(actions.detach().cpu()))
        return (
            torch.as_tensor(obs, dtype=torch.float32, device=self.device),
            torch.as_tensor(reward, dtype=torch.float32, device=self.device),
            torch.as_tensor(np.logical_or(terminated, truncated), dtype=torch.bool, device=self.device),
        )


_CAPTURE_STREAMS: dict = {}


def _capture_stream(device: torch.device) -> "torch.cuda.Stream":
    """One persistent stream per device for hipGraph warmup AND capture.
    torch.cuda.graph() otherwise captures on a FRESH internal stream each
    time; the BLAS workspace the first GEMM allocates for that new stream
    then lives inside the captured graph's private pool and is cached by
    the BLAS handle forever — pinning the whole pool after the graph
    dies (measured ~250 MiB leaked per capture). With a persistent
    stream, the workspace is allocated once during warmup, OUTSIDE any
    capture."""
    key = (device.type, device.index)
    st = _CAPTURE_STREAMS.get(key)
    if st is None:
        st = torch.cuda.Stream(device=device)
        _CAPTURE_STREAMS[key] = st
    return st


class VecEnvNE(NEProblem):
    """Whole-population vectorized rollouts: one env row per solution, one
    vmapped policy forward per step."""

    def __init__(
        self,
        env: Union[Callable[[int], object], object, str],
        network: Union[str, nn.Module, Callable[[], nn.Module]],
        *,
        network_args: Optional[dict] = None,
        max_num_steps: Optional[int] = None,
        observation_normalization: bool = True,
        decrease_rewards_by: float = 0.0,
        alive_bonus_schedule: Optional[tuple] = None,
        device=None,
        seed: Optional[int] = None,
        initial_bounds=(-0.00001, 0.00001),
        env_config: Optional[dict] = None,
        max_num_envs: Optional[int] = None,
        action_noise_stdev: Optional[float] = None,
        num_episodes: int = 1,
        use_hip_graph: bool = False,
        num_actors=None,
        num_gpus_per_actor=None,
        num_subbatches=None,
        subbatch_size=None,
        actor_config=None,
    ):
        env_config = dict(env_config or {})
        if isinstance(env, str):
            env_name = env
            def factory(num_envs: int):
                from . import gym_compat

                return GymVectorEnvAdapter(gym_compat.make_vec(env_name, num_envs=num_envs, **env_config), device=device or "cpu")

            self._env_factory = factory
            self._env = None
            probe = factory(1)
            self._obs_dim, self._act_dim = probe.obs_dim, probe.act_dim
        elif callable(env) and not hasattr(env, "step"):
            self._env_factory = env
            self._env = None
            probe = env(1)
            self._obs_dim, self._act_dim = probe.obs_dim, probe.act_dim
        else:
            self._env_factory = None
            self._env = env
            self._obs_dim, self._act_dim = env.obs_dim, env.act_dim

        self._max_num_steps = max_num_steps
        self._max_num_envs = None if max_num_envs is None else int(max_num_envs)
        self._action_noise_stdev = None if action_noise_stdev is None else float(action_noise_stdev)
        self._num_episodes = max(1, int(num_episodes))
        self._obs_norm_enabled = bool(observation_normalization)
        self._decrease_rewards_by = float(decrease_rewards_by)
        self._alive_bonus_schedule = alive_bonus_schedule
        super().__init__(
            "max",
            network,
            network_args=network_args,
            initial_bounds=initial_bounds,
            device=device,
            seed=seed,
            store_solution_stats=False,
            num_actors=num_actors,
            num_gpus_per_actor=num_gpus_per_actor,
            actor_config=actor_config,
            num_subbatches=num_subbatches,
            subbatch_size=subbatch_size,
        )
        self._obs_norm = RunningNorm(shape=self._obs_dim, device=self.network_device)
        self._use_hip_graph = bool(use_hip_graph)
        self._graph_state: Optional[dict] = None
        self._env_cache: Optional[dict] = None
        self._policy: Optional[Policy] = None
        self.last_eval_interaction_count = 0
        self._total_interactions = 0
        self._episode_count = 0
        self._pending_stats: Optional[RunningNorm] = None
        self.after_eval_hook.append(self._counters_status)

    # -- plumbing -------------------------------------------------------------

    def _network_constants(self) -> dict:
        return {"obs_length": self._obs_dim, "act_length": self._act_dim, "obs_space": None, "act_space": None}

    @property
    def obs_norm(self) -> RunningNorm:
        return self._obs_norm

    def observation_normalization_data(self):
        return {"mean": self._obs_norm.mean.cpu(), "stdev": self._obs_norm.stdev.cpu(), "count": self._obs_norm.count}

    def _get_env(self, num_envs: int):
        if self._env is not None and self._env.num_envs == num_envs:
            return self._env
        if self._env_factory is None:
            raise ValueError(f"The fixed env has {self._env.num_envs} rows but the batch needs {num_envs}; provide an env factory instead")
        # cache per size: max_num_envs splitting an uneven population
        # alternates between two sizes every generation
        if self._env_cache is None:
            self._env_cache = {}
        env = self._env_cache.get(num_envs)
        if env is None:
            env = self._env_factory(num_envs)
            self._env_cache[num_envs] = env
        self._env = env
        return env

    def _alive_bonus(self, t: int) -> float:
        if self._alive_bonus_schedule is None:
            return 0.0
        t0, t1, bonus = self._alive_bonus_schedule
        if t < t0:
            return 0.0
        if t >= t1:
            return float(bonus)
        return float(bonus) * (t - t0) / max(1, (t1 - t0))

    def _counters_status(self, batch) -> dict:
        self._merge_pending_stats()
        return {
            "total_interaction_count": self._total_interactions,
            "total_episode_count": self._episode_count,
        }

    def _merge_pending_stats(self):
        if self._pending_stats is None or not self._obs_norm_enabled:
            self._pending_stats = None
            return
        pending = self._pending_stats
        self._pending_stats = None
        comm = self._comm
        if comm is not None and comm.world_size > 1:
            c, s, ss = pending.stats_triple()
            packed = torch.cat([c.to(s.device, s.dtype).reshape(1), s.reshape(-1), ss.reshape(-1)])
            # During a sharded gradient generation this reduction is fused
            # into the gradient all-reduce (Problem.request_fused_reduce):
            # one collective carries gradients + (count, Σ, Σ²).
            def merge(reduced: torch.Tensor):
                self._obs_norm.update((reduced[0], reduced[1 : 1 + self._obs_dim], reduced[1 + self._obs_dim :]))

            self.request_fused_reduce(packed, merge)
        else:
            self._obs_norm.update(pending)

    # -- the hot loop ---------------------------------------

    @torch.no_grad()  # rollouts never need autograd (user nets may carry requires_grad params)
    def _evaluate_batch(self, batch: SolutionBatch):
        n = len(batch)
        if self._max_num_envs is not None and n > self._max_num_envs:
            # reference vecgymne.py: cap the batched env size and evaluate
            # the population in env-sized pieces
            for piece in batch.split(max_size=self._max_num_envs):
                self._evaluate_batch(piece)
            return
        if self._num_episodes > 1:
            total = torch.zeros(n, dtype=torch.float32)
            steps = 0
            for _ in range(self._num_episodes):
                self._rollout_once(batch)
                total = total + torch.Tensor.as_subclass(batch._evals[:, 0], torch.Tensor).to("cpu", torch.float32)
                steps += self.last_eval_interaction_count
                batch.forget_evals()
            batch.set_evals((total / self._num_episodes).to(self._eval_dtype).to(batch.device))
            self.last_eval_interaction_count = steps
            return
        self._rollout_once(batch)

    def _episode_body(self, env, policy, obs, active, fitness, steps_dev, pending, max_steps: int, *, early_exit: bool):
        """One episode over the batched env, accumulating IN PLACE into
        `fitness`/`steps_dev`/`pending`. On GPU the body never host-syncs:
        steps accumulate in a device scalar, stateless policies skip the
        (masked) recurrent reset, and the all-done early-exit probe runs
        every kCheck steps instead of every step (each bool()/int() read
        drains the whole HIP pipeline). With early_exit=False the body is
        hipGraph-capturable (no data-dependent host control flow at all)."""
        device = fitness.device
        check_every = 16 if fitness.is_cuda else 1
        if self._obs_norm_enabled and self._obs_norm.has_data:
            # obs-norm statistics are frozen for the whole episode (updates
            # merge after evaluation), so hoist mean/stdev out of the step
            # loop; inside a graph capture this recomputes once per REPLAY
            # from the in-place-updated buffers.
            norm_mean = self._obs_norm.mean.to(device)
            norm_stdev = self._obs_norm.stdev.to(device)
        else:
            norm_mean = norm_stdev = None
        norm_clip = getattr(self._obs_norm, "_clip", None)
        for t in range(max_steps):
            if self._obs_norm_enabled:
                pending.update(obs, mask=active)
                if norm_mean is None:
                    obs_in = obs
                else:
                    obs_in = (obs - norm_mean) / norm_stdev
                    if norm_clip is not None:
                        obs_in = torch.clamp(obs_in, norm_clip[0], norm_clip[1])
            else:
                obs_in = obs
            actions = policy(obs_in)
            if self._action_noise_stdev is not None:
                actions = actions + self._action_noise_stdev * torch.randn_like(actions)
            obs, reward, done = env.step(actions)
            obs = obs.to(device)
            bonus = self._alive_bonus(t)
            step_reward = reward.to(device) - self._decrease_rewards_by + bonus
            fitness.add_(step_reward * active)
            steps_dev.add_(active.sum())
            done = done.to(device)
            if policy.h is not None:
                policy.reset(done & active)
            active = active & ~done
            if early_exit and (t + 1) % check_every == 0 and not bool(active.any()):
                break

    def _rollout_once(self, batch: SolutionBatch):
        n = len(batch)
        env = self._get_env(n)
        device = self.network_device
        if self._policy is None:
            self._policy = self.make_functional_policy()
        policy = self._policy
        params = batch.access_values(keep_evals=True).to(device, torch.float32)

        from ..ops.dispatch import _seed_from_generator

        episode_seed = _seed_from_generator(self._generator, self._device) & 0x7FFFFFFF
        max_steps = self._max_num_steps or getattr(env.spec, "episode_length", None) or 1000

        if self._use_hip_graph and device.type == "cuda" and not (self._obs_norm_enabled and not self._obs_norm.has_data):
            # (The has_data guard runs the FIRST generation eagerly so the
            # capture does not bake normalize()'s no-data identity branch.)
            fitness, steps_done, pending = self._rollout_graphed(env, policy, params, episode_seed, n, max_steps)
        else:
            policy.set_parameters(params)
            obs = env.reset(seed=episode_seed).to(device)
            fitness = torch.zeros(n, dtype=torch.float32, device=device)
            active = torch.ones(n, dtype=torch.bool, device=device)
            pending = RunningNorm(shape=self._obs_dim, device=device)
            steps_dev = torch.zeros((), dtype=torch.int64, device=device)
            self._episode_body(env, policy, obs, active, fitness, steps_dev, pending, max_steps, early_exit=True)
            steps_done = int(steps_dev)
        batch.set_evals(fitness.to(self._eval_dtype).to(batch.device))
        if self._pending_stats is None:
            self._pending_stats = pending
        else:
            # multi-episode evaluations merge at the hook; fold, don't drop
            self._pending_stats.update(pending)
        self.last_eval_interaction_count = steps_done
        self._total_interactions += steps_done
        self._episode_count += n

    # -- hipGraph-captured rollouts -------------------------------------------

    def _rollout_graphed(self, env, policy, params, episode_seed: int, n: int, max_steps: int):
        """Replay the whole T-step episode as ONE hipGraph.

        The eager loop pays per-step host costs (vmap dispatch, dozens of
        small kernel launches) that dwarf the device work for small nets;
        capturing the unrolled episode removes all of it — one launch per
        generation. Requirements (opt-in via use_hip_graph=True):
        - torch-native env on the problem's CUDA device with static shapes;
        - cross-step device state declared in `env.graph_state_attrs`
          (host-side counters must be restored by env.reset, which runs
          before every replay);
        - fixed horizon: the graph always runs max_steps (done rows are
          masked out exactly like the eager path, so fitness/steps/stats
          are identical, but no early exit happens).
        Captures are cached per (n, max_steps, policy length) so workloads
        that alternate batch sizes (e.g. max_num_envs splitting an uneven
        population) do not re-capture every call."""
        if self._graph_state is None:
            self._graph_state = {}
        key = (n, max_steps, int(params.shape[-1]), bool(self._obs_norm_enabled))
        gr = self._graph_state.get(key)
        if gr is None or gr["env"] is not env:
            gr = self._capture_rollout_graph(env, policy, params, episode_seed, n, max_steps, key)
            self._graph_state[key] = gr
        else:
            gr["params_buf"].copy_(params)
            self._graph_prepare(gr, episode_seed)
        gr["graph"].replay()
        pending = RunningNorm(shape=self._obs_dim, device=params.device)
        if self._obs_norm_enabled:
            gr["pending"]._has_data = True
            pending.update(gr["pending"])  # snapshot: the static buffers are re-zeroed by the next replay
        return gr["fitness"].clone(), int(gr["steps"]), pending

    def _graph_prepare(self, gr: dict, episode_seed: int):
        """Reset the env eagerly and re-home its cross-step state into the
        capture-time static buffers (env.step REBINDS state attributes, so
        after any episode they point at graph-internal tensors)."""
        env = gr["env"]
        obs0 = env.reset(seed=episode_seed).to(gr["fitness"].device)
        for attr, buf in gr["state_bufs"].items():
            buf.copy_(getattr(env, attr))
            setattr(env, attr, buf)
        if gr["obs_attr"] is None:
            gr["obs_buf"].copy_(obs0)

    def _capture_rollout_graph(self, env, policy, params, episode_seed: int, n: int, max_steps: int, key):
        device = self.network_device
        obs0 = env.reset(seed=episode_seed)
        if not (torch.is_tensor(obs0) and obs0.is_cuda):
            raise ValueError(
                "use_hip_graph=True needs a torch-native env living on the GPU device "
                "(numpy/gymnasium adapter envs cannot be hipGraph-captured)"
            )
        attrs = getattr(env, "graph_state_attrs", None)
        if attrs is None:
            attrs = ("_obs",) if torch.is_tensor(getattr(env, "_obs", None)) else ()
        originals = {a: getattr(env, a) for a in attrs}
        state_bufs = {a: v.detach().clone() for a, v in originals.items()}
        obs_attr = next((a for a in attrs if originals[a] is obs0), None)
        obs_buf = state_bufs[obs_attr] if obs_attr is not None else obs0.detach().clone()

        params_buf = params.detach().clone()
        policy.set_parameters(params_buf)
        pending = RunningNorm(shape=self._obs_dim, device=device)
        fitness = torch.zeros(n, dtype=torch.float32, device=device)
        active = torch.ones(n, dtype=torch.bool, device=device)
        steps = torch.zeros((), dtype=torch.int64, device=device)
        gr = {
            "key": key,
            "env": env,
            "state_bufs": state_bufs,
            "obs_attr": obs_attr,
            "obs_buf": obs_buf,
            "params_buf": params_buf,
            "pending": pending,
            "fitness": fitness,
            "steps": steps,
            "graph": None,
        }

        def body():
            pending._count.zero_()
            pending._sum.zero_()
            pending._sum_sq.zero_()
            fitness.zero_()
            steps.zero_()
            active.fill_(True)
            policy.reset()  # recurrent state (if any) re-initializes inside the capture
            self._episode_body(env, policy, obs_buf, active, fitness, steps, pending, max_steps, early_exit=False)

        self._graph_prepare(gr, episode_seed)
        side = _capture_stream(torch.device(device))
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):  # warm up allocator/BLAS workspaces off the capture
                body()
                self._graph_prepare(gr, episode_seed)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        # The cyclic GC must not run DURING capture: collecting stale CUDA
        # garbage (dead graphs/tensors from earlier work) mid-capture frees
        # device memory on non-captured streams, which aborts the HIP
        # runtime. Drain it now, then hold it off until capture ends.
        import gc

        gc.collect()
        gc.disable()
        try:
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph, stream=side):
                body()
        finally:
            gc.enable()
        gr["graph"] = graph
        # capture records without executing: prepare() state is still intact,
        # so the caller's replay() right after this produces this
        # generation's real result
        return gr

    # -- policy export --------------------------------------------------------

    # -- reference-parity accessors (vecgymne.py:590-716) ---------------------

    @property
    def episode_count(self) -> int:
        return self._episode_count

    def set_episode_count(self, n: int):
        self._episode_count = int(n)

    @property
    def interaction_count(self) -> int:
        return self._total_interactions

    def set_interaction_count(self, n: int):
        self._total_interactions = int(n)

    @property
    def observation_normalization(self) -> bool:
        return self._obs_norm_enabled

    @property
    def max_num_envs(self) -> Optional[int]:
        return self._max_num_envs

    def get_env(self):
        """The underlying batched env (created lazily on first evaluation)."""
        return getattr(self, "_env", None)

    def get_observation_stats(self) -> RunningNorm:
        return self._obs_norm

    def set_observation_stats(self, rn: RunningNorm):
        self._obs_norm = rn.to(self.network_device)

    def update_observation_stats(self, rn) -> RunningNorm:
        """Merge another RunningNorm (or (count, sum, sumsq) triple) into
        this problem's stats — the reference's actor sync protocol entry
        point (vecgymne.py:658-716)."""
        self._obs_norm.update(rn if not isinstance(rn, RunningNorm) else rn.stats_triple())
        return self._obs_norm

    def pop_observation_stats(self) -> Optional[RunningNorm]:
        """Take (and clear) the stats collected since the last pop — what a
        worker ships to the main process in the reference protocol."""
        pending = self._pending_stats
        self._pending_stats = None
        return pending

    def save_solution(self, solution, path: str):
        """Persist a solution as a ready-to-run policy: safetensors weights
        + JSON sidecar (see deploy.save_policy). `solution` may be a
        Solution or a flat tensor."""
        from .deploy import save_policy

        x = solution.values if hasattr(solution, "values") else solution
        policy = self.to_policy(torch.Tensor.as_subclass(torch.as_tensor(x), torch.Tensor))
        return save_policy(policy, path)

    def to_policy(self, x: torch.Tensor) -> nn.Module:
        if self._policy is None:
            self._policy = self.make_functional_policy()
        module = self._policy.to_torch_module(torch.as_tensor(x, dtype=torch.float32).cpu())
        if self._obs_norm_enabled and self._obs_norm.has_data:
            norm_cpu = self._obs_norm.to("cpu")
            module = nn.Sequential(norm_cpu.to_layer(), module)
        module.requires_grad_(False)  # inference artifact
        return module


VecGymNE = VecEnvNE  # reference-compatible alias (vecgymne.py:95)


# -- env-infra helpers (reference net/vecrl.py) ------------------------------
# brax/jax are not installed in the MI355X image (no jax wheel for ROCm
# here); the brax entry points stay importable and raise with a clear
# message, the numpy<->torch plumbing is fully functional.


def array_type(x) -> str:
    """'torch' | 'numpy' | 'scalar' | 'unknown' (reference vecrl.py)."""
    import numpy as _np

    if isinstance(x, torch.Tensor):
        return "torch"
    if isinstance(x, _np.ndarray):
        return "numpy"
    if isinstance(x, (int, float, bool, _np.number)):
        return "scalar"
    return "unknown"


def convert_to_torch(x, *, device=None) -> torch.Tensor:
    """numpy/scalar/tensor -> torch tensor (zero-copy where possible)."""
    import numpy as _np

    if isinstance(x, torch.Tensor):
        t = x
    elif isinstance(x, _np.ndarray):
        t = torch.from_numpy(_np.ascontiguousarray(x))
    else:
        t = torch.as_tensor(x)
    return t.to(device) if device is not None else t


def convert_to_torch_bool(x, *, device=None) -> torch.Tensor:
    return convert_to_torch(x, device=device).to(torch.bool)


def convert_from_torch(x: torch.Tensor):
    """torch tensor -> numpy array (cpu copy only if needed)."""
    return x.detach().cpu().numpy()


def jax_to_torch(x):
    """Zero-copy jax→torch via the DLPack protocol (reference
    net/vecrl.py:53-81). Requires jax (no ROCm jax wheel ships in this
    image; with a user-provided jax the bridge works as-is)."""
    return torch.from_dlpack(x)


def torch_to_jax(x: torch.Tensor):
    """Zero-copy torch→jax via DLPack (reference net/vecrl.py:53-81)."""
    try:
        import jax
    except ImportError as e:
        raise ImportError(
            "torch_to_jax requires jax, which has no ROCm wheel in this image; "
            "provide your own jax build to use the brax bridge"
        ) from e
    return jax.dlpack.from_dlpack(torch.utils.dlpack.to_dlpack(x.detach()))


def is_brax_env(env) -> bool:
    """True if env comes from brax (not installed in this image — always
    False unless the user provides brax themselves)."""
    mod = type(env).__module__ or ""
    return mod.startswith("brax")


def make_brax_env(env_name: str, **kwargs):
    raise ImportError(
        "brax requires jax, which has no ROCm wheel in this image."
        " Use SyntheticTorchEnv / a torch-native batched env (the VecEnvNE"
        " contract: reset()->obs, step(act)->(obs, reward, done, info)),"
        " or GymVectorEnvAdapter over gymnasium."
    )


def make_gym_env(env_name: str, **kwargs):
    """Instantiate a single env by name (reference vecrl.py:668): real
    gymnasium when installed, the vendored classic-control registry
    otherwise."""
    from . import gym_compat

    return gym_compat.make(env_name, **kwargs)


def make_vector_env(env_name: str, *, num_envs: int, **kwargs):
    """Batched env by name, wrapped into the torch contract used by
    VecEnvNE (reference vecrl.py:764): real gymnasium when installed,
    the vendored autoresetting SyncVectorEnv otherwise."""
    from . import gym_compat

    vec = gym_compat.make_vec(env_name, num_envs=num_envs, **kwargs)
    return GymVectorEnvAdapter(vec)


class TorchWrapper:
    """Wraps a classic (numpy-API) env so that reset/step speak torch
    tensors (reference vecrl.py:362); also normalizes old/new gym step
    APIs via the gymne helpers."""

    def __init__(self, env, *, device=None):
        from .gymne import reset_env, take_step_in_env

        self.env = env
        self._device = device
        self._reset_env = reset_env
        self._step_env = take_step_in_env

    def __getattr__(self, name):
        return getattr(self.env, name)

    def reset(self):
        return convert_to_torch(self._reset_env(self.env), device=self._device)

    def step(self, action: torch.Tensor):
        obs, reward, done, info = self._step_env(self.env, convert_from_torch(action))
        return (
            convert_to_torch(obs, device=self._device),
            convert_to_torch(reward, device=self._device),
            convert_to_torch_bool(done, device=self._device),
            info,
        )
