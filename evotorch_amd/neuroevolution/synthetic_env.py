"""Synthetic vectorized control environment (Humanoid-shaped).

There is no simulator available offline (no mujoco/brax/gymnasium on the
GPU box), so the flagship RL benchmark (BASELINE.json: "PGPE Humanoid-v4
linear policy") runs against this synthetic batched environment with
Humanoid-v4's observation/action geometry (obs 376, act 17) and a fixed
low-rank neural dynamics:

    h   = V·o                      V: R×O
    o'  = tanh(Uᵀ·h + D2ᵀ·a + c)   U_T: R×O (U transposed), D2_T: A×O
    r   = wr·o' + alive_bonus − act_cost·‖a‖²/A
    o₀  = 0.1·N(0, I)  (philox-deterministic per global member index)

(Green-field MI355X substitute for the reference's brax/gym env plumbing,
/root/reference/src/evotorch/neuroevolution/net/vecrl.py:616-1664 — the
rollout CONTRACT matches VecGymNE's batched env, the dynamics are
synthetic.)

The fused gfx950 kernel `evotorch_amd._C.rollout_linear`
(evotorch_amd/ops/hip/rollout.hip) runs the entire T-step episode of every
member with policy weights and env matrices resident in LDS (bf16 storage,
fp32 accumulate); `rollout_eager` below is the exact eager reference the
kernel is tested against (it quantizes the same operands to bf16).
"""

from typing import Optional, Tuple

import torch

__all__ = ["SyntheticEnvSpec", "rollout_eager"]

HUMANOID_OBS = 376
HUMANOID_ACT = 17


class SyntheticEnvSpec:
    """Holds the fixed environment matrices and packs them into the blob
    layout the HIP kernel expects (see rollout.hip: V, U_T, D2_T, c, wr,
    mean, std)."""

    def __init__(
        self,
        *,
        obs_dim: int = HUMANOID_OBS,
        act_dim: int = HUMANOID_ACT,
        rank: int = 16,
        episode_length: int = 200,
        alive_bonus: float = 1.0,
        act_cost: float = 0.05,
        env_seed: int = 1234,
        policy_hidden: int = 0,   # 0 = linear policy; H = one tanh hidden layer
        device="cpu",
    ):
        self.obs_dim = int(obs_dim)
        self.act_dim = int(act_dim)
        self.rank = int(rank)
        self.policy_hidden = int(policy_hidden)
        self.episode_length = int(episode_length)
        self.alive_bonus = float(alive_bonus)
        self.act_cost = float(act_cost)
        self.env_seed = int(env_seed)
        g = torch.Generator().manual_seed(self.env_seed)
        O, A, R = self.obs_dim, self.act_dim, self.rank
        # mildly contractive dynamics so rollouts neither explode nor die
        self.V = torch.randn(R, O, generator=g) * (1.2 / O**0.5)
        self.U_T = torch.randn(R, O, generator=g) * (1.2 / R**0.5)
        self.D2_T = torch.randn(A, O, generator=g) * (0.5 / A**0.5)
        self.c = torch.randn(O, generator=g) * 0.05
        self.wr = torch.randn(O, generator=g) * (1.0 / O**0.5)
        self.device = torch.device(device)

    @property
    def solution_length(self) -> int:
        if self.policy_hidden > 0:
            H = self.policy_hidden
            return H * self.obs_dim + H + self.act_dim * H + self.act_dim
        return self.act_dim * self.obs_dim + self.act_dim

    def env_blob(self, mean: torch.Tensor, std: torch.Tensor, device=None) -> torch.Tensor:
        """Pack (V, U_T, D2_T, c, wr, mean, std) contiguously as fp32. The
        static matrices are cached per device; only mean/std change per
        generation."""
        device = torch.device(device or self.device)
        cache = getattr(self, "_matrix_cache", None)
        if cache is None or cache[0] != device:
            mats = torch.cat(
                [p.reshape(-1).to(torch.float32) for p in (self.V, self.U_T, self.D2_T, self.c, self.wr)]
            ).to(device)
            cache = (device, mats)
            self._matrix_cache = cache
        tail = torch.cat([mean.reshape(-1).to(torch.float32), std.reshape(-1).to(torch.float32)]).to(device)
        return torch.cat([cache[1], tail]).contiguous()

    def initial_obs(self, n_members: int, member_offset: int, init_seed: int, device=None) -> torch.Tensor:
        """0.1 * N(0,1), matching the kernel's philox stream exactly.

        Routed through ops.sample_gaussian (mu=0, sigma=0.1): the HIP K1
        kernel on GPU, the vectorized numpy philox reference on CPU —
        bitwise-identical values either way (fmaf(0.1, z, 0) == 0.1*z),
        and no per-row Python loop (env.reset was costing ~150 ms per
        generation at 2048 envs through the scalar reference)."""
        from ..ops import sample_gaussian

        dev = torch.device(device or self.device)
        O = self.obs_dim
        out = torch.empty(n_members, O, dtype=torch.float32, device=dev)
        mu = torch.zeros(O, dtype=torch.float32, device=dev)
        sigma = torch.full((O,), 0.1, dtype=torch.float32, device=dev)
        sample_gaussian(out, mu, sigma, seed=int(init_seed), row_offset=int(member_offset))
        return out


def rollout_eager(
    spec: SyntheticEnvSpec,
    params: torch.Tensor,
    mean: torch.Tensor,
    std: torch.Tensor,
    *,
    steps: Optional[int] = None,
    init_seed: int = 0,
    member_offset: int = 0,
    bf16_operands: bool = True,
) -> Tuple[torch.Tensor, Tuple[float, torch.Tensor, torch.Tensor]]:
    """Eager (device-agnostic) reference of the fused rollout kernel.

    Returns (fitness[N], (count, obs_sum[O], obs_sumsq[O])).
    With bf16_operands=True the matrices are quantized to bf16 before the
    fp32-accumulated products, mirroring the kernel's LDS storage."""
    steps = spec.episode_length if steps is None else int(steps)
    N = params.shape[0]
    O, A, R = spec.obs_dim, spec.act_dim, spec.rank
    device = params.device
    dt = torch.float32

    def q(x):
        x = x.to(device, torch.float32)
        return x.to(torch.bfloat16).to(torch.float32) if bf16_operands else x

    # bf16 OPERANDS / fp32 accumulation, exactly as the kernel stores them
    H = spec.policy_hidden
    if H > 0:
        off = 0
        W1 = q(params[:, off : off + H * O].reshape(N, H, O)); off += H * O
        b1 = params[:, off : off + H].to(device, dt); off += H
        W2 = q(params[:, off : off + A * H].reshape(N, A, H)); off += A * H
        b = params[:, off :].to(device, dt)
    else:
        W = q(params[:, : A * O].reshape(N, A, O))
        b = params[:, A * O :].to(device, dt)
    V = q(spec.V)
    U_T = q(spec.U_T)
    D2_T = q(spec.D2_T)
    c = spec.c.to(device, dt)
    wr = spec.wr.to(device, dt)
    mean_f = mean.to(device, dt)
    inv_std = 1.0 / std.to(device, dt)

    obs = q(spec.initial_obs(N, member_offset, init_seed, device=device))
    fitness = torch.zeros(N, dtype=dt, device=device)
    obs_sum = torch.zeros(O, dtype=dt, device=device)
    obs_sumsq = torch.zeros(O, dtype=dt, device=device)
    for _ in range(steps):
        obs_n = q((obs - mean_f) * inv_std)
        if H > 0:
            hid = q(torch.tanh(torch.einsum("nho,no->nh", W1, obs_n) + b1))
            act = torch.clamp(torch.einsum("nah,nh->na", W2, hid) + b, -1.0, 1.0)
        else:
            act = torch.clamp(torch.einsum("nao,no->na", W, obs_n) + b, -1.0, 1.0)
        act_cost_term = spec.act_cost * (act**2).sum(-1) / A
        act_b = q(act)
        h = q(obs @ V.T)  # (N, R)
        o_new = torch.tanh(h @ U_T + act_b @ D2_T + c)
        fitness = fitness + o_new @ wr + spec.alive_bonus - act_cost_term
        obs_sum += o_new.sum(0)
        obs_sumsq += (o_new**2).sum(0)
        obs = q(o_new)
    return fitness, (float(N * steps), obs_sum, obs_sumsq)
