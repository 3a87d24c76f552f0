"""Model-predictive control with the functional CEM API (mirrors the
reference's Model_Predictive_Control_with_CEM notebooks): at every control
step, CEM optimizes an H-step action sequence against a known dynamics
model, the first action is applied, and the optimization restarts from the
shifted plan (warm start).

Plant: torque-limited pendulum swing-up (classic underactuated benchmark).

Run: python examples/mpc_cem_pendulum.py [--steps 120] [--horizon 20]
"""

import argparse
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd.algorithms.functional import cem, cem_ask, cem_tell

G, M, L_POLE, DT, MAX_TORQUE = 10.0, 1.0, 1.0, 0.05, 2.0


def pendulum_step(theta, theta_dot, u):
    """One step of pendulum dynamics (batched over leading dims)."""
    u = u.clamp(-MAX_TORQUE, MAX_TORQUE)
    theta_ddot = 3 * G / (2 * L_POLE) * torch.sin(theta) + 3.0 / (M * L_POLE**2) * u
    theta_dot = (theta_dot + theta_ddot * DT).clamp(-8.0, 8.0)
    theta = theta + theta_dot * DT
    return theta, theta_dot


def plan_cost(plans, theta0, theta_dot0):
    """Cost of (N, H) action sequences from the current state."""
    n, horizon = plans.shape
    theta = torch.full((n,), float(theta0))
    theta_dot = torch.full((n,), float(theta_dot0))
    cost = torch.zeros(n)
    for t in range(horizon):
        theta, theta_dot = pendulum_step(theta, theta_dot, plans[:, t])
        angle = torch.atan2(torch.sin(theta), torch.cos(theta))
        cost = cost + angle**2 + 0.1 * theta_dot**2 + 0.001 * plans[:, t] ** 2
    return cost


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=120)
    ap.add_argument("--horizon", type=int, default=20)
    ap.add_argument("--iters", type=int, default=6)
    ap.add_argument("--popsize", type=int, default=200)
    args = ap.parse_args()

    torch.manual_seed(0)
    theta, theta_dot = math.pi, 0.0  # hanging down
    plan = torch.zeros(args.horizon)

    for step in range(args.steps):
        state = cem(center_init=plan, stdev_init=1.0, parenthood_ratio=0.1,
                    objective_sense="min", stdev_max_change=0.5)
        for _ in range(args.iters):
            pop = cem_ask(state, popsize=args.popsize)
            costs = plan_cost(pop, theta, theta_dot)
            state = cem_tell(state, pop, costs)
        plan = state.center
        th = torch.tensor(theta)
        theta_t, theta_dot_t = pendulum_step(th, torch.tensor(theta_dot), plan[0:1].squeeze())
        theta, theta_dot = float(theta_t), float(theta_dot_t)
        plan = torch.cat([plan[1:], plan[-1:]])  # shift (warm start)
        if (step + 1) % 30 == 0:
            angle = math.atan2(math.sin(theta), math.cos(theta))
            print(f"step {step+1}: angle={angle:+.3f} rad, speed={theta_dot:+.3f}")

    angle = math.atan2(math.sin(theta), math.cos(theta))
    print(f"final: angle={angle:+.4f} rad (0 = upright), speed={theta_dot:+.4f}")
    assert abs(angle) < 0.2, "pendulum did not swing up"
    print("swing-up OK")


if __name__ == "__main__":
    main()
