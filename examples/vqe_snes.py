"""Variational quantum eigensolver with SNES (mirrors the reference's
Variational_Quantum_Eigensolvers_with_SNES.ipynb): the solution vector
parameterizes a hardware-efficient ansatz circuit (single-qubit RY
rotations + a CZ entangling ladder, repeated in layers), simulated as a
plain torch statevector; fitness is the energy <psi|H|psi> of a
transverse-field Ising Hamiltonian. SNES drives the energy to the exact
ground state obtained by dense diagonalization.

Run: python examples/vqe_snes.py [--qubits 4] [--layers 3]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from evotorch_amd import Problem
from evotorch_amd.algorithms import SNES
from evotorch_amd.decorators import vectorized

I2 = torch.eye(2, dtype=torch.complex64)
X = torch.tensor([[0, 1], [1, 0]], dtype=torch.complex64)
Z = torch.tensor([[1, 0], [0, -1]], dtype=torch.complex64)


def kron_all(ops):
    out = ops[0]
    for op in ops[1:]:
        out = torch.kron(out, op)
    return out


def tfim_hamiltonian(n, h=1.0):
    """Transverse-field Ising: -sum Z_i Z_{i+1} - h * sum X_i."""
    dim = 2**n
    H = torch.zeros(dim, dim, dtype=torch.complex64)
    for i in range(n - 1):
        ops = [I2] * n
        ops[i] = Z
        ops[i + 1] = Z
        H -= kron_all(ops)
    for i in range(n):
        ops = [I2] * n
        ops[i] = X
        H -= h * kron_all(ops)
    return H


def apply_ry_layer(state, angles, n):
    """Batched RY on every qubit: state (B, 2^n), angles (B, n)."""
    for q in range(n):
        half = angles[:, q : q + 1] / 2
        c, s = torch.cos(half).to(state.dtype), torch.sin(half).to(state.dtype)
        view = state.reshape(state.shape[0], 2**q, 2, 2 ** (n - q - 1))
        a, b = view[:, :, 0, :], view[:, :, 1, :]
        view = torch.stack([c.unsqueeze(-1) * a - s.unsqueeze(-1) * b,
                            s.unsqueeze(-1) * a + c.unsqueeze(-1) * b], dim=2)
        state = view.reshape(state.shape[0], -1)
    return state


def apply_cz_ladder(state, n):
    for q in range(n - 1):
        # adjacent qubits q, q+1 are the two middle axes of this view
        view = state.reshape(state.shape[0], 2**q, 2, 2, 2 ** (n - q - 2)).clone()
        view[:, :, 1, 1, :] = -view[:, :, 1, 1, :]
        state = view.reshape(state.shape[0], -1)
    return state


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--qubits", type=int, default=4)
    ap.add_argument("--layers", type=int, default=3)
    ap.add_argument("--generations", type=int, default=400)
    args = ap.parse_args()
    n, layers = args.qubits, args.layers

    H = tfim_hamiltonian(n)
    exact = float(torch.linalg.eigvalsh(H).min())

    @vectorized
    def energy(params: torch.Tensor) -> torch.Tensor:
        batch = params.shape[0]
        state = torch.zeros(batch, 2**n, dtype=torch.complex64)
        state[:, 0] = 1.0
        theta = params.reshape(batch, layers, n)
        for layer in range(layers):
            state = apply_ry_layer(state, theta[:, layer], n)
            if layer < layers - 1:
                state = apply_cz_ladder(state, n)
        return torch.einsum("bi,ij,bj->b", state.conj(), H.to(state.dtype), state).real

    problem = Problem("min", energy, solution_length=layers * n, initial_bounds=(-0.1, 0.1), seed=1,
                      store_solution_stats=True)
    searcher = SNES(problem, popsize=40, stdev_init=0.5)
    searcher.run(args.generations)
    found = float(searcher.status["best_eval"])
    print(f"TFIM n={n}: VQE energy {found:.5f}, exact ground state {exact:.5f}, "
          f"error {abs(found-exact):.2e}")
    assert found - exact < 0.1, "VQE did not approach the ground state"
    print("VQE OK")


if __name__ == "__main__":
    main()
