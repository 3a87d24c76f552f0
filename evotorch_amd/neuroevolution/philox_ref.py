"""Exact CPU reference of the device philox4x32-10 + Box-Muller stream
(evotorch_amd/ops/hip/philox.h). Used to generate kernel-identical initial
observations on CPU and to test the K1 sampling kernel bit-for-bit at the
integer level."""

import numpy as np
import torch

__all__ = ["philox4x32_10", "philox_normal_rows", "philox_normals", "philox_normals_2d"]

_M0 = np.uint32(0xD2511F53)
_M1 = np.uint32(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)


def philox4x32_10(c0, c1, c2, c3, k0, k1):
    """Vectorized philox4x32-10 over numpy uint32 arrays."""
    c0 = np.asarray(c0, dtype=np.uint32).copy()
    c1 = np.asarray(c1, dtype=np.uint32).copy()
    c2 = np.asarray(c2, dtype=np.uint32).copy()
    c3 = np.asarray(c3, dtype=np.uint32).copy()
    k0 = np.uint32(k0)
    k1 = np.uint32(k1)
    with np.errstate(over="ignore"):
        for _ in range(10):
            p0 = c0.astype(np.uint64) * np.uint64(_M0)
            p1 = c2.astype(np.uint64) * np.uint64(_M1)
            h0 = (p0 >> np.uint64(32)).astype(np.uint32)
            l0 = p0.astype(np.uint32)
            h1 = (p1 >> np.uint64(32)).astype(np.uint32)
            l1 = p1.astype(np.uint32)
            n0 = h1 ^ c1 ^ k0
            n1 = l1
            n2 = h0 ^ c3 ^ k1
            n3 = l0
            c0, c1, c2, c3 = n0, n1, n2, n3
            k0 = np.uint32((int(k0) + int(_W0)) & 0xFFFFFFFF)
            k1 = np.uint32((int(k1) + int(_W1)) & 0xFFFFFFFF)
    return c0, c1, c2, c3


def _u32_to_uniform(v: np.ndarray) -> np.ndarray:
    return ((v >> np.uint32(8)).astype(np.float32) + np.float32(1.0)) * np.float32(1.0 / 16777216.0)


def _box_muller4(r0, r1, r2, r3):
    u0, u1, u2, u3 = (_u32_to_uniform(x) for x in (r0, r1, r2, r3))
    rad0 = np.sqrt(np.float32(-2.0) * np.log(u0, dtype=np.float32))
    rad1 = np.sqrt(np.float32(-2.0) * np.log(u2, dtype=np.float32))
    two_pi = np.float32(6.2831853071795864)
    z0 = rad0 * np.cos(two_pi * u1, dtype=np.float32)
    z1 = rad0 * np.sin(two_pi * u1, dtype=np.float32)
    z2 = rad1 * np.cos(two_pi * u3, dtype=np.float32)
    z3 = rad1 * np.sin(two_pi * u3, dtype=np.float32)
    return np.stack([z0, z1, z2, z3], axis=-1)  # (..., 4)


def philox_normals(seed: int, stream_id: int, n_elements: int, *, idx4_offset: int = 0) -> torch.Tensor:
    """The first n_elements of stream (seed, stream_id) starting at counter
    idx4_offset — matches philox_normal4 element indexing (element e of the
    stream is produced by counter e//4, so idx4_offset = elem_offset//4)."""
    n4 = (n_elements + 3) // 4
    idx4 = np.arange(n4, dtype=np.uint64) + np.uint64(idx4_offset)
    c0 = idx4.astype(np.uint32)
    c1 = (idx4 >> np.uint64(32)).astype(np.uint32)
    c2 = np.full(n4, np.uint32(stream_id & 0xFFFFFFFF), dtype=np.uint32)
    c3 = np.zeros(n4, dtype=np.uint32)
    r0, r1, r2, r3 = philox4x32_10(c0, c1, c2, c3, np.uint32(seed & 0xFFFFFFFF), np.uint32((seed >> 32) & 0xFFFFFFFF))
    z = _box_muller4(r0, r1, r2, r3).reshape(-1)[:n_elements]
    return torch.from_numpy(np.ascontiguousarray(z))


def philox_normals_2d(seed: int, row_offset: int, rows: int, length: int) -> torch.Tensor:
    """(rows, length) normals with STREAM-PER-ROW addressing: row r draws
    from stream `row_offset + r`, counter c covering columns [4c, 4c+4).
    Matches the K1 counter-addressed sampling kernel
    (ops/hip/es_kernels.hip::sample_gaussian_kernel) for any row partition
    of a virtual population: generating rows [a, b) here equals slicing
    rows [a, b) of the full population."""
    len4 = (length + 3) // 4
    col4 = np.tile(np.arange(len4, dtype=np.uint64), rows)
    stream = np.repeat(np.arange(row_offset, row_offset + rows, dtype=np.uint64), len4)
    c0 = col4.astype(np.uint32)
    c1 = (col4 >> np.uint64(32)).astype(np.uint32)
    c2 = stream.astype(np.uint32)
    c3 = np.zeros_like(c2)
    r0, r1, r2, r3 = philox4x32_10(c0, c1, c2, c3, np.uint32(seed & 0xFFFFFFFF), np.uint32((seed >> 32) & 0xFFFFFFFF))
    z = _box_muller4(r0, r1, r2, r3).reshape(rows, len4 * 4)[:, :length]
    return torch.from_numpy(np.ascontiguousarray(z))


def philox_normal_rows(seed: int, member_offset: int, n_members: int, row_len: int) -> torch.Tensor:
    """(n_members, row_len) normals where row m uses stream_id
    member_offset + m — the kernel's per-member initial-obs stream."""
    rows = [philox_normals(seed, member_offset + m, row_len) for m in range(n_members)]
    return torch.stack(rows, dim=0)
