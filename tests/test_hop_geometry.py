"""The HIP path's (diag, win) hop geometry must reproduce the oracle's
position-based mask for every (layout, rank pair, lookback) combination.

This is the CPU-side guarantee that the GPU ring dispatch (which cannot run
multi-process here) masks exactly like the verified oracle."""

import pytest
import torch

from ring_attention_amd.ops.ring_flash import SKIP, bucket_mode
from ring_attention_amd.ops.ring_flash_hip import _hop_geometry


def oracle_mask(rq, rk, n, R, striped, causal, lookback):
    """Full (n, n) attend matrix from the oracle's bucket logic (bucket=1)."""
    allowed = torch.zeros(n, n, dtype=torch.bool)
    for bi in range(n):
        for bj in range(n):
            mode = bucket_mode(bi, bj, rq, rk, bucket_size=1, shard_len=n,
                               ring_size=R, causal=causal, striped=striped,
                               lookback=lookback, device="cpu")
            if mode is SKIP:
                allowed[bi, bj] = False
            elif isinstance(mode, torch.Tensor):
                allowed[bi, bj] = not bool(mode[0, 0])
            else:
                allowed[bi, bj] = True
    return allowed


def hip_mask(rq, rk, n, R, striped, causal, lookback):
    skip, diag, win = _hop_geometry(rq, rk, n, R, striped, causal, lookback)
    if skip:
        return torch.zeros(n, n, dtype=torch.bool)
    i = torch.arange(n)[:, None]
    j = torch.arange(n)[None, :]
    qpos = i + diag                      # kernel semantics: qpos(i) = i*qs + diag
    allowed = torch.ones(n, n, dtype=torch.bool)
    if causal:
        allowed &= j <= qpos
    if lookback is not None:
        allowed &= (qpos - j) <= win
    return allowed


@pytest.mark.parametrize("R", [2, 4])
@pytest.mark.parametrize("striped", [False, True])
@pytest.mark.parametrize("lookback", [None, 5, 13])
def test_hop_geometry_matches_oracle(R, striped, lookback):
    n = 8
    for rq in range(R):
        for rk in range(R):
            a = oracle_mask(rq, rk, n, R, striped, True, lookback)
            b = hip_mask(rq, rk, n, R, striped, True, lookback)
            assert torch.equal(a, b), (
                f"mismatch rq={rq} rk={rk} R={R} striped={striped} lb={lookback}\n{a}\n{b}")


def test_hop_geometry_noncausal():
    for R in (2, 4):
        for rq in range(R):
            for rk in range(R):
                a = oracle_mask(rq, rk, 8, R, False, False, None)
                b = hip_mask(rq, rk, 8, R, False, False, None)
                assert torch.equal(a, b)
