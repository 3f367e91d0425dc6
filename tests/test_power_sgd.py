"""PowerSGD gradient averaging (reference tests/test_optimizer.py PowerSGD cases)."""

import threading

import pytest
import torch
import torch.nn as nn

from hivemind_amd import DHT
from hivemind_amd.optim.power_sgd_averager import PowerSGDGradientAverager


def make_dht_swarm(n):
    root = DHT(start=True)
    return [root] + [DHT(initial_peers=[root.endpoint], start=True) for _ in range(n - 1)]


def test_power_sgd_full_rank_matches_exact_average():
    """With rank >= matrix dim, PowerSGD is exact up to orthogonalization noise."""
    torch.manual_seed(0)
    dhts = make_dht_swarm(2)
    models = [nn.Linear(8, 8, bias=True) for _ in range(2)]
    averagers = [
        PowerSGDGradientAverager(
            model.parameters(), averager_rank=8, dht=dht, prefix="psgd", target_group_size=2,
            min_group_size=2, min_matchmaking_time=1.0, request_timeout=0.5, start=True,
        )
        for model, dht in zip(models, dhts)
    ]
    grads = [torch.randn(8, 8) for _ in range(2)]
    for model, g, avg in zip(models, grads, averagers):
        model.weight.grad = g.clone()
        model.bias.grad = torch.ones(8) * (1 if avg is averagers[0] else 3)
        avg.accumulate_grads_(1)

    controls = [avg.step(wait=False, timeout=60) for avg in averagers]
    for c in controls:
        assert c.result(90) is not None

    expected_weight = (grads[0] + grads[1]) / 2
    expected_bias = torch.full((8,), 2.0)
    for avg in averagers:
        with avg.get_tensors() as tensors:
            weight_avg, bias_avg = tensors
            # rank-8 approximation of an 8x8 matrix is exact (modulo fp error)
            assert torch.allclose(weight_avg, expected_weight, atol=1e-3), (weight_avg - expected_weight).abs().max()
            # 1-D tensors ride uncompressed: exact
            assert torch.allclose(bias_avg, expected_bias, atol=1e-5)
    for avg in averagers:
        avg.shutdown()
    for d in dhts:
        d.shutdown()


def test_power_sgd_low_rank_error_feedback():
    """Low-rank rounds accumulate the residual; repeated rounds recover the mean."""
    torch.manual_seed(1)
    dhts = make_dht_swarm(2)
    models = [nn.Linear(16, 16, bias=False) for _ in range(2)]
    averagers = [
        PowerSGDGradientAverager(
            model.parameters(), averager_rank=2, dht=dht, prefix="psgd2", target_group_size=2,
            min_group_size=2, min_matchmaking_time=1.0, request_timeout=0.5, start=True,
        )
        for model, dht in zip(models, dhts)
    ]
    # identical constant gradient each round: check the error-feedback identity
    # sum(transmitted) + final_residual == num_rounds * grad, and both peers agree
    true_grad = torch.randn(16, 16)
    num_rounds = 4
    transmitted_sum = torch.zeros(16, 16)
    for round_idx in range(num_rounds):
        for model, avg in zip(models, averagers):
            model.weight.grad = true_grad.clone()
            avg.accumulate_grads_(1)
        controls = [avg.step(wait=False, timeout=60) for avg in averagers]
        for c in controls:
            assert c.result(90) is not None
        with averagers[0].get_tensors() as t0, averagers[1].get_tensors() as t1:
            assert torch.allclose(t0[0], t1[0], atol=1e-4), "peers must reconstruct the same average"
            transmitted_sum += t0[0]
    final_residual = averagers[0]._ms[0].clone()
    reconstruction = transmitted_sum + final_residual
    expected = num_rounds * true_grad
    assert torch.allclose(reconstruction, expected, atol=1e-3), (reconstruction - expected).abs().max()
    # error feedback makes the cumulative transmission converge toward the mean:
    # the final residual is bounded (it does not blow up round over round)
    assert final_residual.norm() < expected.norm()
    for avg in averagers:
        avg.shutdown()
    for d in dhts:
        d.shutdown()
