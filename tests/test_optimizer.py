"""Optimizer integration tests (reference tests/test_optimizer.py shape)."""

import threading
import time

import numpy as np
import pytest
import torch
import torch.nn as nn
import torch.nn.functional as F

from hivemind_amd import DHT, Optimizer
from hivemind_amd.optim.grad_averager import GradientAverager
from hivemind_amd.optim.state_averager import TrainingStateAverager


def make_dht_swarm(n):
    root = DHT(start=True)
    return [root] + [DHT(initial_peers=[root.endpoint], start=True) for _ in range(n - 1)]


def test_grad_averager_two_peers():
    dhts = make_dht_swarm(2)
    models = [nn.Linear(10, 1, bias=False) for _ in range(2)]
    averagers = [
        GradientAverager(
            model.parameters(), dht, prefix="gatest", target_group_size=2,
            min_matchmaking_time=1.0, request_timeout=0.5, start=True,
        )
        for model, dht in zip(models, dhts)
    ]
    # peer 0: grad = ones, 8 samples; peer 1: grad = -ones, 24 samples
    for i, (model, grads, samples) in enumerate(zip(models, [1.0, -1.0], [8, 24])):
        out = model(torch.ones(samples, 10))
        loss = (out * grads).mean()
        loss.backward()
        averagers[i].accumulate_grads_(samples)

    controls = [avg.step(wait=False, timeout=60) for avg in averagers]
    for c in controls:
        assert c.result(90) is not None

    # peer i's local grad: d/dw mean_s(sign_i * w@x_s) = sign_i * ones
    # expected swarm average: (8 * ones + 24 * (-ones)) / 32 = -0.5 * ones
    expected = (8 * torch.ones(10) + 24 * (-torch.ones(10))) / 32
    for avg in averagers:
        with avg.use_averaged_gradients() as _:
            got = models[0].weight.grad.flatten() if avg is averagers[0] else models[1].weight.grad.flatten()
            assert torch.allclose(got, expected, atol=1e-4), (got, expected)
    for avg in averagers:
        avg.shutdown()
    for d in dhts:
        d.shutdown()


def test_state_averager_basic():
    dhts = make_dht_swarm(2)
    models = [nn.Linear(6, 2) for _ in range(2)]
    avgs = [
        TrainingStateAverager(
            dht=dht,
            optimizer=lambda pg: torch.optim.SGD(pg, lr=0.1),
            params=list(model.parameters()),
            prefix="satest",
            target_group_size=2,
            min_matchmaking_time=1.0,
            request_timeout=0.5,
            start=True,
        )
        for model, dht in zip(models, dhts)
    ]
    expected = [
        (p1.detach() + p2.detach()) / 2 for p1, p2 in zip(models[0].parameters(), models[1].parameters())
    ]
    results = []
    threads = [
        threading.Thread(target=lambda a=a: results.append(a.step(averaging_round=True, averaging_opts=dict(timeout=30))))
        for a in avgs
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join(60)
    for model in models:
        for got, ref in zip(model.parameters(), expected):
            assert torch.allclose(got.detach(), ref, atol=1e-5)
    for a in avgs:
        a.shutdown()
    for d in dhts:
        d.shutdown()


def _run_training_peer(dht, X, y, target_batch_size, batch_per_step, results, idx, reuse_grad_buffers=False, **opt_kwargs):
    torch.manual_seed(idx)
    model = nn.Sequential(nn.Linear(5, 16), nn.ReLU(), nn.Linear(16, 2))
    opt = Optimizer(
        dht=dht,
        run_id="conv_test",
        target_batch_size=target_batch_size,
        batch_size_per_step=batch_per_step,
        optimizer=lambda pg: torch.optim.SGD(pg, lr=0.5),  # grads are per-sample averaged
        params=[{"params": list(model.parameters())}],
        matchmaking_time=1.0,
        averaging_timeout=30.0,
        reuse_grad_buffers=reuse_grad_buffers,
        averager_opts=dict(request_timeout=0.5, min_group_size=2),
        tracker_opts=dict(min_refresh_period=0.2, default_refresh_period=0.5),
        verbose=False,
        **opt_kwargs,
    )
    rng = np.random.RandomState(idx)
    for step in range(100):
        sel = rng.choice(len(X), batch_per_step)
        xb, yb = X[sel], y[sel]
        logits = model(xb)
        loss = F.cross_entropy(logits, yb)
        loss.backward()
        opt.step()
        if not reuse_grad_buffers:
            opt.zero_grad()
        if opt.local_epoch >= 7:
            break
    with torch.no_grad():
        acc = (model(X).argmax(-1) == y).float().mean().item()
    results[idx] = (acc, opt.local_epoch)
    opt.shutdown()


def test_optimizer_convergence_two_peers():
    """Two peers jointly train a toy classifier to high accuracy
    (reference test_optimizer.py:344 / test_training.py:18 scenario)."""
    torch.manual_seed(0)
    n = 256
    X = torch.randn(n, 5)
    w_true = torch.randn(5, 2)
    y = (X @ w_true).argmax(-1)

    dhts = make_dht_swarm(2)
    results = [None, None]
    threads = [
        threading.Thread(
            target=_run_training_peer,
            args=(dhts[i], X, y, 64, 16, results, i),
        )
        for i in range(2)
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join(180)
    assert all(r is not None for r in results), f"some peers did not finish: {results}"
    for acc, epoch in results:
        assert epoch >= 7, f"peer stopped at epoch {epoch}"
        assert acc > 0.75, f"accuracy too low: {acc}"
    for d in dhts:
        d.shutdown()


def test_training_averager_legacy():
    """Legacy TrainingAverager: step() averages params and optimizer statistics
    across two peers (reference test for optim/training_averager.py)."""
    from hivemind_amd.optim.training_averager import TrainingAverager

    dhts = make_dht_swarm(2)
    models = [nn.Linear(4, 1, bias=False) for _ in range(2)]
    with torch.no_grad():
        models[0].weight.fill_(1.0)
        models[1].weight.fill_(3.0)
    opts = [torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9) for m in models]
    averagers = [
        TrainingAverager(
            opt, average_parameters=True, average_gradients=False,
            average_opt_statistics=("momentum_buffer",),
            dht=dht, start=True, prefix="legacy_ta", target_group_size=2,
            min_matchmaking_time=1.0, request_timeout=0.5,
        )
        for opt, dht in zip(opts, dhts)
    ]
    results = [None, None]

    def run(i):
        results[i] = averagers[i].step(timeout=60)

    threads = [threading.Thread(target=run, args=(i,)) for i in range(2)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(90)
    assert all(r is not None for r in results), results
    for m in models:
        assert torch.allclose(m.weight.detach(), torch.full((1, 4), 2.0), atol=1e-4), m.weight
    for avg in averagers:
        avg.shutdown()
    for d in dhts:
        d.shutdown()


def test_optimizer_state_dict_roundtrip():
    """state_dict/load_state_dict preserve inner optimizer state + local_epoch
    (reference optimizer.py state_dict compatibility)."""
    from hivemind_amd import Optimizer

    dht = DHT(start=True)
    model = nn.Linear(6, 2)
    opt = Optimizer(
        dht=dht, run_id="sdict", target_batch_size=8, batch_size_per_step=4,
        optimizer=lambda pg: torch.optim.SGD(pg, lr=0.1, momentum=0.9),
        params=[{"params": list(model.parameters())}],
        matchmaking_time=0.5, averaging_timeout=20.0,
        tracker_opts=dict(min_refresh_period=0.1, default_refresh_period=0.2),
    )
    try:
        X = torch.randn(16, 6)
        for _ in range(4):
            model(X).pow(2).mean().backward()
            opt.step()
            opt.zero_grad()
        sd = opt.state_dict()
        assert "state" in sd and "local_epoch" in sd["state"]
        epoch_before = opt.local_epoch
        opt.load_state_dict(sd)
        assert opt.local_epoch == epoch_before
    finally:
        opt.shutdown()
        dht.shutdown()
