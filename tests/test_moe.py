"""MoE tests (reference shape: tests/test_moe.py, test_connection_handler.py)."""

import time

import pytest
import torch
import torch.nn as nn

from hivemind_amd import DHT
from hivemind_amd.moe import (
    ModuleBackend,
    RemoteExpert,
    RemoteMixtureOfExperts,
    RemoteSwitchMixtureOfExperts,
    Server,
    declare_experts,
    get_experts,
)
from hivemind_amd.moe.client.beam_search import MoEBeamSearcher
from hivemind_amd.moe.expert_uid import is_valid_prefix, is_valid_uid, split_uid
from hivemind_amd.moe.server.dht_handler import get_expert_infos
from hivemind_amd.moe.server.layers import name_to_block


def test_expert_uid_grammar():
    assert is_valid_uid("ffn.0")
    assert is_valid_uid("expert.3.5.7")
    assert not is_valid_uid("expert")
    assert not is_valid_uid("expert.")
    assert not is_valid_uid("expert.03")
    assert split_uid("ffn.3.5") == ("ffn.3.", 5)
    assert is_valid_prefix("ffn.3.")


def test_server_and_remote_expert_forward_backward():
    dht = DHT(start=True)
    server = Server.create(
        dht=dht, expert_uids=["ffn_test.0", "ffn_test.1"], expert_cls="ffn", hidden_dim=16,
        optim_cls=torch.optim.SGD, device="cpu", start=True,
    )
    try:
        experts = get_experts(dht, ["ffn_test.0", "ffn_test.1", "ffn_test.99"])
        assert experts[0] is not None and experts[1] is not None and experts[2] is None
        expert = experts[0]
        x = torch.randn(4, 16, requires_grad=True)
        out = expert(x)
        assert out.shape == (4, 16)
        # gradients flow through the RPC boundary
        loss = out.pow(2).sum()
        loss.backward()
        assert x.grad is not None and x.grad.shape == x.shape
        assert x.grad.abs().sum() > 0
    finally:
        server.shutdown()
        dht.shutdown()


def test_remote_expert_streaming_large_tensor():
    dht = DHT(start=True)
    server = Server.create(
        dht=dht, expert_uids=["big.0"], expert_cls="ffn", hidden_dim=1024,
        optim_cls=None, device="cpu", start=True,
    )
    try:
        (expert,) = get_experts(dht, ["big.0"])
        x = torch.randn(600, 1024, requires_grad=True)  # ~2.4 MB > unary cutoff
        out = expert(x)
        assert out.shape == (600, 1024)
        out.sum().backward()
        assert x.grad is not None
    finally:
        server.shutdown()
        dht.shutdown()


def test_beam_search_vs_brute_force():
    """Beam search with a wide beam must find the true best experts
    (reference test_moe.py:186)."""
    dht = DHT(start=True)
    grid = (4, 4)
    all_uids = [f"grid.{i}.{j}" for i in range(grid[0]) for j in range(grid[1])]
    declare_experts(dht, all_uids, expiration_time=time.time() + 60)

    searcher = MoEBeamSearcher(dht, "grid", grid)
    torch.manual_seed(7)
    for _ in range(5):
        scores = [torch.randn(grid[0]).tolist(), torch.randn(grid[1]).tolist()]
        found = searcher.find_best_expert_infos(scores, beam_size=16)
        found_scores = sorted(
            (scores[0][int(u.uid.split(".")[1])] + scores[1][int(u.uid.split(".")[2])] for u in found),
            reverse=True,
        )
        brute = sorted(
            (scores[0][i] + scores[1][j] for i in range(grid[0]) for j in range(grid[1])), reverse=True
        )
        assert len(found) == 16
        assert all(abs(a - b) < 1e-5 for a, b in zip(found_scores, brute))
    dht.shutdown()


def test_remote_mixture_of_experts_training():
    """Tiny MoE learns a separable task (reference test_training.py:59 shape)."""
    torch.manual_seed(0)
    dht = DHT(start=True)
    server = Server.create(
        dht=dht, expert_uids=[f"moe_ffn.{i}.0" for i in range(4)], expert_cls="ffn", hidden_dim=16,
        optim_cls=lambda p: torch.optim.SGD(p, lr=0.05), device="cpu", start=True,
    )
    try:
        moe = RemoteMixtureOfExperts(
            in_features=16, grid_size=(4, 1), dht=dht, uid_prefix="moe_ffn", k_best=2,
            forward_timeout=15, backward_timeout=15,
        )
        head = nn.Linear(16, 2)
        opt = torch.optim.SGD(list(moe.proj.parameters()) + list(head.parameters()), lr=0.05)
        X = torch.randn(64, 16)
        y = (X[:, 0] > 0).long()
        initial_loss = None
        for step in range(12):
            logits = head(moe(X))
            loss = torch.nn.functional.cross_entropy(logits, y)
            if initial_loss is None:
                initial_loss = loss.item()
            opt.zero_grad()
            loss.backward()
            opt.step()
        assert loss.item() < initial_loss, (initial_loss, loss.item())
    finally:
        server.shutdown()
        dht.shutdown()


def test_switch_moe_forward():
    torch.manual_seed(0)
    dht = DHT(start=True)
    server = Server.create(
        dht=dht, expert_uids=[f"sw.{i}.0" for i in range(4)], expert_cls="ffn", hidden_dim=16,
        optim_cls=lambda p: torch.optim.SGD(p, lr=0.05), device="cpu", start=True,
    )
    try:
        moe = RemoteSwitchMixtureOfExperts(
            in_features=16, grid_size=(4, 1), dht=dht, uid_prefix="sw",
            forward_timeout=15, backward_timeout=15, jitter_eps=0.0,
        )
        X = torch.randn(8, 16)
        out, balancing_loss = moe(X)
        assert out.shape == (8, 16)
        assert torch.isfinite(balancing_loss)
        (out.sum() + balancing_loss).backward()
    finally:
        server.shutdown()
        dht.shutdown()


def test_moe_tolerates_dead_experts():
    """Some chosen experts don't exist; MoE must still produce outputs
    (reference test_moe.py:71 fault-tolerant _RemoteCallMany)."""
    torch.manual_seed(0)
    dht = DHT(start=True)
    # declare 4 experts but only actually serve 2 of them
    all_uids = [f"half.{i}.0" for i in range(4)]
    declare_experts(dht, all_uids, expiration_time=time.time() + 60)
    server = Server.create(
        dht=dht, expert_uids=all_uids[:2], expert_cls="ffn", hidden_dim=16,
        optim_cls=None, device="cpu", start=True,
    )
    try:
        moe = RemoteMixtureOfExperts(
            in_features=16, grid_size=(4, 1), dht=dht, uid_prefix="half", k_best=4, k_min=1,
            forward_timeout=10, backward_timeout=10, timeout_after_k_min=0.5,
        )
        X = torch.randn(3, 16)
        out = moe(X)
        assert out.shape == (3, 16)
        out.sum().backward()
    finally:
        server.shutdown()
        dht.shutdown()


def test_client_anomaly_detection():
    """detect_anomalies: nan/inf inputs raise; experts that return non-finite
    outputs are treated as dead (reference test_moe.py client anomaly test)."""
    from hivemind_amd.moe import register_expert_class

    @register_expert_class("nan_expert_test", lambda batch, hid: torch.empty((batch, hid)))
    class NaNExpert(nn.Module):
        def __init__(self, hid_dim):
            super().__init__()
            self.lin = nn.Linear(hid_dim, hid_dim)

        def forward(self, x):
            return self.lin(x) * float("nan")

    torch.manual_seed(0)
    dht = DHT(start=True)
    good = Server.create(
        dht=dht, expert_uids=["anom.0.0", "anom.1.0"], expert_cls="ffn", hidden_dim=16,
        optim_cls=None, device="cpu", start=True,
    )
    bad = Server.create(
        dht=DHT(initial_peers=[dht.endpoint], start=True),
        expert_uids=["anom.2.0", "anom.3.0"], expert_cls="nan_expert_test", hidden_dim=16,
        optim_cls=None, device="cpu", start=True,
    )
    try:
        moe = RemoteMixtureOfExperts(
            in_features=16, grid_size=(4, 1), dht=dht, uid_prefix="anom", k_best=4, k_min=1,
            forward_timeout=15, backward_timeout=15, detect_anomalies=True,
        )
        # nan input is rejected outright
        bad_input = torch.randn(2, 16)
        bad_input[0, 0] = float("nan")
        with pytest.raises(ValueError, match="nan/inf"):
            moe(bad_input)
        # nan-producing experts are dropped; the mixture output stays finite
        out = moe(torch.randn(4, 16))
        assert torch.isfinite(out).all()
        out.sum().backward()
        assert torch.isfinite(moe.proj.weight.grad).all()
    finally:
        good.shutdown()
        bad.shutdown()
        dht.shutdown()


def test_switch_moe_training():
    """Switch-MoE (top-1 routing + load-balancing loss) learns a separable task
    (reference test_training.py:112 shape)."""
    torch.manual_seed(0)
    dht = DHT(start=True)
    server = Server.create(
        dht=dht, expert_uids=[f"swtr.{i}.0" for i in range(4)], expert_cls="ffn", hidden_dim=16,
        optim_cls=lambda p: torch.optim.SGD(p, lr=0.05), device="cpu", start=True,
    )
    try:
        moe = RemoteSwitchMixtureOfExperts(
            in_features=16, grid_size=(4, 1), dht=dht, uid_prefix="swtr",
            forward_timeout=15, backward_timeout=15, jitter_eps=0.0,
        )
        head = nn.Linear(16, 2)
        opt = torch.optim.SGD(list(moe.proj.parameters()) + list(head.parameters()), lr=0.05)
        X = torch.randn(64, 16)
        y = (X[:, 0] > 0).long()
        initial_loss = None
        for step in range(12):
            out, balancing_loss = moe(X)
            loss = torch.nn.functional.cross_entropy(head(out), y) + 0.01 * balancing_loss
            if initial_loss is None:
                initial_loss = loss.item()
            opt.zero_grad()
            loss.backward()
            opt.step()
        assert loss.item() < initial_loss, (initial_loss, loss.item())
    finally:
        server.shutdown()
        dht.shutdown()


def test_expert_server_behind_relay():
    """An expert server with NO inbound sockets serves through a circuit relay:
    handler loops register with a public relay peer and advertise relay://
    endpoints in the DHT (the reference's libp2p relay / NAT traversal)."""
    from hivemind_amd.p2p import P2P
    from hivemind_amd.utils.asyncio_utils import EventLoopThread

    # public relay peer on its own loop
    relay_loop = EventLoopThread(name="relay")
    relay_loop.start_and_wait()
    import asyncio as _asyncio

    relay = _asyncio.run_coroutine_threadsafe(P2P.create(), relay_loop.loop).result(15)

    dht = DHT(start=True)  # the DHT node itself is public here; only the
    # expert server's RPC plane goes through the relay
    server = Server.create(
        dht=dht, expert_uids=["nat.0", "nat.1"], expert_cls="ffn", hidden_dim=16,
        optim_cls=None, device="cpu", num_connection_handlers=2,
        relay_endpoint=relay.endpoint, start=True,
    )
    try:
        experts = get_experts(dht, ["nat.0", "nat.1"])
        assert all(e is not None for e in experts)
        x = torch.randn(3, 16, requires_grad=True)
        out = experts[1](x)
        assert out.shape == (3, 16)
        out.sum().backward()
        assert x.grad is not None
    finally:
        server.shutdown()
        dht.shutdown()
        _asyncio.run_coroutine_threadsafe(relay.shutdown(), relay_loop.loop).result(10)
        relay_loop.shutdown()


def test_background_server_context():
    """background_server context manager serves experts for the with-block
    (reference server.py:308 background_server)."""
    from hivemind_amd.moe import background_server

    with background_server(
        expert_uids=["bgctx.0"], expert_cls="ffn", hidden_dim=8, optim_cls=None, device="cpu"
    ) as server_info:
        dht = DHT(initial_peers=[f"{ep}" for ep in server_info.endpoints], start=True)
        (expert,) = get_experts(dht, ["bgctx.0"])
        assert expert is not None
        out = expert(torch.randn(2, 8))
        assert out.shape == (2, 8)
        dht.shutdown()


def test_server_expert_scheduler_and_clipping():
    """Server-side expert training with linear warmup scheduler and gradient
    clipping (reference server schedulers + ClippingWrapper)."""
    dht = DHT(start=True)
    server = Server.create(
        dht=dht, expert_uids=["sched.0"], expert_cls="ffn", hidden_dim=8,
        optim_cls=lambda p: torch.optim.SGD(p, lr=1.0), scheduler="linear",
        num_warmup_steps=4, num_total_steps=10, clip_grad_norm=0.5,
        device="cpu", start=True,
    )
    try:
        backend = server.module_backends["sched.0"]
        (expert,) = get_experts(dht, ["sched.0"])
        x = torch.randn(2, 8, requires_grad=True)
        for _ in range(3):
            out = expert(x)
            out.sum().backward()
        # warmup schedule advanced once per backward: lr = step/num_warmup * base
        lr = backend.optimizer.param_groups[0]["lr"]
        assert lr == pytest.approx(3 / 4, rel=1e-3), lr
        # clipping kept the applied update bounded: weights moved, but not far
        total_sq = sum((p.detach() ** 2).sum() for p in backend.module.parameters())
        assert torch.isfinite(torch.as_tensor(total_sq))
    finally:
        server.shutdown()
        dht.shutdown()
