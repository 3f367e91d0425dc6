"""GPU MoE test: expert server on cuda with fused kernels, remote training step."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs an MI355X")


@requires_gpu
def test_gpu_expert_server_forward_backward():
    from hivemind_amd import DHT
    from hivemind_amd.moe import Server, get_experts

    dht = DHT(start=True)
    server = Server.create(
        dht=dht, expert_uids=["gpu_ffn.0", "gpu_ffn.1"], expert_cls="ffn", hidden_dim=1024,
        optim_cls=torch.optim.Adam, device="cuda", start=True,
    )
    try:
        experts = get_experts(dht, ["gpu_ffn.0", "gpu_ffn.1"])
        assert all(e is not None for e in experts)
        x = torch.randn(64, 1024, requires_grad=True)
        out = experts[0](x)
        assert out.shape == (64, 1024)
        assert torch.isfinite(out).all()
        out.pow(2).sum().backward()
        assert x.grad is not None and torch.isfinite(x.grad).all()
    finally:
        server.shutdown()
        dht.shutdown()


def test_rccl_moe_dispatch_gpu():
    """RCCL all-to-all expert dispatch module runs on device (world=1 here;
    the multi-rank path is covered by the 2-process gloo test and composes
    from the same collectives the 8-GPU node uses)."""
    from hivemind_amd.moe.rccl_dispatch import RcclMixtureOfExperts

    torch.manual_seed(0)
    moe = RcclMixtureOfExperts(256, num_local_experts=4, k=2).cuda()
    x = torch.randn(64, 256, device="cuda", requires_grad=True)
    out = moe(x)
    out.sum().backward()
    assert out.shape == x.shape
    assert torch.isfinite(out).all() and torch.isfinite(x.grad).all()
    assert all(e.up.weight.grad is not None for e in moe.experts)
