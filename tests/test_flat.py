"""Flat parameter layout + fused RMSProp vs torch.optim.RMSprop oracle."""

import torch

from torchbeast_amd.models import AtariNet
from torchbeast_amd.parallel import flat as tbflat


def _tiny_net():
    torch.manual_seed(0)
    return torch.nn.Sequential(
        torch.nn.Linear(6, 16), torch.nn.ReLU(), torch.nn.Linear(16, 3)
    )


def test_flatten_preserves_forward():
    net = _tiny_net()
    x = torch.randn(5, 6)
    before = net(x).detach().clone()
    flat = tbflat.flatten_parameters(net)
    after = net(x).detach()
    torch.testing.assert_close(before, after)
    # Params are views of the flat buffer.
    flat.add_(1.0)
    assert not torch.allclose(net(x), after)


def test_flat_grads_accumulate():
    net = _tiny_net()
    tbflat.flatten_parameters(net)
    flat_grad = tbflat.attach_flat_grads(net)
    loss = net(torch.randn(4, 6)).sum()
    loss.backward()
    assert flat_grad.abs().sum() > 0
    grads = torch.cat([p.grad.reshape(-1) for p in net.parameters()])
    torch.testing.assert_close(grads, flat_grad)


def test_fused_rmsprop_matches_torch_rmsprop():
    lr, alpha, eps, clip = 0.01, 0.9, 0.05, 1.0

    net_a = _tiny_net()
    net_b = _tiny_net()
    net_b.load_state_dict(net_a.state_dict())

    flat_param = tbflat.flatten_parameters(net_a)
    flat_grad = tbflat.attach_flat_grads(net_a)
    fused = tbflat.FusedRMSProp(flat_param, flat_grad, lr, alpha, eps, clip)

    ref_opt = torch.optim.RMSprop(net_b.parameters(), lr=lr, alpha=alpha, eps=eps)

    for step in range(5):
        torch.manual_seed(100 + step)
        x = torch.randn(8, 6)
        fused.zero_grad()
        net_a(x).pow(2).sum().backward()
        fused.step()

        ref_opt.zero_grad()
        net_b(x).pow(2).sum().backward()
        torch.nn.utils.clip_grad_norm_(net_b.parameters(), clip)
        ref_opt.step()

    for pa, pb in zip(net_a.parameters(), net_b.parameters()):
        torch.testing.assert_close(pa, pb, rtol=1e-5, atol=1e-6)


def test_fused_rmsprop_state_roundtrip():
    net = _tiny_net()
    flat_param = tbflat.flatten_parameters(net)
    flat_grad = tbflat.attach_flat_grads(net)
    opt = tbflat.FusedRMSProp(flat_param, flat_grad, 0.01)
    sched = tbflat.LinearLR(opt, steps_per_update=10, total_steps=100)

    opt.zero_grad()
    net(torch.randn(2, 6)).sum().backward()
    opt.step()
    sched.step()

    state = {"opt": opt.state_dict(), "sched": sched.state_dict()}

    opt2 = tbflat.FusedRMSProp(flat_param.clone(), flat_grad.clone(), 0.02)
    sched2 = tbflat.LinearLR(opt2, 10, 100)
    opt2.load_state_dict(state["opt"])
    sched2.load_state_dict(state["sched"])
    torch.testing.assert_close(opt2.square_avg, opt.square_avg)
    assert sched2.updates == 1
    assert opt2.base_lr == 0.01


def test_linear_lr_decay_matches_reference_lambda():
    net = AtariNet((4, 36, 36), 4)
    flat_param = tbflat.flatten_parameters(net)
    flat_grad = tbflat.attach_flat_grads(net)
    opt = tbflat.FusedRMSProp(flat_param, flat_grad, 1.0)
    T, B, total = 8, 4, 320
    sched = tbflat.LinearLR(opt, T * B, total)
    for epoch in range(1, 11):
        sched.step()
        expected = 1 - min(epoch * T * B, total) / total
        assert abs(opt.lr_factor - expected) < 1e-9
