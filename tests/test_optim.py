"""External-grad optimizers vs torch.optim references (reference math:
optim/sgd.py:57-89, optim/adam.py:37-93)."""

import torch

from atomo_amd.optim import ExternalAdam, ExternalSGD


def _clone_setup(n=100):
    torch.manual_seed(0)
    p_ref = torch.randn(n, requires_grad=True)
    flat = p_ref.detach().clone()
    grads = [torch.randn(n) for _ in range(5)]
    return p_ref, flat, grads


def test_external_sgd_matches_torch():
    p_ref, flat, grads = _clone_setup()
    torch_opt = torch.optim.SGD([p_ref], lr=0.1, momentum=0.9, weight_decay=0.01)
    mine = ExternalSGD(flat, lr=0.1, momentum=0.9, weight_decay=0.01)
    for g in grads:
        p_ref.grad = g.clone()
        torch_opt.step()
        mine.step(g)
    assert torch.allclose(p_ref.detach(), flat, atol=1e-6)


def test_external_sgd_nesterov():
    p_ref, flat, grads = _clone_setup()
    torch_opt = torch.optim.SGD([p_ref], lr=0.05, momentum=0.8, nesterov=True)
    mine = ExternalSGD(flat, lr=0.05, momentum=0.8, nesterov=True)
    for g in grads:
        p_ref.grad = g.clone()
        torch_opt.step()
        mine.step(g)
    assert torch.allclose(p_ref.detach(), flat, atol=1e-6)


def test_external_adam_matches_torch():
    p_ref, flat, grads = _clone_setup()
    torch_opt = torch.optim.Adam([p_ref], lr=0.01)
    mine = ExternalAdam(flat, lr=0.01)
    for g in grads:
        p_ref.grad = g.clone()
        torch_opt.step()
        mine.step(g)
    assert torch.allclose(p_ref.detach(), flat, atol=1e-6)


def test_sgd_state_roundtrip():
    _, flat, grads = _clone_setup()
    mine = ExternalSGD(flat, lr=0.1, momentum=0.9)
    mine.step(grads[0])
    sd = mine.state_dict()
    flat2 = flat.clone()
    other = ExternalSGD(flat2, lr=0.1, momentum=0.9)
    other.load_state_dict(sd)
    mine.step(grads[1])
    other.step(grads[1])
    assert torch.allclose(flat, flat2)


def test_sgd_state_load_is_in_place():
    """A captured hipGraph holds the momentum buffer's pointer: loading
    optimizer state must copy INTO the existing tensor."""
    import torch

    from atomo_amd.optim import make_optimizer

    flat = torch.zeros(16)
    opt = make_optimizer("sgd", flat, lr=0.1, momentum=0.9)
    buf_before = opt.buf
    opt.buf.fill_(1.0)
    sd = opt.state_dict()
    opt.buf.fill_(5.0)
    opt.load_state_dict(sd)
    assert opt.buf is buf_before
    assert torch.all(opt.buf == 1.0)
