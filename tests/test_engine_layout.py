"""Forward-layout permutation: the engine's device layout must be a bijection
of the reference flat state_dict layout (checkpoint compatibility)."""
import numpy as np
import torch

from es_pytorch_amd.core.engine import forward_perm


def test_forward_perm_bijection():
    dims = [11, 64, 3]
    n = 11 * 64 + 64 + 64 * 3 + 3
    perm = forward_perm(dims)
    assert perm.shape == (n,)
    assert len(set(perm.tolist())) == n  # bijection


def test_forward_perm_matches_torch_linear():
    torch.manual_seed(0)
    dims = [5, 7, 2]
    lin1 = torch.nn.Linear(5, 7)
    lin2 = torch.nn.Linear(7, 2)
    flat = torch.cat([lin1.weight.flatten(), lin1.bias.flatten(),
                      lin2.weight.flatten(), lin2.bias.flatten()]).detach()
    perm = forward_perm(dims)
    fwd = flat[perm]
    # forward layout: W1^T (5,7), b1 (7), W2^T (7,2), b2 (2)
    w1t = fwd[:35].reshape(5, 7)
    assert torch.equal(w1t, lin1.weight.detach().T)
    b1 = fwd[35:42]
    assert torch.equal(b1, lin1.bias.detach())
    w2t = fwd[42:56].reshape(7, 2)
    assert torch.equal(w2t, lin2.weight.detach().T)
    # inverse: scatter back
    flat2 = torch.empty_like(flat)
    flat2[perm] = fwd
    assert torch.equal(flat2, flat)


def test_forward_perm_roundtrip_random():
    dims = [376, 256, 256, 17]
    n = sum(I * O + O for I, O in zip(dims[:-1], dims[1:]))
    perm = forward_perm(dims)
    x = torch.randn(n)
    y = torch.empty(n)
    y[perm] = x[perm]
    assert torch.equal(x, y)
