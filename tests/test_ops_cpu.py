"""CPU-path op tests: fused_linear / tanh_gauss against torch compositions
(these CPU implementations are themselves the oracles for the HIP kernels in
the GPU-marked tests)."""

import math

import torch
import torch.nn.functional as F
from torch.distributions.normal import Normal

from smartcal_amd.ops.linear import FusedLinear, fused_linear
from smartcal_amd.ops.sampling import tanh_gauss_sample


def test_fused_linear_cpu_matches_composition():
    torch.manual_seed(0)
    B, K, N = 5, 11, 7
    x = torch.randn(B, K)
    W = torch.randn(N, K)
    b = torch.randn(N)
    g = torch.randn(N)
    be = torch.randn(N)
    y = fused_linear(x, W, b, g, be, act="elu")
    ref = F.elu(F.layer_norm(F.linear(x, W, b), (N,), g, be))
    assert torch.allclose(y, ref, atol=1e-6)


def test_fused_linear_module_shapes_and_grad():
    m = FusedLinear(6, 4)
    x = torch.randn(3, 6, requires_grad=True)
    y = m(x)
    assert y.shape == (3, 4)
    y.sum().backward()
    assert x.grad is not None
    assert m.weight.grad is not None
    assert m.ln_weight.grad is not None


def test_tanh_gauss_matches_reference_composition():
    torch.manual_seed(0)
    B, A = 4, 3
    mu = torch.randn(B, A, requires_grad=True)
    logsigma = (torch.randn(B, A) * 0.2).requires_grad_(True)
    eps = torch.randn(B, A)
    max_action = 1.0
    action, logprob = tanh_gauss_sample(mu, logsigma, max_action,
                                        reparameterize=True, eps=eps)

    # reference composition (enet_sac.py:446-466) with the same eps
    sigma = logsigma.exp()
    z = mu + sigma * eps
    probs = Normal(mu, sigma)
    a_t = torch.tanh(z)
    ref_action = a_t * max_action
    lp = probs.log_prob(z)
    lp = lp - torch.log(max_action * (1 - a_t.pow(2)) + 1e-6)
    ref_logprob = lp.sum(1, keepdim=True)

    assert torch.allclose(action, ref_action, atol=1e-6)
    assert torch.allclose(logprob, ref_logprob, atol=1e-5)

    # gradients of both paths agree
    (action.sum() + logprob.sum()).backward()
    g_mu = mu.grad.clone()
    g_ls = logsigma.grad.clone()
    mu.grad = None
    logsigma.grad = None
    (ref_action.sum() + ref_logprob.sum()).backward()
    assert torch.allclose(g_mu, mu.grad, atol=1e-5)
    assert torch.allclose(g_ls, logsigma.grad, atol=1e-5)


def test_tanh_gauss_logprob_constant():
    # the Normal log-prob at z = mu + sigma*eps equals
    # -eps^2/2 - logsigma - log sqrt(2 pi)
    mu = torch.zeros(1, 1)
    logsigma = torch.zeros(1, 1)
    eps = torch.zeros(1, 1)
    _, logprob = tanh_gauss_sample(mu, logsigma, 1.0, reparameterize=False,
                                   eps=eps)
    expected = -0.5 * math.log(2 * math.pi) - math.log(1.0 * 1.0 + 1e-6)
    assert abs(float(logprob) - expected) < 1e-5
