"""Property-based fuzzing (hypothesis) of the CPU reference paths: FusedLinear
autograd vs a plain torch.nn.Linear graph, and weighted_loss vs a hand-built
reference, over random shapes/activations/dtypes."""
import numpy as np
import pytest
import torch

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from shifu_amd.ops.linear import FusedLinear
from shifu_amd.ops.loss import weighted_loss

ACTS = ["none", "sigmoid", "tanh", "relu", "leakyrelu"]


@settings(max_examples=25, deadline=None)
@given(b=st.integers(1, 33), fin=st.integers(1, 40), fout=st.integers(1, 24),
       act=st.sampled_from(ACTS), seed=st.integers(0, 10_000))
def test_fused_linear_matches_torch(b, fin, fout, act, seed):
    torch.manual_seed(seed)
    layer = FusedLinear(fin, fout, activation=act, seed=seed)
    ref = torch.nn.Linear(fin, fout)
    with torch.no_grad():
        ref.weight.copy_(layer.weight)
        ref.bias.copy_(layer.bias)
    x = torch.randn(b, fin, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)

    y = layer(x)
    act_fn = {"none": lambda t: t, "sigmoid": torch.sigmoid, "tanh": torch.tanh,
              "relu": torch.relu,
              "leakyrelu": lambda t: torch.nn.functional.leaky_relu(t, 0.01)}[act]
    y2 = act_fn(ref(x2))
    assert torch.allclose(y, y2, atol=1e-5), (b, fin, fout, act)

    g = torch.randn_like(y)
    y.backward(g)
    y2.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(layer.weight.grad, ref.weight.grad, atol=1e-5)
    assert torch.allclose(layer.bias.grad, ref.bias.grad, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(1, 200), kind=st.sampled_from(["weighted_mse", "sigmoid_ce"]),
       seed=st.integers(0, 10_000),
       wzero=st.booleans())
def test_weighted_loss_properties(n, kind, seed, wzero):
    torch.manual_seed(seed)
    z = torch.randn(n, requires_grad=True)
    y = (torch.rand(n) > 0.5).float()
    w = torch.rand(n) + 0.1
    if wzero:
        w[::2] = 0.0   # zero-weight rows must not contribute
    loss = weighted_loss(z, y, w, kind)
    assert torch.isfinite(loss)
    assert float(loss) >= 0.0
    loss.backward()
    assert torch.isfinite(z.grad).all()
    # zero-weight rows get zero gradient
    if wzero:
        assert torch.all(z.grad[::2] == 0.0)
    # gradient direction sanity: increasing a logit whose target is 1 must not
    # increase the loss (dL/dz <= 0 where y=1) for sigmoid_ce
    if kind == "sigmoid_ce" and not wzero:
        assert torch.all(z.grad[y == 1.0] <= 1e-7)
