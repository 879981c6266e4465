"""CPU reference-path op tests: fused linear fwd/bwd vs torch autograd,
weighted losses vs hand math, embedding sparse grads, fused optimizers vs
torch optimizers."""
import math

import numpy as np
import pytest
import torch

from shifu_amd.ops.embedding import MultiEmbedding
from shifu_amd.ops.flat import FlatParams, split_params
from shifu_amd.ops.linear import FusedLinear, fused_linear
from shifu_amd.ops.loss import weighted_loss
from shifu_amd.ops.optim import FusedOptimizer


@pytest.mark.parametrize("act", ["none", "sigmoid", "tanh", "relu", "leakyrelu"])
def test_fused_linear_matches_autograd(act):
    torch.manual_seed(0)
    x = torch.randn(16, 7, requires_grad=True)
    w = torch.randn(5, 7, requires_grad=True)   # [out, in]
    b = torch.randn(5, requires_grad=True)

    y = fused_linear(x, w, b, act)
    loss = (y * torch.arange(y.numel()).reshape(y.shape).float()).sum()
    loss.backward()

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    z = x2 @ w2.t() + b2
    acts = {"none": lambda t: t, "sigmoid": torch.sigmoid, "tanh": torch.tanh,
            "relu": torch.relu,
            "leakyrelu": lambda t: torch.nn.functional.leaky_relu(t, 0.01)}
    y2 = acts[act](z)
    loss2 = (y2 * torch.arange(y2.numel()).reshape(y2.shape).float()).sum()
    loss2.backward()

    assert torch.allclose(y, y2, atol=1e-6)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-5)
    assert torch.allclose(b.grad, b2.grad, atol=1e-5)


def test_fused_linear_module_xavier_init():
    layer = FusedLinear(100, 50, activation="tanh", seed=1)
    limit = math.sqrt(6.0 / 150)
    assert layer.weight.abs().max() <= limit + 1e-6
    assert layer.bias.abs().max() == 0.0
    layer2 = FusedLinear(100, 50, activation="tanh", seed=1)
    assert torch.equal(layer.weight, layer2.weight)  # deterministic


@pytest.mark.parametrize("kind", ["weighted_mse", "sigmoid_ce"])
def test_weighted_loss_value_and_grad(kind):
    torch.manual_seed(1)
    z = torch.randn(64, requires_grad=True)
    y = (torch.rand(64) > 0.5).float()
    w = torch.rand(64) * 2

    loss = weighted_loss(z, y, w, kind)
    loss.backward()

    p = torch.sigmoid(z.detach())
    nnz = (w != 0).float().sum()  # TF SUM_BY_NONZERO_WEIGHTS normalization
    if kind == "weighted_mse":
        expect = (w * (p - y) ** 2).sum() / nnz
        dz = w * 2 * (p - y) * p * (1 - p) / nnz
    else:
        expect = (w * torch.nn.functional.binary_cross_entropy_with_logits(
            z.detach(), y, reduction="none")).sum() / nnz
        dz = w * (p - y) / nnz
    assert torch.allclose(loss, expect, atol=1e-6)
    assert torch.allclose(z.grad, dz, atol=1e-6)


def test_multi_embedding_forward_and_sparse_grad():
    emb = MultiEmbedding([10, 20], dim=4, seed=0)
    ids = torch.tensor([[1, 5], [9, 19], [1, 5]])
    out = emb(ids)
    assert out.shape == (3, 8)
    # row 0 and 2 identical ids -> identical embeddings
    assert torch.equal(out[0], out[2])
    out.sum().backward()
    g = emb.arena.grad
    assert g.is_sparse
    gd = g.coalesce().to_dense()
    # duplicated (1,5) rows accumulate twice
    assert torch.allclose(gd[1], torch.full((4,), 2.0))
    assert torch.allclose(gd[10 + 5], torch.full((4,), 2.0))
    assert torch.allclose(gd[9], torch.ones(4))


def test_flat_params_binding_and_grad_accumulation():
    torch.manual_seed(0)
    m = torch.nn.Sequential(FusedLinear(4, 3, "relu"), FusedLinear(3, 1, "none"))
    dense, sparse = split_params(m)
    assert len(sparse) == 0
    flat = FlatParams(dense)
    assert flat.numel() == 4 * 3 + 3 + 3 * 1 + 1
    x = torch.randn(8, 4)
    y = m(x).sum()
    y.backward()
    flat.sync_grads()
    # flat grad holds every param grad
    for p, (off, n) in zip(flat.params, flat._offsets):
        assert torch.allclose(flat.flat_grad[off:off + n], p.grad.reshape(-1))
    # param data is a view of flat
    flat.flat.mul_(0.5)
    for p, (off, n) in zip(flat.params, flat._offsets):
        assert p.data.data_ptr() == flat.flat[off:off + n].data_ptr()


@pytest.mark.parametrize("opt", ["sgd", "adam", "adagrad"])
def test_fused_optimizer_matches_torch(opt):
    torch.manual_seed(0)
    w0 = torch.randn(50)

    p_ref = torch.nn.Parameter(w0.clone())
    torch_opt = {"sgd": lambda: torch.optim.SGD([p_ref], lr=0.1),
                 "adam": lambda: torch.optim.Adam([p_ref], lr=0.1),
                 "adagrad": lambda: torch.optim.Adagrad([p_ref], lr=0.1, eps=1e-8)}[opt]()

    p_mine = torch.nn.Parameter(w0.clone())
    flat = FlatParams([p_mine])
    mine = FusedOptimizer(flat, [], optimizer=opt, lr=0.1, l2_reg=0.0)

    for step in range(5):
        g = torch.randn(50, generator=torch.Generator().manual_seed(step))
        p_ref.grad = g.clone()
        torch_opt.step()
        flat.flat_grad.copy_(g)
        mine.step()
        assert torch.allclose(p_mine.data, p_ref.data, atol=1e-5), f"step {step}"


def test_adadelta_matches_torch():
    """TF/torch Adadelta semantics coincide (rho, eps inside both sqrts):
    compare the fused arena step against torch.optim.Adadelta."""
    torch.manual_seed(0)
    w0 = torch.randn(50)
    p_ref = torch.nn.Parameter(w0.clone())
    torch_opt = torch.optim.Adadelta([p_ref], lr=1.0, rho=0.95, eps=1e-8)
    p_mine = torch.nn.Parameter(w0.clone())
    flat = FlatParams([p_mine])
    mine = FusedOptimizer(flat, [], optimizer="adadelta", lr=1.0, l2_reg=0.0)
    for step in range(5):
        g = torch.randn(50, generator=torch.Generator().manual_seed(step))
        p_ref.grad = g.clone()
        torch_opt.step()
        flat.flat_grad.copy_(g)
        mine.step()
        assert torch.allclose(p_mine.data, p_ref.data, atol=1e-6), f"step {step}"


def test_l2_weight_decay_applied():
    p = torch.nn.Parameter(torch.ones(10))
    flat = FlatParams([p])
    opt = FusedOptimizer(flat, [], optimizer="sgd", lr=0.5, l2_reg=0.1)
    flat.flat_grad.zero_()
    opt.step()
    # w -= lr * (g + l2*w) = 1 - 0.5*0.1*1 = 0.95
    assert torch.allclose(p.data, torch.full((10,), 0.95))


def test_emb_optimizer_sgd_rowwise():
    emb = MultiEmbedding([10], dim=2, seed=0)
    before = emb.arena.data.clone()
    ids = torch.tensor([[3], [3], [7]])
    out = emb(ids)
    out.sum().backward()
    opt = FusedOptimizer(FlatParams([]), [emb.arena], optimizer="sgd", lr=0.0,
                         emb_optimizer="sgd", emb_lr=0.1)
    opt.step()
    # row 3 got grad 2.0 per dim, row 7 got 1.0
    assert torch.allclose(emb.arena.data[3], before[3] - 0.1 * 2.0 * torch.ones(2))
    assert torch.allclose(emb.arena.data[7], before[7] - 0.1 * 1.0 * torch.ones(2))
    assert torch.equal(emb.arena.data[0], before[0])


def test_fm_second_order_matches_autograd():
    from shifu_amd.ops.fm import fm_second_order
    torch.manual_seed(0)
    B, F, D = 8, 3, 4
    e1 = torch.randn(B, F * D, requires_grad=True)
    out = fm_second_order(e1, F, D)
    out.pow(2).sum().backward()

    e2 = e1.detach().clone().requires_grad_(True)
    v = e2.reshape(B, F, D)
    s = v.sum(dim=1)
    ref = 0.5 * (s * s - (v * v).sum(dim=1)).sum(dim=1)
    ref.pow(2).sum().backward()
    assert torch.allclose(out, ref, atol=1e-5)
    assert torch.allclose(e1.grad, e2.grad, atol=1e-5)
