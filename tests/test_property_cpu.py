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


@settings(max_examples=20, deadline=None)
@given(b=st.integers(1, 17), f=st.integers(1, 6), d=st.integers(1, 9),
       seed=st.integers(0, 10_000))
def test_fm_second_order_equals_pairwise(b, f, d, seed):
    """fm2[b] must equal the brute-force sum of pairwise dot products
    sum_{i<j} <v_i, v_j> (the identity the fused form exploits)."""
    from shifu_amd.ops.fm import fm_second_order
    torch.manual_seed(seed)
    v = torch.randn(b, f, d, requires_grad=True)
    out = fm_second_order(v.reshape(b, f * d), f, d)
    brute = torch.zeros(b)
    for i in range(f):
        for j in range(i + 1, f):
            brute += (v[:, i] * v[:, j]).sum(dim=1)
    assert torch.allclose(out, brute, atol=1e-4), (b, f, d)
    out.sum().backward()
    if f == 1:
        # no pairs: fm2 == 0 and its gradient vanishes (s - v == 0)
        assert torch.allclose(v.grad, torch.zeros_like(v.grad), atol=1e-5)
        return
    v2 = v.detach().clone().requires_grad_(True)
    brute2 = torch.zeros(b)
    for i in range(f):
        for j in range(i + 1, f):
            brute2 += (v2[:, i] * v2[:, j]).sum(dim=1)
    brute2.sum().backward()
    assert torch.allclose(v.grad, v2.grad, atol=1e-4)


@settings(max_examples=20, deadline=None)
@given(b=st.integers(1, 20), d=st.integers(1, 12), seed=st.integers(0, 10_000))
def test_multi_embedding_matches_torch_embedding(b, d, seed):
    from shifu_amd.ops.embedding import MultiEmbedding
    torch.manual_seed(seed)
    vocab = [7, 13]
    emb = MultiEmbedding(vocab, d, seed=seed)
    t0 = torch.nn.Embedding(7, d)
    t1 = torch.nn.Embedding(13, d)
    with torch.no_grad():
        t0.weight.copy_(emb.arena[:7])
        t1.weight.copy_(emb.arena[7:])
    g = torch.Generator().manual_seed(seed + 1)
    ids = torch.stack([torch.randint(0, 7, (b,), generator=g),
                       torch.randint(0, 13, (b,), generator=g)], dim=1)
    out = emb(ids)
    ref = torch.cat([t0(ids[:, 0]), t1(ids[:, 1])], dim=1)
    assert torch.allclose(out, ref, atol=1e-6)
    gr = torch.randn_like(out)
    out.backward(gr)
    ref.backward(gr)
    dense = emb.arena.grad.coalesce().to_dense()
    assert torch.allclose(dense[:7], t0.weight.grad, atol=1e-6)
    assert torch.allclose(dense[7:], t1.weight.grad, atol=1e-6)


@settings(max_examples=15, deadline=None)
@given(n=st.integers(1, 60), nd=st.integers(1, 5), seed=st.integers(0, 10_000),
       gz=st.booleans())
def test_csv_parsers_agree(tmp_path_factory, n, nd, seed, gz):
    """Fuzz: the native C++ reader and the pure-Python parser must agree on
    arbitrary well-formed rows (values spanning magnitudes and signs)."""
    import os
    from shifu_amd.data.csv_loader import load_csv_files
    from shifu_amd.io import load_csv_native, native_io
    rng = np.random.default_rng(seed)
    rows = []
    for _ in range(n):
        t = float(rng.integers(0, 2))
        w = float(abs(rng.standard_normal()))
        feats = rng.standard_normal(nd) * (10.0 ** rng.integers(-6, 7, nd))
        cat = int(rng.integers(0, 1000))
        rows.append([t, w] + [float(x) for x in feats] + [cat])
    td = tmp_path_factory.mktemp("csv")
    path = os.path.join(str(td), "part.csv" + (".gz" if gz else ""))
    import gzip as gzmod
    op = (lambda p: gzmod.open(p, "wt")) if gz else (lambda p: open(p, "w"))
    with op(path) as f:
        for r in rows:
            f.write("|".join(repr(v) for v in r) + "\n")
    cols = dict(selected_numeric=list(range(2, 2 + nd)),
                selected_categorical=[2 + nd], target_column=0, weight_column=1)
    py = load_csv_files([path], **cols)
    nat = load_csv_native([path], **cols)
    if native_io() is None:
        pytest.skip("native reader not built")
    assert len(py) == len(nat) == n
    np.testing.assert_allclose(nat.dense, py.dense, rtol=1e-6, atol=1e-30)
    np.testing.assert_array_equal(nat.cats, py.cats)
    np.testing.assert_allclose(nat.target, py.target, atol=1e-6)
    np.testing.assert_allclose(nat.weight, py.weight, rtol=1e-6)
