"""Model-layer tests.  Reference: tests/test_layers.py:7-70."""

import torch

from shallowspeed_amd.models import MLP, SGD, Linear, Sequential, SoftmaxMSE


def test_sequential_fwd_bwd_shapes():
    gb = 16
    model = Sequential([
        Linear(20, 16, activation="relu"),
        Linear(16, 10),
        SoftmaxMSE(gb),
    ])
    x = torch.randn(16, 20)
    y = model.forward(x, 0)
    assert y.shape == (16, 10)
    assert y.dtype == torch.float32
    torch.testing.assert_close(y.sum(-1), torch.ones(16))  # probs

    t = torch.zeros(16, 10)
    t[torch.arange(16), torch.randint(0, 10, (16,))] = 1
    d = model.backward(t, 0)
    assert d.shape == (16, 20)
    for p in model.parameters():
        if p.requires_grad:
            assert p.grad.shape == p.data.shape
    # nonzero grads after bwd, zero after zero_grad (reference :38-49)
    assert any(p.grad.abs().sum() > 0 for p in model.parameters())
    model.zero_grad()
    assert all(p.grad.abs().sum() == 0 for p in model.parameters())


def test_grad_matches_autograd_end_to_end():
    """Whole-stage backward vs torch.autograd on an identical model."""
    gb = 8
    model = Sequential([
        Linear(12, 9, activation="relu"),
        Linear(9, 7),
        SoftmaxMSE(gb),
    ])
    x = torch.randn(8, 12)
    t = torch.zeros(8, 7)
    t[torch.arange(8), torch.randint(0, 7, (8,))] = 1

    probs = model.forward(x, 0)
    model.backward(t, 0)

    # autograd twin
    w1 = model.layers[0].weight.data.clone().requires_grad_(True)
    b1 = model.layers[0].bias.data.clone().requires_grad_(True)
    w2 = model.layers[1].weight.data.clone().requires_grad_(True)
    b2 = model.layers[1].bias.data.clone().requires_grad_(True)
    h = torch.clamp(x @ w1.t() + b1, min=0)
    z = h @ w2.t() + b2
    s = torch.softmax(z, dim=-1)
    loss = ((t - s) ** 2).sum() / gb
    loss.backward()

    torch.testing.assert_close(probs, s.detach())
    torch.testing.assert_close(model.layers[0].weight.grad, w1.grad)
    torch.testing.assert_close(model.layers[0].bias.grad, b1.grad)
    torch.testing.assert_close(model.layers[1].weight.grad, w2.grad)
    torch.testing.assert_close(model.layers[1].bias.grad, b2.grad)


def test_mubatch_grad_accumulation_equals_full_batch():
    """Two half-µbatches accumulate to the full-batch gradient
    (reference gradient-accumulation semantics, layers.py:135-136)."""
    gb = 8

    def build():
        return Sequential([
            Linear(6, 5, activation="relu"),
            Linear(5, 4),
            SoftmaxMSE(gb),
        ])

    x = torch.randn(8, 6)
    t = torch.zeros(8, 4)
    t[torch.arange(8), torch.randint(0, 4, (8,))] = 1

    full = build()
    full.forward(x, 0)
    full.backward(t, 0)

    acc = build()
    for m in range(2):
        acc.forward(x[m * 4:(m + 1) * 4], m)
    for m in (1, 0):
        acc.backward(t[m * 4:(m + 1) * 4], m)

    for pf, pa in zip(full.parameters(), acc.parameters()):
        torch.testing.assert_close(pf.grad, pa.grad)


def test_mlp_stage_construction():
    """Stage slicing with one-element overlap, activation placement,
    in/out dims (reference tests/test_layers.py:52-70,
    layers.py:242-263)."""
    sizes = [784, 128, 127, 126, 125, 124, 123, 10]
    # single stage: 7 Linears + loss head
    m = MLP(sizes, 0, 1, 128)
    assert len(m.layers) == 8
    assert m.layers[-2].activation is None  # last Linear: no relu
    assert m.in_dim == 784 and m.out_dim == 10

    # 4 stages over 8 boundaries: stage_size=2
    stages = [MLP(sizes, s, 4, 128) for s in range(4)]
    nlin = [sum(1 for l in st.layers if isinstance(l, Linear)) for st in stages]
    assert nlin == [2, 2, 2, 1]  # last stage clipped (overlap slicing)
    assert stages[0].in_dim == 784
    assert stages[1].in_dim == 127  # overlap boundary
    assert stages[3].out_dim == 10
    for st in stages[:-1]:
        assert all(l.activation == "relu" for l in st.layers
                   if isinstance(l, Linear))


def test_shape_seeded_init_partition_invariant():
    """The same (in,out) shape gives bit-identical weights regardless
    of partitioning — the reference's determinism root
    (layers.py:104-112)."""
    sizes = [32, 16, 8, 6, 5, 4]
    full = MLP(sizes, 0, 1, 8)
    s0 = MLP(sizes[:], 0, 3, 8)
    s1 = MLP(sizes[:], 1, 3, 8)
    s2 = MLP(sizes[:], 2, 3, 8)
    parts = s0.parameters() + s1.parameters() + s2.parameters()
    fulls = full.parameters()
    assert len(parts) == len(fulls)
    for a, b in zip(fulls, parts):
        torch.testing.assert_close(a.data, b.data, rtol=0, atol=0)


def test_sgd_step():
    model = Sequential([Linear(4, 3)])
    opt = SGD(model.parameters(), lr=0.1)
    x = torch.randn(5, 4)
    y = model.forward(x, 0)
    model.backward(torch.ones_like(y), 0)
    before = [p.data.clone() for p in model.parameters()]
    opt.step()
    for b, p in zip(before, model.parameters()):
        torch.testing.assert_close(p.data, b - 0.1 * p.grad)


def test_eval_mode_no_stash():
    model = Sequential([Linear(4, 3, activation="relu")])
    model.eval()
    model.forward(torch.randn(2, 4), 0)
    assert len(model.layers[0]._cache) == 0


def test_standalone_mse_loss_module():
    """Softmax + MSELoss modules chained == fused SoftmaxMSE head
    (reference layers.py:83-96 + 145-166 as separate modules)."""
    from shallowspeed_amd.models import MSELoss, Softmax

    gb = 8
    x = torch.randn(8, 6)
    t = torch.zeros(8, 6)
    t[torch.arange(8), torch.randint(0, 6, (8,))] = 1

    chain = Sequential([Linear(10, 6), Softmax(), MSELoss(gb)])
    fused = Sequential([Linear(10, 6), SoftmaxMSE(gb)])
    xin = torch.randn(8, 10)
    y1 = chain.forward(xin, 0)
    y2 = fused.forward(xin, 0)
    torch.testing.assert_close(y1, y2)
    d1 = chain.backward(t, 0)
    d2 = fused.backward(t, 0)
    torch.testing.assert_close(d1, d2)
    # loss value helper (reference functional.py:38-40)
    from shallowspeed_amd.ops import functional as F

    assert F.mse_loss(y1, t, gb).item() >= 0


def test_layernorm_gelu_vs_autograd():
    """LayerNorm + GELU modules (beyond-reference extensions) against
    torch.autograd."""
    from shallowspeed_amd.models import GELU, LayerNorm

    torch.manual_seed(3)
    B, C = 16, 32
    model = Sequential([LayerNorm(C), GELU()])
    x = torch.randn(B, C)
    dy = torch.randn(B, C)
    y = model.forward(x, 0)
    dx = model.backward(dy, 0)

    xr = x.clone().requires_grad_(True)
    g = torch.ones(C, requires_grad=True)
    b = torch.zeros(C, requires_grad=True)
    yr = torch.nn.functional.gelu(
        torch.nn.functional.layer_norm(xr, (C,), g, b), approximate="tanh")
    yr.backward(dy)
    torch.testing.assert_close(y, yr.detach(), atol=2e-5, rtol=2e-5)
    torch.testing.assert_close(dx, xr.grad, atol=2e-5, rtol=2e-5)
    ln = model.layers[0]
    torch.testing.assert_close(ln.gamma.grad, g.grad, atol=2e-5, rtol=2e-5)
    torch.testing.assert_close(ln.beta.grad, b.grad, atol=2e-5, rtol=2e-5)


def test_linear_gelu_fused_vs_autograd():
    torch.manual_seed(4)
    model = Sequential([Linear(12, 8, activation="gelu")])
    x = torch.randn(6, 12)
    dy = torch.randn(6, 8)
    y = model.forward(x, 0)
    dx = model.backward(dy, 0)

    w = model.layers[0].weight.data.clone().requires_grad_(True)
    b = model.layers[0].bias.data.clone().requires_grad_(True)
    xr = x.clone().requires_grad_(True)
    yr = torch.nn.functional.gelu(xr @ w.t() + b, approximate="tanh")
    yr.backward(dy)
    torch.testing.assert_close(y, yr.detach(), atol=2e-5, rtol=2e-5)
    torch.testing.assert_close(dx, xr.grad, atol=2e-5, rtol=2e-5)
    torch.testing.assert_close(model.layers[0].weight.grad, w.grad,
                               atol=2e-5, rtol=2e-5)
    torch.testing.assert_close(model.layers[0].bias.grad, b.grad,
                               atol=2e-5, rtol=2e-5)


def test_layernorm_mlp_trains():
    """A LN+GELU MLP block stack trains end-to-end through the Worker."""
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import LayerNorm, SoftmaxXent
    from shallowspeed_amd.parallel import NaiveParallelSchedule, Topology, Worker

    gbs = 32
    model = Sequential([
        Linear(20, 32, activation="gelu"),
        LayerNorm(32),
        Linear(32, 10),
        SoftmaxXent(gbs),
    ])
    model.in_dim, model.out_dim = 20, 10
    model.materialize_device("cpu")
    opt = SGD(model.parameters(), lr=0.1)
    ds = Dataset(gbs, 8, n_samples=256, in_dim=20, n_classes=10).load(0, 1)
    w = Worker(Topology(), model, ds, opt)

    def acc():
        model.eval()
        p = model.forward(ds.x, 0)
        model.train()
        return (p.argmax(-1) == ds.y.argmax(-1)).float().mean().item()

    a0 = acc()
    for _ in range(25):
        for b in range(ds.num_batches()):
            w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)
    assert acc() > a0 + 0.15


def test_sgd_momentum_weight_decay_matches_torch():
    """Fused-SGD semantics (momentum + weight decay) vs torch.optim.SGD."""
    model = Sequential([Linear(6, 5), Linear(5, 4)])
    opt = SGD(model.parameters(), lr=0.1, momentum=0.9, weight_decay=0.01)
    tw = [p.data.clone().requires_grad_(True) for p in model.parameters()]
    topt = torch.optim.SGD(tw, lr=0.1, momentum=0.9, weight_decay=0.01)
    for step in range(3):
        for p, t in zip(model.parameters(), tw):
            g = torch.randn_like(p.data)
            p.grad.copy_(g)
            t.grad = g.clone()
        opt.step()
        topt.step()
    for p, t in zip(model.parameters(), tw):
        torch.testing.assert_close(p.data, t.detach(), atol=1e-6, rtol=1e-5)
