"""Model-level fuzzing (hypothesis): random layer stacks and shapes,
forward+backward against a torch.autograd twin (CPU f32 oracle)."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from shallowspeed_amd.models import GELU, LayerNorm, Linear, ReLU, Sequential


@st.composite
def stack_spec(draw):
    n_layers = draw(st.integers(1, 4))
    dims = [draw(st.integers(2, 24)) for _ in range(n_layers + 1)]
    acts = [draw(st.sampled_from([None, "relu", "gelu"]))
            for _ in range(n_layers)]
    extras = [draw(st.sampled_from(["none", "ln", "gelu_mod", "relu_mod"]))
              for _ in range(n_layers)]
    batch = draw(st.integers(1, 9))
    return dims, acts, extras, batch


def build_pair(dims, acts, extras):
    """Our stack + a torch twin sharing the same parameters."""
    ours, twin_params = [], []
    for i, act in enumerate(acts):
        lin = Linear(dims[i], dims[i + 1], activation=act)
        ours.append(lin)
        twin_params.append(("linear", lin, act))
        if extras[i] == "ln":
            ln = LayerNorm(dims[i + 1])
            ours.append(ln)
            twin_params.append(("ln", ln, None))
        elif extras[i] == "gelu_mod":
            ours.append(GELU())
            twin_params.append(("gelu", None, None))
        elif extras[i] == "relu_mod":
            ours.append(ReLU())
            twin_params.append(("relu", None, None))
    return Sequential(ours), twin_params


def twin_forward(twin_params, x, grads_into):
    h = x
    for kind, mod, act in twin_params:
        if kind == "linear":
            w = mod.weight.data.clone().requires_grad_(True)
            b = mod.bias.data.clone().requires_grad_(True)
            grads_into.append((mod.weight, w))
            grads_into.append((mod.bias, b))
            h = h @ w.t() + b
            if act == "relu":
                # torch.relu's subgradient at exactly 0 is 0, matching
                # the strict >0 mask convention (reference layers.py:70;
                # clamp(min=0) passes gradient at the boundary)
                h = torch.relu(h)
            elif act == "gelu":
                h = torch.nn.functional.gelu(h, approximate="tanh")
        elif kind == "ln":
            g = mod.gamma.data.clone().requires_grad_(True)
            be = mod.beta.data.clone().requires_grad_(True)
            grads_into.append((mod.gamma, g))
            grads_into.append((mod.beta, be))
            h = torch.nn.functional.layer_norm(h, (h.shape[-1],), g, be,
                                               mod.eps)
        elif kind == "gelu":
            h = torch.nn.functional.gelu(h, approximate="tanh")
        elif kind == "relu":
            h = torch.relu(h)
    return h


@given(spec=stack_spec(), seed=st.integers(0, 10_000))
@settings(max_examples=60, deadline=None)
def test_random_stack_matches_autograd(spec, seed):
    dims, acts, extras, batch = spec
    torch.manual_seed(seed)
    model, twin = build_pair(dims, acts, extras)
    x = torch.randn(batch, dims[0])
    dy = torch.randn(batch, dims[-1])

    y = model.forward(x.clone(), 0)
    dx = model.backward(dy, 0)

    grads_into = []
    xr = x.clone().requires_grad_(True)
    yr = twin_forward(twin, xr, grads_into)
    yr.backward(dy)

    torch.testing.assert_close(y, yr.detach(), atol=2e-4, rtol=2e-4)
    torch.testing.assert_close(dx, xr.grad, atol=2e-4, rtol=2e-4)
    for p, t in grads_into:
        torch.testing.assert_close(p.grad, t.grad, atol=2e-4, rtol=2e-4)
