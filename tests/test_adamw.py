"""AdamW optimizer (beyond-reference extension): CPU path vs
torch.optim.AdamW oracle, checkpoint roundtrip, and (GPU-marked) the
fused multi-tensor HIP kernel vs the CPU path."""

import pytest
import torch

from shallowspeed_amd.models import MLP, AdamW
from shallowspeed_amd.parallel import Topology

SIZES = [24, 16, 12, 10]


def _grads_like(model, seed):
    g = torch.Generator().manual_seed(seed)
    return [torch.randn(p.data.shape, generator=g) for p in model.parameters()
            if p.requires_grad]


def test_adamw_cpu_matches_torch_oracle():
    model = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    opt = AdamW(model.parameters(), lr=0.01, weight_decay=0.02)

    ref_params = [p.data.clone().requires_grad_(True)
                  for p in model.parameters()]
    ref_opt = torch.optim.AdamW(ref_params, lr=0.01, betas=(0.9, 0.999),
                                eps=1e-8, weight_decay=0.02)
    for step in range(5):
        grads = _grads_like(model, seed=step)
        for p, g in zip(model.parameters(), grads):
            p.grad.copy_(g)
        opt.step()
        for rp, g in zip(ref_params, grads):
            rp.grad = g.clone()
        ref_opt.step()
    for p, rp in zip(model.parameters(), ref_params):
        torch.testing.assert_close(p.data, rp.detach(), rtol=1e-5, atol=1e-6)


def test_adamw_checkpoint_roundtrip(tmp_path):
    from shallowspeed_amd.checkpoint import load_checkpoint, save_checkpoint

    model = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    opt = AdamW(model.parameters(), lr=0.01)
    for step in range(3):
        for p, g in zip(model.parameters(), _grads_like(model, step)):
            p.grad.copy_(g)
        opt.step()
    save_checkpoint(tmp_path, model, Topology(), step=3, optimizer=opt)

    model2 = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    opt2 = AdamW(model2.parameters(), lr=0.01)
    load_checkpoint(tmp_path, model2, Topology(), optimizer=opt2)
    assert opt2.step_count == 3
    for a, b in zip(opt._m, opt2._m):
        torch.testing.assert_close(a, b, rtol=0, atol=0)
    # continue both one step: identical trajectory
    for o, m in ((opt, model), (opt2, model2)):
        for p, g in zip(m.parameters(), _grads_like(m, 99)):
            p.grad.copy_(g)
        o.step()
    for a, b in zip(model.parameters(), model2.parameters()):
        torch.testing.assert_close(a.data, b.data, rtol=0, atol=0)


@pytest.mark.gpu
def test_adamw_fused_kernel_matches_cpu(gpu_device):
    cpu_model = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    gpu_model = MLP(SIZES, 0, 1, 16).materialize_device(gpu_device)
    cpu_opt = AdamW(cpu_model.parameters(), lr=0.02, weight_decay=0.01)
    gpu_opt = AdamW(gpu_model.parameters(), lr=0.02, weight_decay=0.01)
    for step in range(4):
        grads = _grads_like(cpu_model, seed=step)
        for p, g in zip(cpu_model.parameters(), grads):
            p.grad.copy_(g)
        for p, g in zip(gpu_model.parameters(), grads):
            p.grad.copy_(g.to(gpu_device))
        cpu_opt.step()
        gpu_opt.step()
    torch.cuda.synchronize()
    for cp, gp in zip(cpu_model.parameters(), gpu_model.parameters()):
        torch.testing.assert_close(gp.data.cpu(), cp.data,
                                   rtol=1e-5, atol=1e-6)
        # bf16 compute copy re-emitted in-kernel must equal bf16 of the
        # GPU's OWN master bitwise (comparing against the CPU master's
        # bf16 would demand bitwise f32 agreement across sqrtf vs sqrt)
        torch.testing.assert_close(gp.lp.float().cpu(),
                                   gp.data.cpu().to(torch.bfloat16).float(),
                                   rtol=0, atol=0)
        if gp.lp_t is not None:
            torch.testing.assert_close(gp.lp_t.cpu(), gp.lp.t().cpu(),
                                       rtol=0, atol=0)


def test_grad_clip_global_norm():
    from shallowspeed_amd.models import SGD

    model = MLP(SIZES, 0, 1, 16).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=0.0, clip_norm=1.0,
              flat_grad=model._flat_grad)
    for p, g in zip(model.parameters(), _grads_like(model, 1)):
        p.grad.copy_(g * 10)  # force norm >> 1
    pre = model._flat_grad.norm().item()
    assert pre > 1.0
    opt.step()
    post = model._flat_grad.norm().item()
    assert abs(post - 1.0) < 1e-4, post
    # below-threshold grads are untouched
    model._flat_grad.mul_(0.3)
    want = model._flat_grad.clone()
    opt.step()
    torch.testing.assert_close(model._flat_grad, want, rtol=1e-6, atol=0)


@pytest.mark.gpu
def test_transpose_bf16_tiled_kernel(gpu_device):
    from shallowspeed_amd.ops import load_ext

    e = load_ext(required=True)
    g = torch.Generator(device="cuda").manual_seed(3)
    x = torch.randn(2048, 1024, generator=g, device=gpu_device).bfloat16()
    dst = torch.empty(1024, 2048, dtype=torch.bfloat16, device=gpu_device)
    e.transpose_bf16(x, dst)
    torch.cuda.synchronize()
    assert torch.equal(dst, x.t().contiguous())


@pytest.mark.gpu
def test_sgd_tiled_transpose_path(gpu_device):
    """Wide weights (cols >= 1024) take the tiled-transpose emission;
    lp_t must still equal lp.t() bitwise after a fused step."""
    from shallowspeed_amd.models import SGD

    model = MLP([1024, 1024, 10], 0, 1, 16).materialize_device(gpu_device)
    opt = SGD(model.parameters(), lr=0.01)
    for p, gr in zip(model.parameters(), _grads_like(model, 5)):
        p.grad.copy_(gr.to(gpu_device))
    opt.step()
    torch.cuda.synchronize()
    assert any(id(p) in {id(q) for q, _, _ in opt._tiled_t}
               for p in model.parameters()), "tiled path not engaged"
    for p in model.parameters():
        if p.lp_t is not None:
            assert torch.equal(p.lp_t, p.lp.t().contiguous())
        torch.testing.assert_close(p.lp.float(),
                                   p.data.to(torch.bfloat16).float(),
                                   rtol=0, atol=0)
