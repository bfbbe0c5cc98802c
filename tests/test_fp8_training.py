"""MX-fp8 training forward: numerics vs bf16 and convergence parity.

Round-1 shipped the fp8 serving tier (csrc/fp8.hip: scaled-MFMA GEMM,
hardware-probed scale layout); this extends it to the TRAINING forward
(f32 masters, bf16 backward, fp8 forward GEMMs on qualifying layers)
and gates on convergence parity with bf16 on the synthetic teacher
task — the fp8-training acceptance criterion."""

import pytest
import torch

pytestmark = pytest.mark.gpu

SIZES = [256, 512, 512, 10]
GBS = 512


def _fresh_model(dev):
    from shallowspeed_amd.models import MLP

    return MLP(SIZES, 0, 1, GBS, loss="xent").materialize_device(dev)


def test_fp8_fwd_close_to_bf16(gpu_device):
    model = _fresh_model(gpu_device)
    model.eval()  # no stash
    g = torch.Generator().manual_seed(7)
    x = torch.randn(GBS, SIZES[0], generator=g).bfloat16().to(gpu_device)
    y_bf = model.forward(x).float()
    n = model.set_fp8_fwd(True)
    assert n == 2, f"expected 2 fp8-eligible layers, got {n}"
    y_f8 = model.forward(x).float()
    torch.cuda.synchronize()
    # probs out of the softmax head: small absolute deviation expected
    # from e4m3 block quantization of two hidden layers
    assert (y_f8 - y_bf).abs().max().item() < 0.08
    assert torch.isfinite(y_f8).all()


def _train_acc(dev, fp8: bool, epochs=25):
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import SGD
    from shallowspeed_amd.ops.functional import row_argmax
    from shallowspeed_amd.parallel import NaiveParallelSchedule, Topology, Worker

    model = _fresh_model(dev)
    if fp8:
        assert model.set_fp8_fwd(True) == 2
    opt = SGD(model.parameters(), lr=0.05, momentum=0.9)
    ds = Dataset(GBS, GBS, n_samples=4096, in_dim=SIZES[0],
                 n_classes=SIZES[-1], device=dev).load(0, 1)
    val = Dataset(GBS, GBS, validation=True, n_samples=1024,
                  in_dim=SIZES[0], n_classes=SIZES[-1], device=dev).load(0, 1)
    w = Worker(Topology(device=dev), model, ds, opt)
    sched = NaiveParallelSchedule(1, 1, 0)
    for _ in range(epochs):
        for b in range(ds.num_batches()):
            w.execute(sched, b)
    model.eval()
    correct = total = 0
    for b in range(val.num_batches()):
        probs = model.forward(val.micro_batch_input(b, 0))
        pred = row_argmax(probs)
        lab = row_argmax(val.micro_batch_target(b, 0))
        correct += (pred == lab).sum().item()
        total += pred.numel()
    torch.cuda.synchronize()
    return correct / total


def test_fp8_training_convergence_parity(gpu_device):
    acc_bf = _train_acc(gpu_device, fp8=False)
    acc_f8 = _train_acc(gpu_device, fp8=True)
    assert acc_bf > 0.5, f"teacher task not learned in bf16: {acc_bf}"
    assert acc_f8 >= acc_bf - 0.05, (acc_bf, acc_f8)
