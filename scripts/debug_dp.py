"""Simulate DP=2 in one process (no comm): manually sum grads."""
import torch

from shallowspeed_amd.data import Dataset
from shallowspeed_amd.models import MLP, SGD
from shallowspeed_amd.parallel import GPipeSchedule, NaiveParallelSchedule, Topology, Worker

SIZES = [24, 16, 12, 8, 6, 10]
GBS, MUB, N, LR = 32, 4, 64, 0.05


def serial(sched_cls):
    model = MLP(SIZES, 0, 1, GBS).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0], n_classes=SIZES[-1]).load(0, 1)
    w = Worker(Topology(), model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(sched_cls(ds.num_mubatches(), 1, 0), b)
    return model


def dp2(sched_cls):
    models, workers, dss = [], [], []
    for r in range(2):
        model = MLP(SIZES, 0, 1, GBS).materialize_device("cpu")
        opt = SGD(model.parameters(), lr=LR)
        ds = Dataset(GBS, (GBS // 2) // MUB, n_samples=N, in_dim=SIZES[0], n_classes=SIZES[-1]).load(r, 2)
        models.append(model); workers.append(Worker(Topology(), model, ds, opt)); dss.append(ds)
    nb = dss[0].num_batches()
    for b in range(nb):
        for r in range(2):
            # run all steps except optimizer; emulate allreduce before step
            sched = sched_cls(dss[r].num_mubatches(), 1, 0)
            cmds = [c for step in sched.steps() for c in step]
            w = workers[r]
            w._batch_id = b
            w._ensure_buffers(sched.num_buffers, dss[r].mubatch_size)
            for c in cmds[:-1]:  # skip OptimizerStep
                w._DISPATCH[type(c)](c)
        # allreduce
        flat0, flat1 = models[0]._flat_grad, models[1]._flat_grad
        s = flat0 + flat1
        flat0.copy_(s); flat1.copy_(s)
        for r in range(2):
            workers[r].optimizer.step()
    return models[0]


ms = serial(NaiveParallelSchedule)
for name, cls in [("naive", NaiveParallelSchedule), ("gpipe", GPipeSchedule)]:
    md = dp2(cls)
    diffs = [(a.data - b.data).abs().max().item() for a, b in zip(ms.parameters(), md.parameters())]
    print(name, "max diff vs serial-naive:", max(diffs))
