import os
import random
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.multiprocessing as mp

SIZES = [24, 16, 12, 8, 6, 10]
GBS, MUB, N, LR = 32, 4, 64, 0.05


def child(rank, world, port, out):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import SCHEDULES, Worker, init_topology

    topo = init_topology(dp=1, pp=world, backend="gloo",
                         device=torch.device("cpu"))
    model = MLP(SIZES, topo.stage_id, world, GBS).materialize_device("cpu")
    init = [p.data.clone() for p in model.parameters()]
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1]).load(0, 1)
    w = Worker(topo, model, ds, opt)
    dumps = []
    for b in range(ds.num_batches()):
        sched = SCHEDULES["naive"](ds.num_mubatches(), world, topo.stage_id)
        w.execute(sched, b)
        dumps.append({"grads": [p.grad.clone() for p in model.parameters()],
                      "w": [p.data.clone() for p in model.parameters()]})
    torch.save({"dumps": dumps, "init": init},
               os.path.join(out, f"dump{topo.stage_id}.pt"))
    torch.distributed.destroy_process_group()


def main():
    tmp = tempfile.mkdtemp()
    mp.spawn(child, args=(2, random.randint(20000, 45000), tmp),
             nprocs=2, join=True)

    from shallowspeed_amd.data import Dataset
    from shallowspeed_amd.models import MLP, SGD
    from shallowspeed_amd.parallel import NaiveParallelSchedule, Topology, Worker

    model = MLP(SIZES, 0, 1, GBS).materialize_device("cpu")
    init = [p.data.clone() for p in model.parameters()]
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1]).load(0, 1)
    w = Worker(Topology(), model, ds, opt)
    sdumps = []
    for b in range(ds.num_batches()):
        sched = NaiveParallelSchedule(ds.num_mubatches(), 1, 0)
        w.execute(sched, b)
        sdumps.append({"grads": [p.grad.clone() for p in model.parameters()],
                       "w": [p.data.clone() for p in model.parameters()]})

    d0 = torch.load(os.path.join(tmp, "dump0.pt"), weights_only=False)
    d1 = torch.load(os.path.join(tmp, "dump1.pt"), weights_only=False)
    for b in range(len(sdumps)):
        gg = d0["dumps"][b]["grads"] + d1["dumps"][b]["grads"]
        ww = d0["dumps"][b]["w"] + d1["dumps"][b]["w"]
        for i, (a, s) in enumerate(zip(gg, sdumps[b]["grads"])):
            d = (a - s).abs().max().item()
            if d: print("batch", b, "grad", i, d)
        for i, (a, s) in enumerate(zip(ww, sdumps[b]["w"])):
            d = (a - s).abs().max().item()
            if d: print("batch", b, "weight", i, d)
    print("done")


if __name__ == "__main__":
    main()
