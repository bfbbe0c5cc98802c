"""Simulate PP=2 in-process: two Workers, send/recv replaced by queues."""
import torch

from shallowspeed_amd.data import Dataset
from shallowspeed_amd.models import MLP, SGD
from shallowspeed_amd.parallel import NaiveParallelSchedule, Topology, Worker

SIZES = [24, 16, 12, 8, 6, 10]
GBS, MUB, N, LR = 32, 4, 64, 0.05


def serial():
    model = MLP(SIZES, 0, 1, GBS).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0], n_classes=SIZES[-1]).load(0, 1)
    w = Worker(Topology(), model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)
    return model


def pp2():
    P = 2
    workers, models = [], []
    chan = {}  # (src,dst) -> list of tensors

    for s in range(P):
        topo = Topology(rank=s, world=P, dp=1, pp=P)
        model = MLP(SIZES, s, P, GBS).materialize_device("cpu")
        opt = SGD(model.parameters(), lr=LR)
        ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=SIZES[0], n_classes=SIZES[-1]).load(0, 1)
        w = Worker(topo, model, ds, opt)
        # monkeypatch p2p
        def mk_send(me):
            def _send(buf_t, dst):
                chan.setdefault((me, dst), []).append(buf_t.clone())
            return _send
        def mk_recv(me):
            def _recv(buf_t, src):
                buf_t.copy_(chan[(src, me)].pop(0))
            return _recv
        w._send = mk_send(s)
        w._recv = mk_recv(s)
        w._send_activations = lambda cmd, w=w: w._send(w._out_bufs[cmd.buffer_idx], w.topo.next_rank)
        w._recv_activations = lambda cmd, w=w: w._recv(w._in_bufs[cmd.buffer_idx], w.topo.prev_rank)
        w._recv_output_grad = lambda cmd, w=w: w._recv(w._out_bufs[cmd.buffer_idx], w.topo.next_rank)
        w._send_input_grad = lambda cmd, w=w: w._send(w._in_bufs[cmd.buffer_idx], w.topo.prev_rank)
        w._DISPATCH = dict(w._DISPATCH)
        from shallowspeed_amd.parallel.instructions import (
            RecvActivations, SendActivations, RecvOutputGrad, SendInputGrad)
        w._DISPATCH[RecvActivations] = w._recv_activations
        w._DISPATCH[SendActivations] = w._send_activations
        w._DISPATCH[RecvOutputGrad] = w._recv_output_grad
        w._DISPATCH[SendInputGrad] = w._send_input_grad
        workers.append(w); models.append(model)

    nb = workers[0].dataset.num_batches()
    for b in range(nb):
        # interleave: run stage0's steps then stage1's (queues buffer)
        progs = []
        for s in range(P):
            sched = NaiveParallelSchedule(MUB, P, s)
            w = workers[s]
            w._batch_id = b
            w._ensure_buffers(sched.num_buffers, w.dataset.mubatch_size)
            progs.append([c for st in sched.steps() for c in st])
        pcs = [0] * P
        from shallowspeed_amd.parallel.instructions import RecvActivations, RecvOutputGrad
        done = 0
        total = sum(len(p) for p in progs)
        while done < total:
            prog = False
            for s in range(P):
                while pcs[s] < len(progs[s]):
                    c = progs[s][pcs[s]]
                    if isinstance(c, RecvActivations) and not chan.get((s - 1, s)):
                        break
                    if isinstance(c, RecvOutputGrad) and not chan.get((s + 1, s)):
                        break
                    workers[s]._DISPATCH[type(c)](c)
                    pcs[s] += 1; done += 1; prog = True
            assert prog, "deadlock"
    return models


ms = serial()
stages = pp2()
got = stages[0].parameters() + stages[1].parameters()
want = ms.parameters()
for i, (g, w) in enumerate(zip(got, want)):
    d = (g.data - w.data).abs().max().item()
    print(i, g.data.shape, "maxdiff", d)
