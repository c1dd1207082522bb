"""Data-parallel plane tests on the gloo backend (CPU, world_size 2).

The reference's distributed tests ran master+slave in one process over
loopback ZeroMQ (veles/tests client/server tests); here the equivalent
is two spawned ranks over torch.distributed gloo on 127.0.0.1 — the
same DistributedContext / coalesced_all_reduce / loader-shard /
Decision-metric path the RCCL production path uses.
"""

import os
import pickle
import socket
import tempfile

import numpy
import pytest

torch = pytest.importorskip("torch")
import torch.multiprocessing as mp  # noqa: E402


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, world, port, fn_name, out_dir):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from veles.znicz_amd.parallel import DistributedContext
    ctx = DistributedContext.create(backend="gloo")
    result = globals()[fn_name](ctx)
    with open(os.path.join(out_dir, "rank%d.pkl" % rank), "wb") as f:
        pickle.dump(result, f)
    ctx.finalize()


def _spawn(fn_name, world=2):
    port = _free_port()
    out_dir = tempfile.mkdtemp()
    mp.spawn(_worker, args=(world, port, fn_name, out_dir), nprocs=world,
             join=True)
    out = []
    for r in range(world):
        with open(os.path.join(out_dir, "rank%d.pkl" % r), "rb") as f:
            out.append(pickle.load(f))
    return out


# -- per-rank bodies (run inside spawned processes) -------------------------

def body_collectives(ctx):
    s = ctx.all_reduce_scalar(ctx.rank + 1)            # 1 + 2
    t1 = torch.full((5,), float(ctx.rank + 1))
    t2 = torch.full((3,), float(10 * (ctx.rank + 1)))
    from veles.znicz_amd.parallel import coalesced_all_reduce
    coalesced_all_reduce(ctx, [t1, t2])
    return {"scalar": s, "t1": t1.numpy().tolist(),
            "t2": t2.numpy().tolist()}


def body_sharded_training(ctx):
    """MNIST MLP, numpy backend, sharded loader; Decision metrics must be
    the (identical) global counts on every rank."""
    from veles import prng
    from veles.backends import NumpyDevice
    from veles.config import root
    from veles.dummy import DummyLauncher
    from veles.loader.base import TRAIN
    prng.seed_all(1234)
    from veles.znicz_amd.models.mnist import MnistWorkflow
    # (config updates must follow the model import — the module sets its
    # defaults into root.mnist at import time)
    root.mnist.decision.update({"max_epochs": 2})
    root.mnist.loader.update({"n_train": 120, "n_valid": 40,
                              "minibatch_size": 40, "seed": 777})
    wf = MnistWorkflow(DummyLauncher())
    wf.dist = ctx
    wf.loader.shard(ctx.rank, ctx.world)
    wf.initialize(device=NumpyDevice())
    wf.run()
    return {"train_err_pt": wf.decision.epoch_n_err_pt[TRAIN],
            "epoch": wf.loader.epoch_number,
            "n_served": int(wf.loader.minibatch_size)}


def body_shard_disjoint(ctx):
    from veles import prng
    from veles.config import root
    from veles.dummy import DummyWorkflow
    from veles.backends import NumpyDevice
    prng.seed_all(1)
    from veles.znicz_amd.models.synthetic import SyntheticClassLoader
    ldr = SyntheticClassLoader(
        DummyWorkflow(), minibatch_size=20, n_train=40, n_valid=0,
        sample_shape=(4, 4, 8), n_classes=4, seed=99)
    ldr.shard(ctx.rank, ctx.world)
    ldr.initialize(device=NumpyDevice())
    served = []
    for _ in range(2):  # one epoch = 2 global minibatches
        ldr.run()
        served.extend(ldr.minibatch_indices.mem[
            :ldr.minibatch_size].tolist())
    return {"indices": served}


# -- tests ------------------------------------------------------------------

@pytest.mark.timeout(300)
def test_gloo_collectives():
    res = _spawn("body_collectives")
    for r in res:
        assert r["scalar"] == 3
        assert r["t1"] == [3.0] * 5
        assert r["t2"] == [30.0] * 3


@pytest.mark.timeout(300)
def test_sharded_loader_disjoint_and_aligned():
    res = _spawn("body_shard_disjoint")
    a, b = set(res[0]["indices"]), set(res[1]["indices"])
    assert a and b
    assert not (a & b), "ranks served overlapping samples"


@pytest.mark.timeout(600)
def test_sharded_training_metrics_agree():
    res = _spawn("body_sharded_training")
    # Decision all-reduces n_err: both ranks must report the SAME global
    # error percentage, and each rank serves half the global batches.
    assert res[0]["train_err_pt"] == pytest.approx(
        res[1]["train_err_pt"], abs=1e-9)
    assert res[0]["epoch"] == res[1]["epoch"] == 2
    assert res[0]["n_served"] == res[1]["n_served"]
