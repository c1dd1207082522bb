#!/usr/bin/env python3
"""Benchmark harness — AlexNet training images/sec on MI355X (BASELINE).

    python bench.py --gpus N --steps K --warmup W [--model alexnet]

Under torchrun (one rank per GPU over RCCL) reads RANK/LOCAL_RANK/
WORLD_SIZE from the env.  Rank 0 prints ONE JSON line with the whole-job
aggregate throughput.  Synthetic data (prototype+noise, random-init
weights), bf16 compute / fp32 accumulate.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int,
                   default=int(os.environ.get("WORLD_SIZE", 1)))
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--model", default="alexnet",
                   choices=["alexnet", "cifar", "vgg16"])
    p.add_argument("--batch", type=int, default=0,
                   help="per-GPU minibatch (0 = model default)")
    p.add_argument("--unit-times", action="store_true",
                   help="print the per-unit time table to stderr")
    p.add_argument("--target-loss", type=float, default=0.0,
                   help="also run time-to-loss: train until the epoch "
                        "train error %% drops below this, report secs")
    p.add_argument("--kill-rank-after", type=int, default=0,
                   help="fault injection: rank (world-1) aborts after "
                        "N steps — exercises the emergency-snapshot / "
                        "failure-detection path")
    return p.parse_args()


def build_workflow(args, rank, world, device):
    from veles.config import root
    from veles.dummy import DummyLauncher
    if args.model == "alexnet":
        from veles.znicz_amd.models.alexnet import AlexNetWorkflow as WF
        cfg = root.alexnet
        batch = args.batch or 256
        cfg.loader.update({"minibatch_size": batch,
                           "n_train": batch * 4, "n_valid": 0,
                           "seed": 0xC0FFEE + rank})
    elif args.model == "cifar":
        from veles.znicz_amd.models.cifar import CifarWorkflow as WF
        cfg = root.cifar
        batch = args.batch or 256
        cfg.loader.update({"minibatch_size": batch,
                           "n_train": batch * 8, "n_valid": 0,
                           "on_device": True, "device_dtype": "bf16",
                           "seed": 0xC0FFEE + rank})
    else:
        from veles.znicz_amd.models.vgg import VGG16Workflow as WF
        cfg = root.vgg16
        batch = args.batch or 64
        cfg.loader.update({"minibatch_size": batch,
                           "n_train": batch * 2, "n_valid": 0,
                           "seed": 0xC0FFEE + rank})
    cfg.decision.update({"max_epochs": 10 ** 9,
                         "fail_iterations": 10 ** 9})
    wf = WF(DummyLauncher())
    return wf, batch


def main():
    args = parse_args()
    import torch
    if not torch.cuda.is_available():
        print(json.dumps({"error": "no GPU visible — bench.py measures "
                                   "the HIP path on MI355X"}))
        sys.exit(1)

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    # ranks % device-count: the gloo test lane runs 2 ranks on 1 GPU
    local_rank = local_rank % max(1, torch.cuda.device_count())
    torch.cuda.set_device(local_rank)

    from veles import prng
    prng.seed_all(1234 + rank)

    from veles.backends import ROCmDevice
    device = ROCmDevice(local_rank)

    dist_ctx = None
    if world > 1:
        from veles.znicz_amd.parallel import DistributedContext
        dist_ctx = DistributedContext.create(
            backend=os.environ.get("VELES_DIST_BACKEND", "nccl"))

    wf, batch = build_workflow(args, rank, world, device)
    if dist_ctx is not None:
        wf.dist = dist_ctx
    wf.initialize(device=device)

    # the reducer is ALWAYS attached: at world>1 it all-reduces gradient
    # buckets over RCCL; at world=1 (_NullCtx) it still runs the
    # comm-stream fused SGD so the update overlaps the rest of backward
    from veles.znicz_amd.parallel import GradientReducer
    wf.grad_reducer = GradientReducer(dist_ctx or _NullCtx(world), device)

    loader, fwds, ev, dec = wf.loader, wf.forwards, wf.evaluator, \
        wf.decision
    gds_rev = [g for g in reversed(wf.gds)]
    comm_sync = wf.comm_sync

    # hipGraph replay of the device step (veles/znicz_amd/hipgraph.py):
    # default on for single-rank runs; multi-rank needs RCCL's graph
    # capture, so it is opt-in there (VELES_HIPGRAPH=1).  VELES_HIPGRAPH=0
    # forces eager.  Capture failure falls back to eager automatically.
    graph_env = os.environ.get("VELES_HIPGRAPH", "auto")
    use_graph = graph_env != "0" and (
        world == 1 or graph_env == "1")
    if use_graph:
        from veles.znicz_amd.hipgraph import GraphStepper
        stepper = GraphStepper(wf)

        def step():
            stepper.step()
    else:
        def step():
            loader.run()
            for f in fwds:
                f.run()
            ev.run()
            dec.run()
            for g in gds_rev:
                g.run()
            comm_sync.run()

    def barrier_sync():
        if dist_ctx is not None:
            dist_ctx.barrier()
        torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    barrier_sync()
    t0 = time.perf_counter()
    for it in range(args.steps):
        if args.kill_rank_after and rank == world - 1 and \
                it == args.kill_rank_after:
            # simulated rank death: the launcher path
            # catches the collective failure and snapshots
            print(json.dumps({"fault_injected": True, "rank": rank,
                              "at_step": it}), flush=True)
            os._exit(13)
        step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist_ctx is not None:
        t = torch.tensor([elapsed], dtype=torch.float64, device="cuda")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    if use_graph and rank == 0:
        print("graph: captured=%s disabled=%s%s" % (
            stepper._graph is not None, stepper._disabled,
            "\n" + stepper.capture_error
            if getattr(stepper, "capture_error", None) else ""),
            file=sys.stderr)

    if args.unit_times and rank == 0:
        wf.print_stats()

    if rank == 0:
        images = batch * world * args.steps
        value = images / elapsed
        sample = loader.minibatch_data.shape
        print(json.dumps({
            "metric": "images/sec (whole node) %s training" % args.model,
            "value": round(value, 1),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {"model": args.model,
                       "global_batch": batch * world,
                       "input": list(sample[1:]),
                       # RGB padded to 8 channels for the NHWC conv DMA
                       # grain — the kernels do MORE work than the
                       # 3-channel config requires (a C=4 materialized-
                       # im2col path exists and is GPU-tested, but
                       # measured net-slower: profiles/perf_r2.md)
                       "channels_padded_from": 3,
                       "parallelism": "dp%d" % world},
        }))
    if args.target_loss > 0:
        tl0 = time.perf_counter()
        from veles.loader.base import TRAIN
        steps_done = 0
        while (dec.epoch_n_err_pt[TRAIN] is None or
               dec.epoch_n_err_pt[TRAIN] > args.target_loss) and \
                steps_done < 100000 and not bool(dec.complete):
            step()
            steps_done += 1
        barrier_sync()
        if rank == 0:
            print(json.dumps({
                "metric": "time-to-loss %s (train err%% <= %g)" % (
                    args.model, args.target_loss),
                "value": round(time.perf_counter() - tl0, 3),
                "unit": "seconds", "n_gpus": world,
                "higher_is_better": False,
                "final_err_pt": dec.epoch_n_err_pt[TRAIN],
                "steps": steps_done}))
    if dist_ctx is not None:
        dist_ctx.finalize()


class _NullCtx(object):
    def __init__(self, world=1):
        self.world = 1
        self.rank = 0

    def all_reduce_(self, t, async_op=False):
        return None

    def all_reduce_scalar(self, v):
        return v

    def barrier(self):
        pass

    def finalize(self):
        pass


if __name__ == "__main__":
    main()
