"""Flagship benchmark — the driver contract.

Measures the BASELINE.json headline metric: epoch time + throughput
(edges/s) for Reddit-shaped GraphSAGE 3-layer h=256, sampling-rate=0.1,
use_pp, on N partitions = N GPUs (one process per GPU over RCCL/xGMI).

  python bench.py --gpus N --steps K --warmup W
  (N>1 is launched by the driver via torch.distributed.run, one rank/GPU)

A "step" is one full training epoch over the partitioned graph (forward,
loss, backward, gradient all-reduce, Adam step) — identical work to the
reference's epoch loop (reference train.py:385-425). Data is synthetic of
the named shape (232,965 nodes / ~114.8M directed edges incl self-loops,
602 features, 41 classes — graph/synthetic.py) with random-init weights;
compute dtype fp32 = the reference's dtype.

Scaling is STRONG: the same fixed graph is partitioned across N GPUs.

value       = full-graph edges / epoch-time  (whole-job aggregate)
vs_baseline = value / (E_ref / 0.3578s), the reference README's measured
              epoch time (note: the README quotes a 4-layer,
              2-GPU, NVIDIA run of the same dataset+rate — the closest
              published number; BASELINE.json names the 3-layer/8-part
              config measured here).
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--dataset", type=str, default="reddit")
    p.add_argument("--model", type=str, default="graphsage")
    p.add_argument("--n-layers", type=int, default=3)
    p.add_argument("--n-hidden", type=int, default=256)
    p.add_argument("--heads", type=int, default=4)
    p.add_argument("--n-linear", type=int, default=0)
    p.add_argument("--sampling-rate", type=float, default=0.1)
    p.add_argument("--partition-method", type=str, default="metis")
    p.add_argument("--data-scale", type=float, default=1.0)
    p.add_argument("--partition-dir", type=str, default="bench_partition")
    p.add_argument("--device", type=str, default="auto")
    p.add_argument("--backend", type=str, default="auto",
                   choices=["auto", "nccl", "gloo"])
    p.add_argument("--use-pp", action=argparse.BooleanOptionalAction,
                   default=True)
    p.add_argument("--dropout", type=float, default=0.5)
    p.add_argument("--halo-dtype", choices=["fp32", "bf16"], default="fp32")
    return p.parse_args()


def main():
    a = parse()
    if a.dataset == "ogbn-papers100M" and a.data_scale >= 0.5:
        # 111M-node tensors leave little slack for allocator fragmentation
        os.environ.setdefault("PYTORCH_ALLOC_CONF", "expandable_segments:True")
    from bnsgcn_amd.runtime.config import create_parser, graph_name_of
    from bnsgcn_amd.runtime.trainer import prepare_partitions, RankState, _forward
    from bnsgcn_amd.graph import load_partition, load_meta
    from bnsgcn_amd.models.models import create_model
    from bnsgcn_amd.parallel import GradReducer, init_distributed

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", a.gpus))
    assert world == a.gpus, f"WORLD_SIZE {world} != --gpus {a.gpus}"

    args = create_parser().parse_args([])
    args.dataset = a.dataset
    args.model = a.model
    args.n_layers = a.n_layers
    args.n_hidden = a.n_hidden
    args.heads = a.heads
    args.n_linear = a.n_linear
    args.sampling_rate = a.sampling_rate
    args.n_partitions = world
    args.partition_method = a.partition_method
    args.data_scale = a.data_scale
    args.partition_dir = a.partition_dir
    args.use_pp = a.use_pp
    args.dropout = a.dropout
    args.halo_dtype = a.halo_dtype
    args.eval = False
    args.fix_seed = True
    args.seed = 0
    args.device = a.device
    args.graph_name = graph_name_of(args)

    cuda = torch.cuda.is_available() and a.device != "cpu"
    # set the device BEFORE the first collective: an NCCL barrier issued
    # while every rank still sits on cuda:0 builds duplicate-device
    # communicators and aborts the whole 8-GPU job
    if a.device == "auto":
        device = f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}" if cuda else "cpu"
    else:
        device = a.device
    if cuda:
        torch.cuda.set_device(torch.device(device))
    if world > 1:
        backend = a.backend if a.backend != "auto" else \
            ("nccl" if cuda else "gloo")
        rank, world = init_distributed(backend)
    if rank == 0:
        args.skip_partition = True   # reuse an existing store for this config
        prepare_partitions(args)
    if world > 1:
        dist.barrier()

    part = load_partition(args.partition_dir, args.graph_name, rank)
    meta = part.meta
    torch.manual_seed(args.seed)
    state = RankState(part, args, device)
    state.plan.set_epoch(0)
    model = create_model(args, n_feat=int(meta["n_feat"]),
                         n_class=int(meta["n_class"]),
                         train_size=int(meta["n_train"])).to(device)
    if a.use_pp or a.model == "gat":
        state.precompute()
        state.raw_feat = None   # bench never evaluates; frees the raw
        if cuda:                # features (57 GB for 1-partition papers100M)
            torch.cuda.empty_cache()
    if world > 1:
        for prm in model.parameters():
            dist.broadcast(prm.data, src=0)

    multilabel = bool(meta.get("multilabel", False))
    if multilabel:
        loss_fcn = torch.nn.BCEWithLogitsLoss(reduction="sum")
        labels_train = state.label[state.train_mask].float()
    else:
        loss_fcn = torch.nn.CrossEntropyLoss(reduction="sum")
        labels_train = state.label[state.train_mask].long()
    reducer = GradReducer(model, int(meta["n_train"]))
    try:        # fused Adam: one multi-tensor kernel instead of ~6 (K11)
        optimizer = torch.optim.Adam(model.parameters(), lr=args.lr,
                                     fused=cuda)
    except (RuntimeError, ValueError):
        optimizer = torch.optim.Adam(model.parameters(), lr=args.lr)

    def step(epoch: int):
        state.plan.set_epoch(epoch)
        model.train()
        logits = _forward(model, state, state.feat)
        if state.loss_rows is not None:     # final layer already restricted
            loss = loss_fcn(logits, labels_train)
        else:
            loss = loss_fcn(logits[state.train_mask], labels_train)
        reducer.zero_grad()
        loss.backward()
        reducer.reduce()
        # build next epoch's sampling plan on a side stream, overlapped
        # with this epoch's queued GPU work
        state.prefetch(epoch + 1)
        reducer.synchronize()
        optimizer.step()
        return loss

    for e in range(a.warmup):
        step(e)

    if cuda:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    t0 = time.perf_counter()
    for e in range(a.steps):
        step(a.warmup + e)
    if cuda:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    final_loss = step(a.warmup + a.steps)   # untimed: numerics guard
    assert torch.isfinite(final_loss), \
        f"non-finite loss after {a.steps} steps — bench result invalid"

    # MAX over ranks (NCCL needs a device-resident tensor)
    t = torch.tensor([elapsed], device=device if cuda else "cpu")
    if world > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t[0])

    if rank == 0:
        n_edges = int(meta.get("full_n_edges", meta["n_edges"]))
        epoch_s = elapsed / a.steps
        value = n_edges / epoch_s
        ref_epoch_s = 0.3578          # BASELINE.md (README.md:94-95)
        baseline = n_edges / ref_epoch_s if a.dataset == "reddit" and \
            a.data_scale == 1.0 else None
        out = {
            "metric": f"training throughput (full-graph edges/s), "
                      f"{a.dataset}-shaped {a.model} {a.n_layers}-layer "
                      f"h={a.n_hidden}, sampling-rate={a.sampling_rate}",
            "value": value,
            "unit": "edges/s",
            "n_gpus": world,
            "steps": a.steps,
            "warmup": a.warmup,
            "ms_per_step": epoch_s * 1e3,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": (value / baseline) if baseline else None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {"model": a.model, "dataset": a.dataset,
                       "n_layers": a.n_layers, "n_hidden": a.n_hidden,
                       "sampling_rate": a.sampling_rate,
                       "n_nodes": int(meta.get("full_n_nodes", meta["n_nodes"])),
                       "n_edges": n_edges, "use_pp": a.use_pp,
                       "partition": a.partition_method,
                       "parallelism": f"partition-parallel p{world}",
                       "epoch_time_s": epoch_s},
        }
        print(json.dumps(out), flush=True)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
