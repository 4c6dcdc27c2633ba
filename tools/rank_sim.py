"""Simulate ONE rank of an N-partition job without a process group.

Collectives are stubbed (recv buffers zero-filled), so this measures the
per-rank COMPUTE time of an N-way-partitioned epoch on a single GPU — the
scaling ceiling the driver's real N-GPU run should approach when the halo
all-to-all overlaps (bandwidth for the real exchange at p=0.1 is tens of
MB per step over 7 xGMI links, well under the compute time).

  python tools/rank_sim.py --parts 8 [--dataset reddit ...]
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def fake_all_to_all_rows(recv, send, recv_counts, send_counts, async_op=False):
    recv.zero_()
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--parts", type=int, default=8)
    ap.add_argument("--rank", type=int, default=0)
    ap.add_argument("--dataset", default="reddit")
    ap.add_argument("--model", default="graphsage")
    ap.add_argument("--n-layers", type=int, default=3)
    ap.add_argument("--n-hidden", type=int, default=256)
    ap.add_argument("--heads", type=int, default=4)
    ap.add_argument("--sampling-rate", type=float, default=0.1)
    ap.add_argument("--data-scale", type=float, default=1.0)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--partition-dir", default="/tmp/ranksim")
    a = ap.parse_args()

    import bnsgcn_amd.parallel.comm as comm
    import bnsgcn_amd.parallel.halo as halo
    import bnsgcn_amd.runtime.trainer as trainer
    comm.all_to_all_rows = fake_all_to_all_rows
    halo.all_to_all_rows = fake_all_to_all_rows
    trainer.all_to_all_rows = fake_all_to_all_rows

    from bnsgcn_amd.runtime.config import create_parser, graph_name_of
    from bnsgcn_amd.runtime.trainer import prepare_partitions, RankState, _forward, forward_train_logits
    from bnsgcn_amd.graph import load_partition
    from bnsgcn_amd.models.models import create_model
    from bnsgcn_amd.parallel import GradReducer

    args = create_parser().parse_args([])
    for k in ("dataset", "model", "n_layers", "n_hidden", "heads",
              "sampling_rate", "data_scale", "partition_dir"):
        setattr(args, k, getattr(a, k))
    args.n_partitions = a.parts
    args.use_pp = True
    args.eval = False
    args.graph_name = graph_name_of(args)
    args.skip_partition = True
    prepare_partitions(args)

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    part = load_partition(args.partition_dir, args.graph_name, a.rank)
    torch.manual_seed(0)
    state = RankState(part, args, device)
    state.plan.set_epoch(0)
    model = create_model(args, n_feat=int(part.meta["n_feat"]),
                         n_class=int(part.meta["n_class"]),
                         train_size=int(part.meta["n_train"])).to(device)
    state.precompute()
    reducer = GradReducer(model, int(part.meta["n_train"]))
    opt = torch.optim.Adam(model.parameters(), lr=1e-2)
    multilabel = bool(part.meta.get("multilabel", False))
    lf = (torch.nn.BCEWithLogitsLoss(reduction="sum") if multilabel
          else torch.nn.CrossEntropyLoss(reduction="sum"))
    labels = (state.label[state.train_mask].float() if multilabel
              else state.label[state.train_mask].long())

    def step(e):
        state.plan.set_epoch(e)
        model.train()
        loss = lf(forward_train_logits(model, state), labels)
        reducer.zero_grad()
        loss.backward()
        reducer.reduce()
        reducer.synchronize()
        opt.step()

    for e in range(a.warmup):
        step(e)
    if device != "cpu":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for e in range(a.steps):
        step(a.warmup + e)
    if device != "cpu":
        torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / a.steps * 1e3
    print(f"rank {a.rank}/{a.parts} {a.dataset} {a.model}: "
          f"{ms:.2f} ms/epoch (compute only, comm stubbed)", flush=True)


if __name__ == "__main__":
    main()
