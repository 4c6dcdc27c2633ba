"""Training runtime: setup, precompute, epoch loop, eval, checkpoint.

Reference counterpart: train.py (run/init_processes and its 15 helpers —
call stacks in SURVEY.md §3). Key structural differences (MI355X-first):

* no per-epoch DGL graph rebuild — HaloPlan.set_epoch does CSR gathers;
* no NODE-id exchange — shared Philox sampling (parallel/plan.py);
* the halo all-to-all overlaps inner-edge SpMM (parallel/halo.py);
* gradient sync is one bucketed RCCL all-reduce (parallel/reducer.py).
Observable behavior kept: loss = sum-reduced CE/BCE over local train rows
scaled 1/n_train at grad sync (train.py:358-361, reducer.py:34), per-epoch
log line, checkpoint names `checkpoint/<graph>_p<rate>_<epoch>.pth.tar` /
`<graph>_final.pth.tar` (train.py:428,452), full-graph eval on rank 0 in a
background thread (train.py:434-442).
"""
from __future__ import annotations

import copy
import os
import time
from concurrent.futures import ThreadPoolExecutor

import numpy as np
import torch
import torch.distributed as dist

from ..graph import load_data, load_partition, partition_and_save
from ..graph.store import Partition
from ..models.context import GraphContext
from ..models.models import create_model
from ..ops.functional import pack_rows_raw, spmm_sum_raw
from ..parallel import GradReducer, HaloPlan, all_to_all_rows, init_distributed
from ..utils.timer import comm_timer
from .config import graph_name_of


# --------------------------------------------------------------- offline

def _load_data_cached(args):
    """Generate the synthetic dataset, caching the edge arrays on disk so
    back-to-back partitionings for different N (the driver's 1/2/4/8-GPU
    scaling sweep) skip the expensive edge sampling."""
    import numpy as np
    from ..graph.csr import CSR, Graph
    from ..graph.ingest import disk_dataset_file, load_disk_data
    data_path = getattr(args, "data_path", None)
    if disk_dataset_file(args.dataset, data_path):
        # pre-downloaded real dataset on disk (reference helper/utils.py
        # loads via dgl/ogb + network; see graph/ingest.py for the layout)
        return load_disk_data(args.dataset, data_path)
    cache_dir = data_path or args.partition_dir
    cache = os.path.join(cache_dir,
                         f"_edges_{args.dataset}_s{args.seed}"
                         f"_x{args.data_scale:g}.npz")
    from ..graph.synthetic import LazyFeat
    if os.path.exists(cache):
        z = np.load(cache)
        n_nodes = int(z["n_nodes"])
        adj = CSR(z["indptr"], z["indices"], n_nodes)
        if "procedural_seed" in z:
            feat = LazyFeat(int(z["procedural_seed"]), int(z["procedural_nf"]),
                            np.arange(n_nodes, dtype=np.int64))
        else:
            feat = z["feat"]
        g = Graph(adj, feat, z["label"], z["train_mask"], z["val_mask"],
                  z["test_mask"], int(z["n_class"]), bool(z["multilabel"]),
                  name=args.dataset)
    else:
        g = load_data(args.dataset, seed=args.seed, scale=args.data_scale)
        if isinstance(g.feat, LazyFeat):
            extra = {"feat": np.zeros((0, g.n_feat), dtype=np.float32),
                     "procedural_seed": g.feat.seed, "procedural_nf": g.n_feat}
        else:
            extra = {"feat": g.feat}
        try:
            os.makedirs(cache_dir, exist_ok=True)
            np.savez(cache, indptr=g.adj_in.indptr, indices=g.adj_in.indices,
                     n_nodes=g.n_nodes, label=g.label,
                     train_mask=g.train_mask, val_mask=g.val_mask,
                     test_mask=g.test_mask, n_class=g.n_class,
                     multilabel=g.multilabel, **extra)
        except OSError:
            pass
    return g


def prepare_partitions(args) -> str:
    """Rank-0 offline step (reference main.py:26-31 + graph_partition).

    With --eval-mode dist in the INDUCTIVE setting, two extra partition
    stores are written ("<name>-evalval" = the train∪val graph,
    "<name>-evaltest" = the full graph) so evaluation can also run as a
    distributed p=1.0 forward instead of the rank-0 CPU pass."""
    name = graph_name_of(args)
    d = os.path.join(args.partition_dir, name)
    if args.skip_partition and os.path.exists(os.path.join(d, "meta.json")):
        return d
    g_full = _load_data_cached(args)
    g = g_full.subgraph(g_full.train_mask, name=g_full.name) \
        if args.inductive else g_full
    extra = {"inductive": args.inductive, "dataset_seed": args.seed,
             "data_scale": args.data_scale,
             "full_n_nodes": g.n_nodes, "full_n_edges": g.n_edges}
    partition_and_save(g, args.n_partitions, args.partition_method,
                       args.partition_dir, name, seed=args.seed,
                       objective=args.partition_obj, extra_meta=extra)
    if (args.inductive and args.eval
            and getattr(args, "eval_mode", "thread") == "dist"):
        g_val = g_full.subgraph(g_full.train_mask | g_full.val_mask,
                                name=g_full.name)
        for suffix, gg in (("-evalval", g_val), ("-evaltest", g_full)):
            partition_and_save(gg, args.n_partitions, args.partition_method,
                               args.partition_dir, name + suffix,
                               seed=args.seed, objective=args.partition_obj,
                               extra_meta=extra)
    return d


# ------------------------------------------------------------- rank setup

def _feat_to_device(feat, device) -> torch.Tensor:
    """numpy features -> device copy; LazyFeat (papers100M procedural
    features) -> materialize directly into device memory (the 1-partition
    papers matrix is 57 GB — it never exists on the host)."""
    from ..graph.synthetic import LazyFeat
    if isinstance(feat, LazyFeat):
        return feat.materialize_torch(device)
    return torch.from_numpy(feat).to(device)


class RankState:
    """Everything one rank needs for the epoch loop."""

    def __init__(self, part: Partition, args, device):
        self.device = torch.device(device)
        self.part = part
        self.args = args
        dev = self.device
        self.feat = _feat_to_device(part.feat, dev)
        lab = torch.from_numpy(part.label)
        self.label = lab.to(dev)
        self.train_mask = torch.from_numpy(part.train_mask).to(dev)
        self.val_mask = torch.from_numpy(part.val_mask).to(dev)
        self.test_mask = torch.from_numpy(part.test_mask).to(dev)
        self.raw_feat = self.feat   # pre-precompute features (dist eval)
        wire = torch.bfloat16 \
            if getattr(args, "halo_dtype", "fp32") == "bf16" else None
        self.plan = HaloPlan(part, args.sampling_rate, seed=args.seed, device=dev,
                             unit_ratio=(args.model == "gat"
                                         and not args.gat_ratio_scale),
                             wire_dtype=wire)
        self.ctx = GraphContext.for_partition(part, self.plan, dev,
                                              need_eperm=(args.model == "gat"))
        self.halo_feat0 = None     # GAT use_pp layer-0 full halo features
        self.n_train_global = int(part.meta["n_train"])
        self.part_train = int(part.train_mask.sum())
        # final-layer loss-row restriction (all models): train-time logits
        # are computed only for labeled rows — identical loss/gradients,
        # and [N, C] logits shrink to [n_train_local, C] (papers100M:
        # 76 GB -> 0.8 GB). BNSGCN_FULL_LOGITS=1 restores the reference's
        # full pass.
        self.loss_rows = None
        if os.environ.get("BNSGCN_FULL_LOGITS") != "1":
            self.loss_rows = torch.nonzero(self.train_mask).flatten()
            self.ctx.loss_rows = self.loss_rows

    def prefetch(self, epoch: int) -> None:
        """Overlap the NEXT epoch's sampling-plan build (and the
        restricted-halo gather when loss-row restriction is on) with the
        current epoch's queued GPU work, on a side stream."""
        self.plan.prefetch(epoch)
        nxt = getattr(self.plan, "_next", None)
        if nxt is not None and self.ctx.loss_rows is not None:
            self.ctx.prefetch_rows_halo(nxt[1], self.plan._prefetch_stream,
                                        gat=(self.args.model == "gat"))

    # ---------------------------------------------------------- precompute
    @torch.no_grad()
    def precompute(self):
        """use_pp pass (reference train.py:170-211): one FULL (unsampled)
        boundary feature exchange + first-layer aggregation, after which
        training layer 0 degenerates to a GEMM."""
        args, ctx = self.args, self.ctx
        st = ctx.full_state()
        send = pack_rows_raw(self.feat, st.pack_idx, None)
        recv = torch.empty(sum(st.recv_counts), self.feat.shape[1],
                           dtype=self.feat.dtype, device=self.feat.device)
        all_to_all_rows(recv, send, st.recv_counts, st.send_counts)
        self.raw_feat = self.feat
        if args.model == "gcn":
            out = spmm_sum_raw(ctx.indptr, ctx.indices, self.feat,
                               ctx.out_norm_inv, ctx.in_norm_inv)
            spmm_sum_raw(st.halo_fwd_indptr, st.halo_fwd_indices, recv,
                         src_scale=st.halo_out_norm_inv,
                         dst_scale=ctx.in_norm_inv, out=out)
            self.feat = out
        elif args.model == "graphsage":
            mean = spmm_sum_raw(ctx.indptr, ctx.indices, self.feat,
                                None, ctx.in_deg_inv)
            spmm_sum_raw(st.halo_fwd_indptr, st.halo_fwd_indices, recv,
                         src_scale=None, dst_scale=ctx.in_deg_inv, out=mean)
            self.feat = torch.cat((self.feat, mean), dim=1)
        elif args.model == "gat":
            self.halo_feat0 = recv   # keep raw full-halo features
        else:
            raise ValueError(args.model)


class EvalState:
    """Lightweight per-rank state over an EVALUATION partition store (the
    train∪val graph / the full graph in the inductive setting): features,
    labels, masks, context + a p=1.0 plan. Used by dist_evaluate."""

    def __init__(self, part: Partition, args, device):
        self.device = torch.device(device)
        self.part = part
        self.args = args
        self.feat = _feat_to_device(part.feat, device)
        self.raw_feat = self.feat
        self.label = torch.from_numpy(part.label).to(device)
        self.val_mask = torch.from_numpy(part.val_mask).to(device)
        self.test_mask = torch.from_numpy(part.test_mask).to(device)
        self.plan = HaloPlan(part, 1.0, seed=args.seed, device=device,
                             unit_ratio=True)
        self.plan.set_epoch(0)
        self.ctx = GraphContext.for_partition(part, self.plan, device)


@torch.no_grad()
def dist_evaluate(state, model) -> dict:
    """Exact full-graph evaluation ACROSS partitions: every rank runs an
    eval-mode forward with the FULL (p=1.0) halo state on its own
    partition, then correctness counts are all-reduced. Replaces the
    reference's rank-0 CPU full-graph pass (train.py:434-442) with a
    collective that runs on the GPUs in milliseconds. Must be called by
    ALL ranks. `state` is the training RankState (transductive: the
    training partitions ARE the full graph) or an EvalState over a
    dedicated eval store (inductive)."""
    plan = state.plan
    saved = plan._state
    plan._state = state.ctx.full_state()
    was_training = model.training
    try:
        model.eval()
        logits = model(state.ctx, state.raw_feat)
    finally:
        plan._state = saved
        if was_training:
            model.train()
    multilabel = bool(state.part.meta.get("multilabel", False))
    out = {}
    if multilabel:
        pred = (logits > 0).float()
        lab = state.label.float()
        stats = []
        for m in (state.val_mask, state.test_mask):
            tp = (pred[m] * lab[m]).sum()
            fp = (pred[m] * (1 - lab[m])).sum()
            fn = ((1 - pred[m]) * lab[m]).sum()
            stats.append(torch.stack([tp, fp, fn]))
        t = torch.cat(stats)
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(t)
        for key, (tp, fp, fn) in zip(("val", "test"), t.view(2, 3)):
            out[key] = float(2 * tp / (2 * tp + fp + fn + 1e-12))
    else:
        pred = logits.argmax(1)
        t = torch.stack([
            (pred[state.val_mask] == state.label[state.val_mask]).sum(),
            state.val_mask.sum(),
            (pred[state.test_mask] == state.label[state.test_mask]).sum(),
            state.test_mask.sum()]).float()
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(t)
        out["val"] = float(t[0] / t[1].clamp_min(1))
        out["test"] = float(t[2] / t[3].clamp_min(1))
    return out


def _forward(model, state: RankState, feat):
    if state.args.model == "gat":
        return model(state.ctx, feat, halo_feat0=state.halo_feat0)
    return model(state.ctx, feat)


def forward_train_logits(model, state: RankState) -> torch.Tensor:
    """Training forward returning logits FOR THE TRAIN ROWS, regardless of
    whether the final-layer loss-row restriction is active."""
    logits = _forward(model, state, state.feat)
    if getattr(state, "loss_rows", None) is not None and model.training:
        return logits
    return logits[state.train_mask]


# -------------------------------------------------------------- evaluation

def _accuracy(logits: torch.Tensor, labels: torch.Tensor, multilabel: bool) -> float:
    if multilabel:
        pred = (logits > 0).float()
        tp = (pred * labels).sum()
        fp = (pred * (1 - labels)).sum()
        fn = ((1 - pred) * labels).sum()
        return float(2 * tp / (2 * tp + fp + fn + 1e-12))
    return float((logits.argmax(1) == labels).float().mean())


class Evaluator:
    """Full-graph inference on CPU (rank 0), reference train.py:13-61.

    Regenerates the deterministic synthetic dataset instead of loading it
    from disk; transductive: one pass over the full graph scoring val+test;
    inductive: val on the train∪val subgraph, test on the full graph."""

    def __init__(self, args):
        self.args = args
        self.device = torch.device(getattr(args, "eval_device", "cpu"))
        self.multilabel = None
        from ..graph.ingest import disk_dataset_file, load_disk_data
        dp = getattr(args, "data_path", None)
        if disk_dataset_file(args.dataset, dp):
            g = load_disk_data(args.dataset, dp)
        else:
            g = load_data(args.dataset, seed=args.seed, scale=args.data_scale)
        self.multilabel = g.multilabel
        self._graphs = {}
        if args.inductive:
            self._graphs["val"] = g.subgraph(g.train_mask | g.val_mask)
            self._graphs["test"] = g
        else:
            self._graphs["full"] = g

    def _ctx_and_tensors(self, g):
        ctx = GraphContext.for_full_graph(
            torch.from_numpy(g.adj_in.indptr), torch.from_numpy(g.adj_in.indices),
            torch.from_numpy(g.in_deg), torch.from_numpy(g.out_deg), self.device)
        feat = _feat_to_device(g.feat, self.device)
        label = torch.from_numpy(g.label).to(self.device)
        return ctx, feat, label, g

    @torch.no_grad()
    def evaluate(self, model_cpu) -> dict:
        model_cpu = model_cpu.to(self.device)
        model_cpu.eval()
        out = {}
        if self.args.inductive:
            ctx, feat, label, g = self._ctx_and_tensors(self._graphs["val"])
            logits = model_cpu(ctx, feat)
            vm = torch.from_numpy(g.val_mask).to(self.device)
            out["val"] = _accuracy(logits[vm], label[vm], self.multilabel)
        else:
            ctx, feat, label, g = self._ctx_and_tensors(self._graphs["full"])
            logits = model_cpu(ctx, feat)
            vm = torch.from_numpy(g.val_mask).to(self.device)
            tm = torch.from_numpy(g.test_mask).to(self.device)
            out["val"] = _accuracy(logits[vm], label[vm], self.multilabel)
            out["test"] = _accuracy(logits[tm], label[tm], self.multilabel)
        return out

    @torch.no_grad()
    def evaluate_test(self, model_cpu) -> float:
        model_cpu = model_cpu.to(self.device)
        model_cpu.eval()
        key = "test" if self.args.inductive else "full"
        ctx, feat, label, g = self._ctx_and_tensors(self._graphs[key])
        logits = model_cpu(ctx, feat)
        tm = torch.from_numpy(g.test_mask).to(self.device)
        return _accuracy(logits[tm], label[tm], self.multilabel)


# --------------------------------------------------------------- the loop

def run(args, rank: int | None = None, world_size: int | None = None) -> dict:
    """One training process (= one partition = one GPU). Returns summary
    stats (epoch-time mean, loss, accuracies) for tests/bench."""
    # pin the device BEFORE the process group exists: an RCCL communicator
    # built while every rank still sits on cuda:0 is duplicate-device and
    # aborts the whole job (same guard as bench.py:88-96)
    pre_rank = int(os.environ.get("RANK", rank if rank is not None else 0))
    pre_world = int(os.environ.get("WORLD_SIZE",
                                   world_size if world_size is not None else 1))
    if args.device == "auto":
        if torch.cuda.is_available():
            local = int(os.environ.get(
                "LOCAL_RANK", pre_rank % max(torch.cuda.device_count(), 1)))
            device = f"cuda:{local % torch.cuda.device_count()}"
        else:
            device = "cpu"
    else:
        device = args.device
    if str(device).startswith("cuda"):
        backend_eff = args.backend
        if backend_eff in (None, "auto"):
            backend_eff = "nccl"
        if backend_eff == "nccl" and pre_world > torch.cuda.device_count():
            raise RuntimeError(
                f"{pre_world} ranks > {torch.cuda.device_count()} GPUs: RCCL "
                "cannot oversubscribe devices (the reference's gloo backend "
                "could, main.py:45). Use --n-partitions <= GPU count, or "
                "--backend gloo (CUDA payloads are host-staged).")
        torch.cuda.set_device(torch.device(device))
    rank, world = init_distributed(args.backend, rank, world_size,
                                   args.master_addr, args.port)

    name = graph_name_of(args)
    part = load_partition(args.partition_dir, name, rank)
    assert part.n_parts == world, (part.n_parts, world)
    if args.fix_seed:
        torch.manual_seed(args.seed)

    state = RankState(part, args, device)
    state.plan.set_epoch(0)
    model = create_model(args, n_feat=state.feat.shape[1],
                         n_class=int(part.meta["n_class"]),
                         train_size=state.n_train_global).to(device)
    if args.use_pp or args.model == "gat":
        state.precompute()

    if getattr(args, "resume", ""):
        sd = torch.load(args.resume, map_location=device)
        model.load_state_dict(sd)
        print(f"Process {rank:03d} | resumed from {args.resume}", flush=True)

    # broadcast initial weights so all ranks start identical (the reference
    # relies on --fix-seed for this, main.py:13-16; we make it robust)
    if world > 1:
        for p in model.parameters():
            dist.broadcast(p.data, src=0)

    multilabel = bool(part.meta.get("multilabel", False))
    if multilabel:
        loss_fcn = torch.nn.BCEWithLogitsLoss(reduction="sum")
        labels_train = state.label[state.train_mask].float()
    else:
        loss_fcn = torch.nn.CrossEntropyLoss(reduction="sum")
        labels_train = state.label[state.train_mask].long()

    reducer = GradReducer(model, state.n_train_global,
                          bucket_bytes=args.bucket_mb << 20)
    try:        # fused Adam: one multi-tensor kernel instead of ~6 (K11)
        optimizer = torch.optim.Adam(model.parameters(), lr=args.lr,
                                     weight_decay=args.weight_decay,
                                     fused=str(device).startswith("cuda"))
    except (RuntimeError, ValueError):
        optimizer = torch.optim.Adam(model.parameters(), lr=args.lr,
                                     weight_decay=args.weight_decay)

    evaluator = None
    pool = None
    pending = None
    best_val, best_state = -1.0, None
    best_dist_test = None
    eval_mode = getattr(args, "eval_mode", "thread")
    dist_eval_on = args.eval and eval_mode == "dist"
    eval_states = None
    if dist_eval_on and args.inductive:
        # inductive: evaluation graphs differ from the training graph —
        # load the dedicated eval stores written by prepare_partitions
        try:
            pv = load_partition(args.partition_dir, name + "-evalval", rank)
            pt = load_partition(args.partition_dir, name + "-evaltest", rank)
            eval_states = (EvalState(pv, args, device),
                           EvalState(pt, args, device))
            have_stores = 1.0
        except FileNotFoundError:
            have_stores = 0.0
        # agree across ranks: if the stores are missing on ANY rank, every
        # rank must fall back together, else dist_evaluate's collectives
        # deadlock against thread-mode ranks
        if world > 1:
            flag = torch.tensor([have_stores],    # NCCL: device-resident
                                device=device if str(device).startswith("cuda")
                                else "cpu")
            dist.all_reduce(flag, op=dist.ReduceOp.MIN)
            have_stores = float(flag[0])
        if have_stores < 1.0:
            if rank == 0:
                print("inductive dist eval stores missing (re-run "
                      "partitioning with --eval-mode dist); falling back "
                      "to thread mode", flush=True)
            eval_mode = "thread"
            dist_eval_on = False
            eval_states = None
    if args.eval and rank == 0:
        if not dist_eval_on:
            evaluator = Evaluator(args)
            pool = ThreadPoolExecutor(max_workers=1)
        os.makedirs("checkpoint", exist_ok=True)
        os.makedirs("results", exist_ok=True)

    train_dur, comm_dur, reduce_dur = [], [], []
    loss_history: list[float] = []
    loss_val = float("nan")
    cuda = str(device).startswith("cuda")
    if cuda:
        torch.cuda.reset_peak_memory_stats()

    for epoch in range(args.n_epochs):
        t0 = time.perf_counter()
        state.plan.set_epoch(epoch)
        model.train()
        logits = _forward(model, state, state.feat)
        if state.loss_rows is not None:     # final layer already restricted
            loss = loss_fcn(logits, labels_train)
        else:
            loss = loss_fcn(logits[state.train_mask], labels_train)
        reducer.zero_grad()
        loss.backward()
        t_red = time.perf_counter()
        reducer.reduce()
        state.prefetch(epoch + 1)        # overlap next epoch's sampling
        reducer.synchronize()
        reduce_dur.append(time.perf_counter() - t_red)
        optimizer.step()
        if cuda:
            torch.cuda.synchronize()
        if epoch >= 5:
            train_dur.append(time.perf_counter() - t0)
        comm_dur.append(comm_timer.tot_time())
        comm_timer.clear()
        loss_val = loss.item()
        loss_history.append(loss_val)

        if (epoch + 1) % args.log_every == 0:
            print(f"Process {rank:03d} | Epoch {epoch:05d} | "
                  f"Time(s) {np.mean(train_dur) if train_dur else 0:.4f} | "
                  f"Comm(s) {np.mean(comm_dur):.4f} | "
                  f"Reduce(s) {np.mean(reduce_dur):.4f} | "
                  f"Loss {loss_val / max(state.part_train, 1):.4f}", flush=True)

        if dist_eval_on and (epoch + 1) % args.log_every == 0:
            # COLLECTIVE: every rank participates
            if eval_states is None:          # transductive: training
                res = dist_evaluate(state, model)    # partitions = full graph
            else:                            # inductive: dedicated stores
                res = {"val": dist_evaluate(eval_states[0], model)["val"],
                       "test": dist_evaluate(eval_states[1], model)["test"]}
            if rank == 0:
                torch.save(model.state_dict(),
                           f"checkpoint/{name}_p{args.sampling_rate:.2f}"
                           f"_{epoch}.pth.tar")
                if res["val"] > best_val:
                    best_val = res["val"]
                    best_dist_test = res["test"]
                    best_state = {k: v.cpu() for k, v in
                                  model.state_dict().items()}

        if evaluator is not None and (epoch + 1) % args.log_every == 0:
            torch.save(model.state_dict(),
                       f"checkpoint/{name}_p{args.sampling_rate:.2f}_{epoch}.pth.tar")
            if pending is not None:
                res, snap = pending.result()
                if res["val"] > best_val:
                    best_val, best_state = res["val"], snap
            model_cpu = copy.deepcopy(model).cpu()
            snap = copy.deepcopy(model_cpu.state_dict())
            pending = pool.submit(lambda m=model_cpu, s=snap:
                                  (evaluator.evaluate(m), s))

    if cuda:
        # reference print_memory (helper/utils.py:244-250, train.py:444)
        print(f"Process {rank:03d} | "
              f"allocated {torch.cuda.memory_allocated() / 2**20:.0f} MB | "
              f"peak {torch.cuda.max_memory_allocated() / 2**20:.0f} MB | "
              f"reserved {torch.cuda.memory_reserved() / 2**20:.0f} MB",
              flush=True)
    summary = {"rank": rank, "epoch_time": float(np.mean(train_dur)) if train_dur
               else float("nan"),
               "comm_time": float(np.mean(comm_dur)) if comm_dur else 0.0,
               "loss": loss_val / max(state.part_train, 1),
               "loss_history": loss_history}
    if dist_eval_on and rank == 0 and best_state is not None:
        torch.save(best_state, f"checkpoint/{name}_final.pth.tar")
        summary["val_acc"] = best_val
        summary["test_acc"] = best_dist_test
        with open(f"results/{args.dataset}_n{args.n_partitions}"
                  f"_p{args.sampling_rate:.2f}.txt", "a") as f:
            f.write(f"val={best_val:.4f} test={best_dist_test:.4f}\n")
        print(f"Max Validation Accuracy {best_val:.4f} | "
              f"Test Accuracy {best_dist_test:.4f}", flush=True)
    if evaluator is not None:
        if pending is not None:
            res, snap = pending.result()
            if res["val"] > best_val:
                best_val, best_state = res["val"], snap
        if best_state is not None:
            torch.save(best_state, f"checkpoint/{name}_final.pth.tar")
            # rebuild with the ORIGINAL feature width (precompute may have
            # widened state.feat; eval consumes raw features)
            final = create_model(args, n_feat=int(part.meta["n_feat"]),
                                 n_class=int(part.meta["n_class"]),
                                 train_size=state.n_train_global)
            final.load_state_dict(best_state)
            test_acc = evaluator.evaluate_test(final)
            summary["val_acc"] = best_val
            summary["test_acc"] = test_acc
            with open(f"results/{args.dataset}_n{args.n_partitions}"
                      f"_p{args.sampling_rate:.2f}.txt", "a") as f:
                f.write(f"val={best_val:.4f} test={test_acc:.4f}\n")
            print(f"Max Validation Accuracy {best_val:.4f} | "
                  f"Test Accuracy {test_acc:.4f}", flush=True)
        if pool is not None:
            pool.shutdown()
    return summary
