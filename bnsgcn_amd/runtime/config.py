"""CLI flag system — compatible with the reference parser
(helper/parser.py:4-61): same flag names (kebab and snake aliases), same
defaults, plus a few MI355X-framework extras (clearly marked).
"""
from __future__ import annotations

import argparse


def create_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="bnsgcn_amd — MI355X-native BNS-GCN")
    # --- reference-compatible flags (helper/parser.py) ---
    p.add_argument("--dataset", type=str, default="reddit")
    p.add_argument("--graph-name", "--graph_name", type=str, default="")
    p.add_argument("--model", type=str, default="graphsage",
                   choices=["gcn", "graphsage", "gat"])
    p.add_argument("--dropout", type=float, default=0.5)
    p.add_argument("--lr", type=float, default=1e-2)
    p.add_argument("--sampling-rate", "--sampling_rate", type=float, default=1.0)
    p.add_argument("--heads", type=int, default=1)
    p.add_argument("--n-epochs", "--n_epochs", type=int, default=200)
    p.add_argument("--n-partitions", "--n_partitions", type=int, default=2)
    p.add_argument("--n-hidden", "--n_hidden", type=int, default=16)
    p.add_argument("--n-layers", "--n_layers", type=int, default=2)
    p.add_argument("--n-linear", "--n_linear", type=int, default=0)
    p.add_argument("--norm", choices=["layer", "batch", "none"], default="layer")
    p.add_argument("--weight-decay", "--weight_decay", type=float, default=0)
    p.add_argument("--log-every", "--log_every", type=int, default=10)
    p.add_argument("--use-pp", "--use_pp", action="store_true")
    p.add_argument("--inductive", action="store_true")
    p.add_argument("--fix-seed", "--fix_seed", action="store_true")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--backend", type=str, default="auto",
                   choices=["auto", "nccl", "gloo", "mpi"])
    p.add_argument("--port", type=int, default=18118)
    p.add_argument("--master-addr", "--master_addr", type=str, default="127.0.0.1")
    p.add_argument("--node-rank", "--node_rank", type=int, default=0)
    p.add_argument("--parts-per-node", "--parts_per_node", type=int, default=10)
    p.add_argument("--partition-method", "--partition_method", type=str,
                   default="metis",
                   choices=["metis", "random", "bfs", "contiguous"])
    p.add_argument("--partition-obj", "--partition_obj", type=str,
                   default="vol", choices=["vol", "cut"])
    p.add_argument("--skip-partition", "--skip_partition", action="store_true")
    p.add_argument("--eval", action="store_true", dest="eval",
                   default=True, help="enable evaluation (default)")
    p.add_argument("--no-eval", action="store_false", dest="eval")
    p.add_argument("--partition-dir", "--partition_dir", "--part-path",
                   "--part_path", dest="partition_dir", type=str,
                   default="partition")
    p.add_argument("--data-path", "--data_path", type=str, default="./dataset/",
                   help="dataset storage dir (reference parity; holds the "
                        "synthetic edge cache here)")
    # --- bnsgcn_amd extras ---
    p.add_argument("--data-scale", type=float, default=1.0,
                   help="shrink the synthetic dataset (papers100M smoke runs)")
    p.add_argument("--device", type=str, default="auto",
                   help="cuda | cpu | auto")
    p.add_argument("--gat-ratio-scale", action="store_true",
                   help="reproduce the reference's (acknowledged-wrong, "
                        "train.py:117) 1/ratio scaling of GAT attention inputs; "
                        "default keeps ratio=1 for GAT")
    p.add_argument("--resume", type=str, default="",
                   help="checkpoint (.pth.tar state_dict) to load before "
                        "training — the reference saves checkpoints but has "
                        "no resume path (SURVEY.md §5.4); this adds one")
    p.add_argument("--eval-mode", choices=["thread", "dist"], default="thread",
                   help="'thread' = reference-style rank-0 full-graph eval in "
                        "a background thread; 'dist' = exact p=1.0 eval-mode "
                        "forward ACROSS the training partitions with "
                        "all-reduced accuracy (transductive only, runs on "
                        "the GPUs — no CPU full-graph pass)")
    p.add_argument("--eval-device", type=str, default="cpu",
                   help="device for rank-0 full-graph evaluation (the "
                        "reference evaluates on CPU; 'cuda' runs it on the "
                        "rank-0 GPU with the HIP kernels)")
    p.add_argument("--bucket-mb", type=int, default=16,
                   help="gradient all-reduce bucket size (MiB)")
    p.add_argument("--halo-dtype", "--halo_dtype", choices=["fp32", "bf16"],
                   default="fp32",
                   help="wire dtype for the per-layer halo all-to-all: bf16 "
                        "halves the xGMI bytes; compute stays fp32")
    return p


def graph_name_of(args) -> str:
    """Reference naming (main.py:18-24); a non-unit --data-scale is
    appended so differently-scaled synthetic stores never collide in the
    same partition dir (a scale-0.125 run followed by a full-scale run
    with --skip-partition semantics would silently reuse the small
    store)."""
    if getattr(args, "graph_name", ""):
        return args.graph_name
    mode = "induc" if args.inductive else "trans"
    name = (f"{args.dataset}-{args.n_partitions}-{args.partition_method}-"
            f"{args.partition_obj}-{mode}")
    scale = getattr(args, "data_scale", 1.0)
    if scale != 1.0:
        name += f"-x{scale:g}"
    return name
