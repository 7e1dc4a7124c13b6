#!/usr/bin/env python3
"""YAML-driven launcher for the distributed loader benchmark (parity:
reference benchmarks/api/run_dist_bench.py).  Single-node configs spawn
every rank locally; multi-node configs print the per-host command to run."""
import argparse
import os
import subprocess
import sys

import yaml

HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default=os.path.join(
        HERE, "bench_dist_config.yml"))
    ap.add_argument("--node-rank", type=int, default=0)
    args = ap.parse_args()
    cfg = yaml.safe_load(open(args.config))
    node = cfg["nodes"][args.node_rank]
    cmd = [sys.executable, os.path.join(HERE,
                                        "bench_dist_neighbor_loader.py"),
           "--ranks", str(cfg["nodes_total_ranks"]),
           "--nodes", str(cfg["dataset"]["nodes"]),
           "--edges", str(cfg["dataset"]["edges"]),
           "--feat-dim", str(cfg["dataset"]["feat_dim"]),
           "--fanout", cfg["sampling"]["fanout"],
           "--batch-size", str(cfg["sampling"]["batch_size"]),
           "--sampling-workers", str(cfg["sampling"]["workers_per_rank"]),
           "--batches", str(cfg["batches"])]
    if len(cfg["nodes"]) > 1:
        print("multi-node config: run on each host:", " ".join(cmd))
        if node["host"] not in ("127.0.0.1", "localhost"):
            return
    subprocess.run(cmd, check=True)


if __name__ == "__main__":
    main()
