#!/usr/bin/env python3
"""Parse and plot the per-rank training CSVs.

Consumes exactly the log schema written by gossip_sgd.py /
gossip_sgd_adpsgd.py (header: reference gossip_sgd.py:264-274; parser
parity: reference visualization/plotting.py:195-228, 255-345, plus the
fairseq-style transformer-log parser at :137-192).  Plot functions
require matplotlib; parsing needs only pandas.
"""

import argparse
import glob
import os
import re

import pandas as pd

CSV_COLUMNS = [
    "Epoch", "itr",
    "BT(s)", "avg:BT(s)", "std:BT(s)",
    "NT(s)", "avg:NT(s)", "std:NT(s)",
    "DT(s)", "avg:DT(s)", "std:DT(s)",
    "Loss", "avg:Loss", "Prec@1", "avg:Prec@1", "Prec@5", "avg:Prec@5",
    "val",
]

# iterations per epoch at the reference's experiment scales
# (reference visualization/plotting.py:196)
ITR_PER_EPOCH = {4: 1251, 8: 625, 16: 312, 32: 156}


def parse_csv(path):
    """Parse one per-rank CSV into (meta, train_df, val_df)."""
    meta = {}
    with open(path) as f:
        assert f.readline().strip() == "BEGIN-TRAINING"
        for _ in range(3):
            key, v = f.readline().strip().split(",")
            meta[key] = int(v)
        header = f.readline().strip().split(",")
        assert header == CSV_COLUMNS, header
        df = pd.read_csv(f, names=CSV_COLUMNS)
    val_df = df[df["itr"] == -1].copy()
    train_df = df[df["itr"] != -1].copy()
    return meta, train_df, val_df


def load_experiment(checkpoint_dir, tag="", world_size=None):
    """Load all ranks' CSVs for one experiment run."""
    pattern = os.path.join(checkpoint_dir, f"{tag}out_r*_n*.csv")
    runs = {}
    for path in sorted(glob.glob(pattern)):
        m = re.search(r"out_r(\d+)_n(\d+)\.csv$", path)
        if m is None:
            continue
        rank, ws = int(m.group(1)), int(m.group(2))
        if world_size is not None and ws != world_size:
            continue
        runs[rank] = parse_csv(path)
    return runs


def summarize(runs):
    """Whole-run summary: avg time/itr, best val top-1, per-rank rows."""
    rows = []
    for rank, (meta, train_df, val_df) in sorted(runs.items()):
        rows.append({
            "rank": rank,
            "world_size": meta.get("World-Size"),
            "batch_size": meta.get("Batch-Size"),
            "avg_itr_time_s": train_df["avg:BT(s)"].iloc[-1]
            if len(train_df) else None,
            "final_train_loss": train_df["avg:Loss"].iloc[-1]
            if len(train_df) else None,
            "best_val_prec1": val_df["val"].max() if len(val_df) else None,
        })
    return pd.DataFrame(rows)


def parse_transformer_out(path):
    """Parse a fairseq-style training log (the reference's transformer
    side-experiment, reference plotting.py:137-192): returns a DataFrame
    of (num_updates, valid_nll_loss) pairs."""
    rows = []
    update_re = re.compile(r"\|\s*num_updates[\s:=]+(\d+)")
    nll_re = re.compile(r"valid[^|]*\|\s*nll_loss[\s:=]+([\d.]+)")
    with open(path) as f:
        for line in f:
            if "valid" not in line:
                continue
            u = update_re.search(line)
            n = nll_re.search(line)
            if u and n:
                rows.append({
                    "num_updates": int(u.group(1)),
                    "valid_nll_loss": float(n.group(1)),
                })
    return pd.DataFrame(rows)


def _plt():
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    return plt


def plot_itrs(runs, out_path, metric="avg:Loss"):
    """Metric vs iteration for every rank (reference plotting.py:255)."""
    plt = _plt()
    fig, ax = plt.subplots()
    for rank, (meta, train_df, _) in sorted(runs.items()):
        itr_abs = train_df["Epoch"] * train_df["itr"].max() + train_df["itr"]
        ax.plot(itr_abs, train_df[metric], label=f"rank {rank}")
    ax.set_xlabel("iteration")
    ax.set_ylabel(metric)
    ax.legend()
    fig.savefig(out_path, bbox_inches="tight")
    return out_path


def plot_scaling(results, out_path):
    """Time-per-iteration (or img/s) vs node count
    (reference plotting.py:295).  `results` is {world_size: value}."""
    plt = _plt()
    fig, ax = plt.subplots()
    ks = sorted(results)
    ax.plot(ks, [results[k] for k in ks], marker="o")
    ax.set_xlabel("nodes")
    ax.set_ylabel("avg time per iteration (s)")
    ax.set_xscale("log", base=2)
    fig.savefig(out_path, bbox_inches="tight")
    return out_path


def experiment_matrix(kind, results_dir="results_dir"):
    """Experiment-tag matrix for the transport comparison plots
    (reference plotting.py:55-134 get_eth_config/get_ib_config/
    get_transformer_config): per algorithm, the CSV path template,
    the per-node-count experiment tags, legend strings and a colormap
    ramp.  Tags follow the launch scripts' checkpoint-dir naming
    (``{tag}out_r{r}_n{n}.csv``).
    """
    import numpy as np
    from matplotlib import cm

    def ramp(cmap, n):
        return [cmap(x) for x in np.linspace(0.3, 0.8, n)]

    nodes = [4, 8, 16, 32]
    fpath = results_dir + "/out_files/{tag}out_r{r}_n{n}.csv"
    if kind == "ib":
        groups = [
            ("SGP", ["SGP-%dIB" % n for n in nodes], cm.Blues),
            ("AR-SGD", ["AR-%dIB" % n for n in nodes], cm.Reds),
        ]
    elif kind == "eth":
        groups = [
            ("AR-SGD", ["AR-%dETH" % n for n in nodes], cm.Reds),
            ("D-PSGD", ["DPSGD-%dETH" % n for n in nodes], cm.Greens),
            ("SGP", ["SGP-%dETH" % n for n in nodes], cm.Blues),
        ]
    elif kind == "transformer":
        nodes = [8, 8]
        fpath = results_dir + "/transformer_{tag}_test.out"
        groups = [
            ("SGP", ["ps_sm", "ps"], cm.Blues),
            ("SGD", ["ar_sm", "ar"], cm.Reds),
        ]
    else:
        raise ValueError(f"unknown experiment matrix '{kind}'")

    fpaths, tags, legends, colors = [], [], [], []
    for name, group_tags, cmap in groups:
        fpaths.append(fpath)
        tags.append(group_tags)
        if kind == "transformer":
            legends.append([f"{name} (25K batch)", f"{name} (400K batch)"])
        else:
            legends.append([f"{name} {n} nodes" for n in nodes])
        colors.append(ramp(cmap, len(group_tags)))
    return nodes, fpaths, tags, legends, colors


def plot_matrix(kind, out_path, results_dir="results_dir",
                metric="avg:Prec@1"):
    """Overlay every algorithm's runs from the tag matrix, one curve per
    (algorithm, node-count): metric vs wall-clock training time
    (reference plotting.py's ETH/IB comparison figures).  Experiments
    whose CSVs are absent are skipped (the matrix describes the full
    grid; partial result dirs are normal)."""
    plt = _plt()
    nodes, fpaths, tags, legends, colors = experiment_matrix(
        kind, results_dir
    )
    fig, ax = plt.subplots()
    plotted = 0
    for fpath, group_tags, group_legends, group_colors in zip(
        fpaths, tags, legends, colors
    ):
        for tag, legend, color, n in zip(
            group_tags, group_legends, group_colors, nodes
        ):
            path = fpath.format(tag=tag, r=0, n=n)
            if not os.path.exists(path):
                continue
            try:
                meta, train_df, _ = parse_csv(path)
            except Exception:
                continue
            x = train_df["BT(s)"].cumsum()
            ax.plot(x, train_df[metric], label=legend, color=color)
            plotted += 1
    ax.set_xlabel("training time (s)")
    ax.set_ylabel(metric)
    if plotted:
        ax.legend()
    fig.savefig(out_path, bbox_inches="tight")
    return out_path


def plot_transformer(df, out_path):
    """Validation NLL vs optimizer steps (reference plotting.py:231)."""
    plt = _plt()
    fig, ax = plt.subplots()
    ax.plot(df["num_updates"], df["valid_nll_loss"], marker=".")
    ax.set_xlabel("num updates")
    ax.set_ylabel("valid nll loss")
    fig.savefig(out_path, bbox_inches="tight")
    return out_path


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--checkpoint_dir", required=True)
    p.add_argument("--tag", default="")
    p.add_argument("--out_dir", default=".")
    args = p.parse_args()
    runs = load_experiment(args.checkpoint_dir, args.tag)
    print(summarize(runs).to_string(index=False))
    if runs:
        plot_itrs(
            runs, os.path.join(args.out_dir, f"{args.tag}loss_vs_itr.png")
        )


if __name__ == "__main__":
    main()
