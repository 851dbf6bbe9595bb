"""Paper-figure generation — rebuild of ``src/results_plot-Adhoc.ipynb``.

Consumes the test CSV (``Adhoc_test_data_*.csv``, schema from
``AdHoc_test.py``) and produces the three ICASSP figures (notebook cells
10/13/16 → ``fig/*.pdf``):
  1. per-instance mean latency (tau) by network size, boxplot per method;
  2. congestion ratio + latency vs network size, dual-axis lineplot;
  3. per-task latency ratio vs baseline, boxplot per method.
Also provides the live training monitor aggregation (cell 5).

Run: python -m multihop_offload_amd.harness.figures --csv out/Adhoc_test_...csv
"""

from __future__ import annotations

import argparse
import os

import numpy as np
import pandas as pd


def _mpl():
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    return plt


def fig_instance_scales(df: pd.DataFrame, out: str):
    """Boxplot of per-instance tau by num_nodes × method (cell 10 analog)."""
    plt = _mpl()
    sizes = sorted(df["num_nodes"].unique())
    methods = list(dict.fromkeys(df["Algo"]))
    fig, ax = plt.subplots(figsize=(8, 4))
    width = 0.8 / len(methods)
    for mi, m in enumerate(methods):
        data = [df[(df.Algo == m) & (df.num_nodes == n)]["tau"].dropna()
                for n in sizes]
        pos = [i + mi * width for i in range(len(sizes))]
        ax.boxplot(data, positions=pos, widths=width * 0.9,
                   showfliers=False,
                   medianprops={"color": f"C{mi}"},
                   boxprops={"color": f"C{mi}"})
        ax.plot([], [], color=f"C{mi}", label=m)
    ax.set_yscale("log")
    ax.set_xticks(range(len(sizes)))
    ax.set_xticklabels(sizes)
    ax.set_xlabel("number of nodes")
    ax.set_ylabel(r"mean task latency $\tau$ (slots)")
    ax.legend()
    fig.savefig(out, bbox_inches="tight")
    plt.close(fig)


def fig_congestion_latency(df: pd.DataFrame, out: str):
    """Congestion ratio (left axis) + tau (right axis) vs size (cell 13)."""
    plt = _mpl()
    sizes = sorted(df["num_nodes"].unique())
    methods = list(dict.fromkeys(df["Algo"]))
    fig, ax1 = plt.subplots(figsize=(8, 4))
    ax2 = ax1.twinx()
    for mi, m in enumerate(methods):
        cong, tau = [], []
        for n in sizes:
            sub = df[(df.Algo == m) & (df.num_nodes == n)]
            cong.append(100 * sub["congest_jobs"].sum()
                        / max(sub["num_jobs"].sum(), 1))
            tau.append(sub["tau"].mean())
        ax1.plot(sizes, cong, f"C{mi}-o", label=f"{m} congestion")
        ax2.plot(sizes, tau, f"C{mi}--s", alpha=0.5, label=f"{m} tau")
    ax1.set_xlabel("number of nodes")
    ax1.set_ylabel("task congestion ratio (%)")
    ax2.set_ylabel(r"$\tau$ (slots)")
    ax2.set_yscale("log")
    ax1.legend(loc="upper left", fontsize=8)
    ax2.legend(loc="upper right", fontsize=8)
    fig.savefig(out, bbox_inches="tight")
    plt.close(fig)


def fig_latency_ratio(df: pd.DataFrame, out: str):
    """Per-instance latency ratio vs baseline, boxplot (cell 16 analog)."""
    plt = _mpl()
    methods = [m for m in dict.fromkeys(df["Algo"]) if m != "baseline"]
    fig, ax = plt.subplots(figsize=(5, 4))
    data = [df[df.Algo == m]["gnn_bl_ratio"].replace(
        [np.inf, -np.inf], np.nan).dropna() for m in methods]
    ax.boxplot(data, tick_labels=methods, showfliers=False)
    ax.axhline(1.0, color="k", lw=0.5)
    ax.set_ylabel("latency ratio vs baseline")
    fig.savefig(out, bbox_inches="tight")
    plt.close(fig)


def training_monitor(df: pd.DataFrame) -> pd.DataFrame:
    """Mean tau per method per graph iteration — the live monitor of
    notebook cell 5 (grouped on the training CSV schema)."""
    return df.groupby(["fid", "method"])["tau"].mean().unstack()


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--csv", required=True, help="Adhoc_test_data_*.csv")
    ap.add_argument("--fig_dir", default="fig")
    args = ap.parse_args(argv)
    df = pd.read_csv(args.csv)
    os.makedirs(args.fig_dir, exist_ok=True)
    base = os.path.splitext(os.path.basename(args.csv))[0]
    fig_instance_scales(df, os.path.join(args.fig_dir,
                                         f"{base}_instance_scales.pdf"))
    fig_congestion_latency(df, os.path.join(args.fig_dir,
                                            f"{base}_congestion.pdf"))
    fig_latency_ratio(df, os.path.join(args.fig_dir, f"{base}_ratio.pdf"))
    print(f"figures written to {args.fig_dir}/")


if __name__ == "__main__":
    main()
