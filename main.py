#!/usr/bin/env python3
"""Benchmark harness: run all three parallelism strategies and emit the
comparison charts.

CLI parity with the reference's ``main.py``
(``--sample_sizes --world_size --epochs --output_dir``); runs DP → MP → TP
per sample size (tolerating a strategy failure like ``main.py:44-55``) and
generates the 8 comparison charts into
``{output_dir}/{sample_size}/worker-{ws}-epoch-{e}/``.
"""
from __future__ import annotations

import argparse
import os
import traceback

import numpy as np

from data_parallel_train import run_data_parallel
from layer_model_parallel_train import run_model_parallel
from tensor_parallel_train import run_tensor_parallel

STRATEGIES = ["data_parallel", "model_parallel", "tensor_parallel"]
LABELS = {"data_parallel": "Data Parallel",
          "model_parallel": "Layer Model Parallel",
          "tensor_parallel": "Tensor Parallel"}
COLORS = {"data_parallel": "#1f77b4", "model_parallel": "#ff7f0e",
          "tensor_parallel": "#2ca02c"}


def run_benchmarks(sample_sizes, world_size, epochs, synthetic=None,
                   backend=None):
    results = {s: {} for s in STRATEGIES}
    for n in sample_sizes:
        for name, runner, logs in [
                ("data_parallel", run_data_parallel, "data_parallel_logs"),
                ("model_parallel", run_model_parallel, "model_parallel_logs"),
                ("tensor_parallel", run_tensor_parallel,
                 "tensor_parallel_logs")]:
            print(f"=== {name} @ {n} samples ===", flush=True)
            try:
                df = runner(world_size, epochs, n, logs,
                            backend=backend, synthetic=synthetic)
            except Exception:  # noqa: BLE001 — strategy tolerance (main.py:44)
                traceback.print_exc()
                df = None
            results[name][n] = df
    return results


def _epoch_rows(df, last_worker_only=False):
    if df is None or len(df) == 0:
        return None
    if last_worker_only:
        df = df[df["worker"] == df["worker"].max()]
    return df


CHART_FILES = ["accuracy_comparison.png", "loss_comparison.png",
               "training_time_comparison.png",
               "compute_vs_comm_comparison.png",
               "cpu_utilization_comparison.png",
               "memory_usage_comparison.png", "idle_time_comparison.png",
               "overall_performance_comparison.png"]


def _emit_chart_set(results, sample_sizes, out, plt):
    """Draw the reference's 8-figure inventory into ``out`` for the given
    sample sizes (one line/bar group per strategy per size — the reference
    draws EVERY size into one figure, ``main.py:69-117``)."""
    os.makedirs(out, exist_ok=True)

    def save(fig, name):
        fig.tight_layout()
        fig.savefig(os.path.join(out, name), dpi=110)
        plt.close(fig)

    # 1/2: accuracy + loss curves, all sizes in one figure (MP: last
    # worker only, like reference main.py:73)
    for metric, fname in [("accuracy", "accuracy_comparison.png"),
                          ("loss", "loss_comparison.png")]:
        fig, ax = plt.subplots(figsize=(9, 6))
        for n in sample_sizes:
            for s in STRATEGIES:
                df = _epoch_rows(results[s].get(n),
                                 last_worker_only=(s == "model_parallel"))
                if df is None:
                    continue
                g = df.groupby("epoch")[metric].mean()
                lbl = (LABELS[s] if len(sample_sizes) == 1
                       else f"{LABELS[s]} ({n} samples)")
                ax.plot(g.index, g.values, marker="o", label=lbl,
                        color=COLORS[s],
                        alpha=1.0 if n == sample_sizes[-1] else 0.55)
        ax.set_xlabel("Epoch")
        ax.set_ylabel("Accuracy (%)" if metric == "accuracy" else "Loss")
        ax.set_title(f"{metric.capitalize()} Comparison: "
                     "Data vs Model vs Tensor Parallel")
        ax.legend()
        ax.grid(True, alpha=0.3)
        save(fig, fname)

    # 3: avg epoch time — grouped bars over sample sizes (main.py:119-149)
    fig, ax = plt.subplots(figsize=(9, 6))
    width = 0.25
    x = np.arange(len(sample_sizes))
    for i, s in enumerate(STRATEGIES):
        vals = []
        for n in sample_sizes:
            df = _epoch_rows(results[s].get(n))
            vals.append(0.0 if df is None
                        else df.groupby("epoch")["epoch_time"].mean().mean())
        ax.bar(x + (i - 1) * width, vals, width, label=LABELS[s],
               color=COLORS[s])
        for xi, v in zip(x + (i - 1) * width, vals):
            if v > 0:
                ax.text(xi, v, f"{v:.2f}s", ha="center", va="bottom",
                        fontsize=8)
    ax.set_xlabel("Sample Size")
    ax.set_ylabel("Average Epoch Time (s)")
    ax.set_title("Training Time Comparison")
    ax.set_xticks(x, [str(n) for n in sample_sizes])
    ax.legend()
    save(fig, "training_time_comparison.png")

    # 4: compute vs comm — one stacked grouped figure (main.py:151-203)
    fig, ax = plt.subplots(figsize=(12, 5))
    for i, s in enumerate(STRATEGIES):
        comp, comm = [], []
        for n in sample_sizes:
            df = _epoch_rows(results[s].get(n))
            comp.append(0.0 if df is None else df["compute_time"].sum())
            comm.append(0.0 if df is None else df["comm_time"].sum())
        xs = x + (i - 1) * width
        ax.bar(xs, comp, width, label=f"{LABELS[s]} compute",
               color=COLORS[s])
        ax.bar(xs, comm, width, bottom=comp,
               label=f"{LABELS[s]} comm", color=COLORS[s], alpha=0.45,
               hatch="//")
    ax.set_xlabel("Sample Size")
    ax.set_ylabel("Cumulative Time (s)")
    ax.set_title("Communication vs Computation Time")
    ax.set_xticks(x, [str(n) for n in sample_sizes])
    ax.legend(fontsize=8)
    save(fig, "compute_vs_comm_comparison.png")

    # 5-7: CPU utilization / memory / idle time (main.py:205-302)
    for metric, ylab, fname in [
            ("avg_cpu", "CPU Utilization (%)",
             "cpu_utilization_comparison.png"),
            ("avg_memory", "Memory Usage (MB)",
             "memory_usage_comparison.png"),
            ("idle_time", "Idle Time (s)", "idle_time_comparison.png")]:
        fig, ax = plt.subplots(figsize=(9, 6))
        for i, s in enumerate(STRATEGIES):
            vals = []
            for n in sample_sizes:
                df = _epoch_rows(results[s].get(n))
                vals.append(0.0 if df is None else df[metric].mean())
            ax.bar(x + (i - 1) * width, vals, width, label=LABELS[s],
                   color=COLORS[s])
        ax.set_xlabel("Sample Size")
        ax.set_ylabel(ylab)
        ax.set_title(f"{ylab} Comparison")
        ax.set_xticks(x, [str(n) for n in sample_sizes])
        ax.legend()
        save(fig, fname)

    # 8: overall performance radar, 6 normalized metrics at the largest
    # size (main.py:304-388)
    n = sample_sizes[-1]
    metrics = ["accuracy", "epoch_time", "compute_time", "comm_time",
               "avg_memory", "idle_time"]
    avail = [s for s in STRATEGIES
             if _epoch_rows(results[s].get(n)) is not None]
    if len(avail) >= 2:
        table = {}
        for s in avail:
            df = _epoch_rows(results[s][n],
                             last_worker_only=(s == "model_parallel"))
            table[s] = [df["accuracy"].max(),
                        df.groupby("epoch")["epoch_time"].mean().mean(),
                        df["compute_time"].sum(), df["comm_time"].sum(),
                        df["avg_memory"].mean(), df["idle_time"].sum()]
        arr = np.array([table[s] for s in avail], dtype=float)
        # normalize each metric to [0,1]; time-like metrics inverted
        # (lower is better)
        norm = np.zeros_like(arr)
        for j in range(arr.shape[1]):
            col = arr[:, j]
            rng = col.max() - col.min()
            v = (col - col.min()) / rng if rng > 0 else np.ones_like(col)
            if metrics[j] != "accuracy":
                v = 1.0 - v
            norm[:, j] = v
        angles = np.linspace(0, 2 * np.pi, len(metrics), endpoint=False)
        angles = np.concatenate([angles, angles[:1]])
        fig, ax = plt.subplots(figsize=(6.5, 6.5),
                               subplot_kw={"projection": "polar"})
        for i, s in enumerate(avail):
            vals = np.concatenate([norm[i], norm[i][:1]])
            ax.plot(angles, vals, label=LABELS[s], color=COLORS[s])
            ax.fill(angles, vals, alpha=0.12, color=COLORS[s])
        ax.set_xticks(angles[:-1])
        ax.set_xticklabels(metrics)
        ax.set_title(f"Overall Performance ({n} samples)")
        ax.legend(loc="lower right", bbox_to_anchor=(1.2, -0.1))
        fig.savefig(os.path.join(out, "overall_performance_comparison.png"),
                    dpi=110, bbox_inches="tight")
        plt.close(fig)


def chart_set_name(results, n) -> str:
    """Published-tree set level (``benchmark_results/{all|datavlayer}/``):
    ``all`` when every strategy produced results, ``datavlayer`` when only
    DP + layer-MP did (the reference's TP-failed runs)."""
    if _epoch_rows(results["tensor_parallel"].get(n)) is None and \
            _epoch_rows(results["data_parallel"].get(n)) is not None:
        return "datavlayer"
    return "all"


def generate_comparison_graphs(results, output_dir, world_size, epochs):
    """Reference figure inventory (``main.py:64-390``: 8 figures, every
    sample size drawn into each) at the top of ``output_dir``, plus the
    published per-run tree ``{output_dir}/{set}/{n}/worker-{ws}-epoch-{e}/``
    with the same 8 figures per sample size
    (``/root/reference/benchmark_results/``)."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    sample_sizes = sorted({n for s in STRATEGIES for n in results[s]})
    if not sample_sizes:
        return
    # top level: the exact output of the reference's flat generate call
    _emit_chart_set(results, sample_sizes, output_dir, plt)
    # published layout: one run dir per sample size under its set
    for n in sample_sizes:
        out = os.path.join(output_dir, chart_set_name(results, n), str(n),
                           f"worker-{world_size}-epoch-{epochs}")
        _emit_chart_set(results, [n], out, plt)


def main():
    ap = argparse.ArgumentParser(description="HorizonML-AMD benchmark suite")
    ap.add_argument("--sample_sizes", type=int, nargs="+",
                    default=[1000, 10000, 50000])
    ap.add_argument("--world_size", type=int, default=5)
    ap.add_argument("--epochs", type=int, default=5)
    ap.add_argument("--output_dir", type=str, default="benchmark_results")
    ap.add_argument("--synthetic", action="store_true", default=None)
    ap.add_argument("--backend", type=str, default=None, nargs="?",
                    choices=[None, "nccl", "gloo"])
    args = ap.parse_args()
    results = run_benchmarks(args.sample_sizes, args.world_size, args.epochs,
                             synthetic=args.synthetic, backend=args.backend)
    generate_comparison_graphs(results, args.output_dir, args.world_size,
                               args.epochs)
    print(f"charts written under {args.output_dir}/")


if __name__ == "__main__":
    main()
