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
                if name == "tensor_parallel":
                    df = runner(world_size, epochs, n, logs,
                                backend=backend, synthetic=synthetic)
                else:
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


def generate_comparison_graphs(results, output_dir, world_size, epochs):
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    sample_sizes = sorted({n for s in STRATEGIES for n in results[s]})
    for n in sample_sizes:
        out = os.path.join(output_dir, str(n),
                           f"worker-{world_size}-epoch-{epochs}")
        os.makedirs(out, exist_ok=True)

        def save(fig, name):
            fig.tight_layout()
            fig.savefig(os.path.join(out, name), dpi=110)
            plt.close(fig)

        # 1/2: accuracy + loss curves (MP: last worker only, like main.py:73)
        for metric, fname in [("accuracy", "accuracy_comparison.png"),
                              ("loss", "loss_comparison.png")]:
            fig, ax = plt.subplots(figsize=(7, 4.5))
            for s in STRATEGIES:
                df = _epoch_rows(results[s].get(n),
                                 last_worker_only=(s == "model_parallel"))
                if df is None:
                    continue
                g = df.groupby("epoch")[metric].mean()
                ax.plot(g.index, g.values, marker="o", label=LABELS[s],
                        color=COLORS[s])
            ax.set_xlabel("Epoch")
            ax.set_ylabel(metric.capitalize())
            ax.set_title(f"{metric.capitalize()} ({n} samples)")
            ax.legend()
            ax.grid(alpha=0.3)
            save(fig, fname)

        # 3: avg epoch time bars
        fig, ax = plt.subplots(figsize=(7, 4.5))
        vals, names, cols = [], [], []
        for s in STRATEGIES:
            df = _epoch_rows(results[s].get(n))
            if df is None:
                continue
            vals.append(df.groupby("epoch")["epoch_time"].mean().mean())
            names.append(LABELS[s])
            cols.append(COLORS[s])
        ax.bar(names, vals, color=cols)
        for i, v in enumerate(vals):
            ax.text(i, v, f"{v:.2f}s", ha="center", va="bottom")
        ax.set_ylabel("Avg epoch time (s)")
        ax.set_title(f"Training time ({n} samples)")
        save(fig, "training_time_comparison.png")

        # 4-6: compute vs comm stacked, per strategy
        for s in STRATEGIES:
            df = _epoch_rows(results[s].get(n))
            if df is None:
                continue
            fig, ax = plt.subplots(figsize=(7, 4.5))
            g = df.groupby("worker")[["compute_time", "comm_time"]].sum()
            ax.bar(g.index.astype(str), g["compute_time"],
                   label="Compute", color="#4c72b0")
            ax.bar(g.index.astype(str), g["comm_time"],
                   bottom=g["compute_time"], label="Comm", color="#dd8452")
            ax.set_xlabel("Worker")
            ax.set_ylabel("Time (s)")
            ax.set_title(f"{LABELS[s]}: compute vs comm ({n} samples)")
            ax.legend()
            save(fig, f"compute_vs_comm_{s}.png")

        # 7: CPU utilization / 8: memory / 9: idle time
        for metric, ylab, fname in [
                ("avg_cpu", "CPU %", "cpu_utilization_comparison.png"),
                ("avg_memory", "Memory (MB)", "memory_usage_comparison.png"),
                ("idle_time", "Idle time (s)", "idle_time_comparison.png")]:
            fig, ax = plt.subplots(figsize=(7, 4.5))
            vals, names, cols = [], [], []
            for s in STRATEGIES:
                df = _epoch_rows(results[s].get(n))
                if df is None:
                    continue
                vals.append(df[metric].mean())
                names.append(LABELS[s])
                cols.append(COLORS[s])
            ax.bar(names, vals, color=cols)
            ax.set_ylabel(ylab)
            ax.set_title(f"{ylab} ({n} samples)")
            save(fig, fname)

    # radar over 6 normalized metrics at the largest sample size (main.py:304)
    n = sample_sizes[-1]
    metrics = ["accuracy", "epoch_time", "compute_time", "comm_time",
               "avg_memory", "idle_time"]
    avail = [s for s in STRATEGIES if _epoch_rows(results[s].get(n)) is not None]
    if len(avail) >= 2:
        import matplotlib.pyplot as plt
        table = {}
        for s in avail:
            df = _epoch_rows(results[s][n],
                             last_worker_only=(s == "model_parallel"))
            table[s] = [df["accuracy"].max(),
                        df.groupby("epoch")["epoch_time"].mean().mean(),
                        df["compute_time"].sum(), df["comm_time"].sum(),
                        df["avg_memory"].mean(), df["idle_time"].sum()]
        arr = np.array([table[s] for s in avail], dtype=float)
        # normalize each metric to [0,1]; time-like metrics inverted (lower
        # is better)
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
        ax.set_title(f"Normalized comparison ({n} samples)")
        ax.legend(loc="lower right", bbox_to_anchor=(1.2, -0.1))
        out = os.path.join(output_dir, str(n),
                           f"worker-{world_size}-epoch-{epochs}")
        os.makedirs(out, exist_ok=True)
        fig.savefig(os.path.join(out, "radar_comparison.png"), dpi=110,
                    bbox_inches="tight")
        plt.close(fig)


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
