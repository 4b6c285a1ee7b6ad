"""CPU tests for the benchmark harness (main.py) and the legacy env-var
entry (train.py) — chart generation and CSV layout parity."""
import os
import subprocess
import sys

import pandas as pd
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_main_benchmark_and_charts(tmp_path, monkeypatch):
    """run_benchmarks over all three strategies + the 8 comparison charts
    (reference ``main.py:17-61`` / ``:64-390`` parity) on a tiny config."""
    monkeypatch.chdir(tmp_path)
    sys.path.insert(0, REPO)
    from main import generate_comparison_graphs, run_benchmarks
    results = run_benchmarks([64], world_size=2, epochs=1, synthetic=True,
                             backend="gloo")
    for strat in ("data_parallel", "model_parallel", "tensor_parallel"):
        df = results[strat][64]
        assert df is not None, f"{strat} produced no results"
        assert "total_training_time" in df.columns
    out = tmp_path / "charts"
    generate_comparison_graphs(results, str(out), 2, 1)
    chart_dir = out / "64" / "worker-2-epoch-1"
    expected = ["accuracy_comparison.png", "loss_comparison.png",
                "training_time_comparison.png",
                "compute_vs_comm_data_parallel.png",
                "compute_vs_comm_model_parallel.png",
                "compute_vs_comm_tensor_parallel.png",
                "cpu_utilization_comparison.png",
                "memory_usage_comparison.png", "idle_time_comparison.png",
                "radar_comparison.png"]
    for name in expected:
        assert (chart_dir / name).is_file(), f"missing chart {name}"


@pytest.mark.timeout(300)
def test_train_py_env_entry(tmp_path):
    """Legacy Docker entry: env-var rendezvous, single worker, legacy CSV
    schema (reference ``train.py:115-116``)."""
    env = dict(os.environ)
    env.update({"RANK": "0", "WORLD_SIZE": "1",
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29631",
                "EPOCHS": "1", "SAMPLE_SIZE": "32", "BATCH_SIZE": "8",
                "SYNTHETIC": "1", "LOGS_DIR": str(tmp_path),
                "MODEL_TYPE": "resnet",
                "PYTHONPATH": REPO})
    r = subprocess.run([sys.executable, os.path.join(REPO, "train.py")],
                       env=env, capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stderr[-2000:]
    csv = tmp_path / "training_logs_worker_0.csv"
    assert csv.is_file()
    df = pd.read_csv(csv)
    assert list(df.columns) == ["Worker", "Epoch", "Loss", "Accuracy",
                                "Time"]
    assert len(df) == 1
