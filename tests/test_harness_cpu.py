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
    from main import CHART_FILES
    # reference inventory (main.py:64-390): 8 figures at the top level ...
    for name in CHART_FILES:
        assert (out / name).is_file(), f"missing top-level chart {name}"
    # ... plus the published per-run tree
    # benchmark_results/{set}/{n}/worker-{ws}-epoch-{e}/ with the same 8
    chart_dir = out / "all" / "64" / "worker-2-epoch-1"
    for name in CHART_FILES:
        assert (chart_dir / name).is_file(), f"missing chart {name}"


def test_chart_set_name_datavlayer():
    """TP-failed runs land under the datavlayer set, like the reference's
    published tree."""
    import pandas as pd

    from main import chart_set_name
    df = pd.DataFrame({"epoch": [1], "worker": [0], "accuracy": [1.0]})
    results = {"data_parallel": {64: df}, "model_parallel": {64: df},
               "tensor_parallel": {64: None}}
    assert chart_set_name(results, 64) == "datavlayer"
    results["tensor_parallel"][64] = df
    assert chart_set_name(results, 64) == "all"


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
