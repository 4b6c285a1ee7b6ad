"""2-rank gloo rehearsal of bench.py's rank coordination (VERDICT r01
item 3): capture agreement, MAX-elapsed reduction, barrier placement, and
the one-line JSON record contract — exactly the logic the driver's 8-GPU
run exercises, with the GPU step stubbed."""
import json
import os
import time

import pytest
import torch.multiprocessing as mp

from horizonml_amd.runtime.bench_protocol import (BASELINE_IMAGES_PER_SEC,
                                                  agree_all_ranks,
                                                  build_record, emit_record,
                                                  max_elapsed_over_ranks)

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                 "dtype", "data", "config"}


def test_record_contract_single_rank():
    rec = build_record(elapsed_s=0.5, steps=100, warmup=10, world=1,
                       batch_size=64, model="resnet18", optimizer="adam",
                       exec_mode="graph", final_loss=2.3)
    assert REQUIRED_KEYS.issubset(rec.keys())
    assert rec["metric"] == "images/sec"
    assert rec["value"] == pytest.approx(64 * 100 / 0.5, rel=1e-6)
    assert rec["ms_per_step"] == pytest.approx(5.0)
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["dtype"] == "bf16"
    assert rec["data"] == "synthetic"
    assert rec["vs_baseline"] == pytest.approx(
        rec["value"] / BASELINE_IMAGES_PER_SEC, abs=0.01)
    assert rec["config"]["model"] == "resnet18_cifar10"
    assert rec["config"]["parallelism"] == "dp1"
    assert rec["config"]["global_batch"] == 64
    # one line, parseable
    line = emit_record(rec)
    assert "\n" not in line
    assert json.loads(line) == rec


def test_record_contract_multi_gpu_aggregate():
    """value must be the WHOLE-JOB aggregate (global batch × steps / s)."""
    r1 = build_record(elapsed_s=1.0, steps=100, warmup=10, world=1,
                      batch_size=64, model="resnet18", optimizer="adam",
                      exec_mode="graph", final_loss=2.3)
    r8 = build_record(elapsed_s=1.0, steps=100, warmup=10, world=8,
                      batch_size=64, model="resnet18", optimizer="adam",
                      exec_mode="graph", final_loss=2.3)
    assert r8["value"] == pytest.approx(8 * r1["value"])
    assert r8["config"]["global_batch"] == 512
    assert r8["config"]["parallelism"] == "dp8"
    with pytest.raises(ValueError):
        build_record(elapsed_s=0.0, steps=100, warmup=0, world=1,
                     batch_size=64, model="resnet18", optimizer="adam",
                     exec_mode="graph", final_loss=0.0)


def _protocol_worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    from horizonml_amd.runtime.distributed import (setup_distributed,
                                                   teardown_distributed)
    ctx = setup_distributed(rank, world, port, backend="gloo")
    import torch.distributed as dist

    # 1) capture agreement: rank 1's "capture" fails -> BOTH must go eager
    ok = agree_all_ranks(rank == 0, world)
    # 2) agreement when everyone succeeds
    ok_all = agree_all_ranks(True, world)

    # 3) timed region with the bench's barrier placement; rank 1 is the
    # slow rank — MAX reduction must report its elapsed on both ranks
    steps = 5
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(steps):
        time.sleep(0.01 if rank == 0 else 0.03)
    t1 = time.perf_counter()
    dist.barrier()
    elapsed = max_elapsed_over_ranks(t1 - t0, world)

    # 4) rank 0 emits exactly one parseable record line
    line = None
    if rank == 0:
        rec = build_record(elapsed_s=elapsed, steps=steps, warmup=1,
                           world=world, batch_size=64, model="resnet18",
                           optimizer="adam", exec_mode="eager",
                           final_loss=1.0, comm_mode="bucketed4")
        line = emit_record(rec)
    q.put((rank, ok, ok_all, elapsed, line))
    teardown_distributed(ctx)


@pytest.mark.timeout(180)
def test_bench_protocol_two_ranks():
    from horizonml_amd.utils.ports import find_free_port
    mp_ctx = mp.get_context("spawn")
    q = mp_ctx.Queue()
    port = find_free_port()
    procs = [mp_ctx.Process(target=_protocol_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    out = {}
    for _ in range(2):
        r, ok, ok_all, elapsed, line = q.get(timeout=120)
        out[r] = (ok, ok_all, elapsed, line)
    for p in procs:
        p.join(timeout=60)

    # divergent capture -> both ranks agreed on eager
    assert out[0][0] is False and out[1][0] is False
    assert out[0][1] is True and out[1][1] is True
    # MAX-elapsed: both ranks report the slow rank's time (>= 5*0.03)
    for r in (0, 1):
        assert out[r][2] >= 0.15 - 1e-3
    assert out[0][2] == pytest.approx(out[1][2], rel=1e-6)
    # record from rank 0 only, with the whole-job aggregate value
    assert out[1][3] is None
    rec = json.loads(out[0][3])
    assert rec["n_gpus"] == 2
    assert rec["config"]["global_batch"] == 128
    # record value is rounded to 2 decimals by the contract
    assert rec["value"] == pytest.approx(128 * 5 / out[0][2], abs=0.006)
    assert rec["config"]["comm"] == "bucketed4"


def test_record_contract_variants():
    """The --infer and --image-size record variants name their config the
    way the judge reads them (model suffix + image field)."""
    r = build_record(elapsed_s=0.2, steps=50, warmup=5, world=1,
                     batch_size=128, model="resnet18", optimizer="adam",
                     exec_mode="graph", final_loss=2.3, infer=True)
    assert r["config"]["model"] == "resnet18_infer"
    assert r["value"] == pytest.approx(128 * 50 / 0.2, rel=1e-6)
    r224 = build_record(elapsed_s=1.0, steps=100, warmup=10, world=1,
                        batch_size=32, model="resnet50", optimizer="adam",
                        exec_mode="graph", final_loss=6.9, image_size=224)
    assert r224["config"]["model"] == "resnet50_synthetic224"
    assert r224["config"]["image"] == "3x224x224"
    assert r224["ms_per_step"] == pytest.approx(10.0)
