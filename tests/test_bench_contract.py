"""The driver contract: bench.py must run the distributed path end-to-end
(torch.distributed env vars, one process per rank) and emit one JSON line
with the agreed fields. Runs on CPU over gloo — the same code path the
round-end 8-GPU scale run exercises over RCCL."""
import io
import json
import os
import sys

import numpy as np
import pytest
import torch.multiprocessing as mp


def _run_bench(rank, world, port, argv, results):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), RANK=str(rank),
        LOCAL_RANK=str(rank), WORLD_SIZE=str(world),
    )
    sys.argv = ["bench.py"] + argv
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import bench

    buf = io.StringIO()
    real = sys.stdout
    sys.stdout = buf
    try:
        bench.main()
    finally:
        sys.stdout = real
    results[rank] = buf.getvalue()


@pytest.mark.parametrize("world,extra", [
    (4, []),
    (2, ["--precond", '{"class": "dist_amg", "coarse_enough": 200,'
                      ' "repart_threshold": 400}']),
    (2, ["--weak"]),
])
def test_bench_distributed_contract(world, extra):
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        argv = ["--size", "24", "--steps", "2", "--warmup", "1",
                "--backend", "cpu"] + extra
        procs = [ctx.Process(target=_run_bench,
                             args=(r, world, 29391 + world, argv, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(300)
        for p in procs:
            assert p.exitcode == 0
        out = dict(results)
    # only rank 0 prints; exactly one JSON line
    lines = [ln for ln in out[0].strip().splitlines() if ln.strip()]
    assert len(lines) == 1
    for r in range(1, world):
        assert out[r].strip() == ""
    d = json.loads(lines[0])
    weak = "--weak" in extra
    assert d["n_gpus"] == world
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["scaling"] == ("weak" if weak else "strong")
    assert d["dtype"] == "fp64"
    assert d["higher_is_better"] is False
    assert d["unit"] == "s"
    assert abs(d["ms_per_step"] - d["value"] * 1000) < 1e-9
    if weak:
        assert d["vs_baseline"] is None
        assert d["config"]["unknowns"] == 24 ** 3 * world
    else:
        assert abs(d["vs_baseline"] - d["value"] / 2.03) < 1e-12
        assert d["config"]["unknowns"] == 24 ** 3
    assert d["config"]["true_rel_resid"] < 1e-6
    assert "Poisson 24^3" in d["metric"]


def test_bench_single_process_contract():
    import subprocess

    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    env.pop("RANK", None)
    r = subprocess.run(
        [sys.executable, "bench.py", "--size", "20", "--steps", "1",
         "--warmup", "1", "--backend", "cpu"],
        capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))), env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    d = json.loads(r.stdout.strip().splitlines()[-1])
    assert d["n_gpus"] == 1
    assert d["config"]["true_rel_resid"] < 1e-6
