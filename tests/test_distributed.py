"""Multi-process DDP tests over gloo (CPU, world_size=2).

Exercises the bucketed all-reduce wrapper (maskrcnn_amd.distributed) the
same way bench.py uses it on RCCL: subprocess workers (tests/ddp_worker.py),
broadcast init, backward -> finalize -> identical averaged grads on every
rank.
"""

import json
import os
import socket
import subprocess
import sys

import pytest

_WORKER = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                       "ddp_worker.py")


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _run_world(mode, world=2, timeout=240):
    env = dict(os.environ,
               MASTER_ADDR="127.0.0.1", MASTER_PORT=str(_free_port()))
    procs = [subprocess.Popen(
        [sys.executable, _WORKER, mode, str(r), str(world)],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE, env=env,
        text=True) for r in range(world)]
    results = {}
    for r, p in enumerate(procs):
        try:
            out, err = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            for pp in procs:
                pp.kill()
            raise
        assert p.returncode == 0, f"rank {r} failed:\n{err[-3000:]}"
        results[r] = json.loads(out.strip().splitlines()[-1])
    return results


@pytest.mark.timeout(300)
def test_bucketed_allreduce_two_ranks():
    res = _run_world("linear")
    # Broadcast worked: same initial weights despite different seeds.
    assert res[0]["w0_sum"] == pytest.approx(res[1]["w0_sum"], rel=1e-7)
    # All-reduce worked: identical averaged grads despite different data.
    assert res[0]["grad_sum"] == pytest.approx(res[1]["grad_sum"],
                                               rel=1e-6)
    assert res[0]["grad_head"] == pytest.approx(res[1]["grad_head"],
                                                rel=1e-5)
    assert res[0]["grad_sum"] > 0


@pytest.mark.timeout(600)
def test_maskrcnn_ddp_step_two_ranks():
    res = _run_world("maskrcnn", timeout=420)
    assert res[0]["grad_sum"] == pytest.approx(res[1]["grad_sum"],
                                               rel=1e-5)
    assert res[0]["grad_sum"] > 0


def test_bench_script_two_proc_gloo(tmp_path):
    """End-to-end insurance for the driver's SCALE run: launch bench.py
    exactly as the driver does (torch.distributed.run, one proc per
    'GPU') on CPU/gloo with a shrunk config and validate the rank-0
    JSON contract line."""
    import json
    import subprocess
    import sys

    port = _free_port()
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(port), "bench.py", "--gpus", "2",
           "--steps", "1", "--warmup", "0", "--batch", "1",
           "--image-size", "256", "--backbone", "resnet50"]
    proc = subprocess.run(
        cmd, capture_output=True, text=True, timeout=900,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert proc.returncode == 0, proc.stderr[-2000:]
    json_lines = [ln for ln in proc.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, proc.stdout[-1000:]
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 2
    assert rec["metric"] == "train_images_per_sec"
    assert rec["value"] > 0
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 2


def test_train_cli_one_epoch(tmp_path):
    """tools/train.py end to end on CPU (shapes, 2 steps, heads) —
    writes an epoch checkpoint + metrics.csv + train_state.pth."""
    import glob
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.run(
        [sys.executable, "tools/train.py", "--dataset", "shapes",
         "--epochs", "1", "--steps", "2", "--layers", "heads",
         "--num-workers", "0", "--logs", str(tmp_path)],
        capture_output=True, text=True, timeout=1200, cwd=root)
    assert proc.returncode == 0, proc.stderr[-2000:]
    ckpts = glob.glob(str(tmp_path / "*" / "mask_rcnn_*_0000.pth"))
    assert len(ckpts) == 1, ckpts
    assert glob.glob(str(tmp_path / "*" / "metrics.csv"))
    assert glob.glob(str(tmp_path / "*" / "train_state.pth"))


@pytest.mark.timeout(900)
def test_staged_training_ddp_two_ranks():
    """heads -> all staged fits with DDP rebuild between stages: ranks
    stay weight-synchronized and no bucket double-counts."""
    res = _run_world("staged", timeout=800)
    assert res[0]["epochs"] == res[1]["epochs"] == 2
    assert res[0]["losses_finite"] and res[1]["losses_finite"]
    assert res[0]["w_head"] == pytest.approx(res[1]["w_head"],
                                             abs=1e-6)


def test_train_cli_resume_from_last(tmp_path):
    """tools/train.py --model last resumes epoch numbering and restores
    optimizer state (train_state.pth round trip through the CLI)."""
    import glob
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    def run(extra):
        return subprocess.run(
            [sys.executable, "tools/train.py", "--dataset", "shapes",
             "--steps", "1", "--layers", "heads", "--num-workers", "0",
             "--logs", str(tmp_path)] + extra,
            capture_output=True, text=True, timeout=1200, cwd=root)

    r = run(["--epochs", "1"])
    assert r.returncode == 0, r.stderr[-2000:]
    r = run(["--epochs", "2", "--model", "last"])
    assert r.returncode == 0, r.stderr[-2000:]
    ckpts = sorted(glob.glob(str(tmp_path / "*" / "mask_rcnn_*.pth")))
    assert any(c.endswith("0001.pth") for c in ckpts), ckpts


def test_shapes_sample_cli(tmp_path):
    """samples/shapes.py train -> detect end to end (the reference's
    train_shapes notebook flow as a script)."""
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "samples/shapes.py", "train", "--epochs", "1",
         "--steps", "1", "--layers", "heads", "--logs", str(tmp_path)],
        capture_output=True, text=True, timeout=1200, cwd=root)
    assert r.returncode == 0, r.stderr[-2000:]
    out_png = str(tmp_path / "det.png")
    r = subprocess.run(
        [sys.executable, "samples/shapes.py", "detect", "--model",
         "last", "--logs", str(tmp_path), "--out", out_png],
        capture_output=True, text=True, timeout=1200, cwd=root)
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(out_png)


def test_bench_script_single_process(tmp_path):
    """The driver's N=1 invocation shape (plain python, no torchrun)."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "1",
         "--warmup", "0", "--batch", "1", "--image-size", "256",
         "--backbone", "resnet50"],
        capture_output=True, text=True, timeout=900, cwd=root)
    assert proc.returncode == 0, proc.stderr[-2000:]
    rec = json.loads([ln for ln in proc.stdout.splitlines()
                      if ln.startswith("{")][0])
    assert rec["n_gpus"] == 1 and rec["value"] > 0
    assert rec["data"] == "synthetic"


@pytest.mark.timeout(300)
def test_no_sync_gradient_accumulation():
    """no_sync() accumulates locally; the synced backward reduces the
    full accumulated grad exactly once (torch-DDP semantics)."""
    res = _run_world("accum")
    for r in res.values():
        assert r["match"], res
    assert res[0]["grad_sum"] == pytest.approx(res[1]["grad_sum"],
                                               rel=1e-6)
