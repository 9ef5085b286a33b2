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
