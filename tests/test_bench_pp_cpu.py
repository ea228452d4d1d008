"""bench.py --parallelism pp: the async-pipeline benchmark mode (the
reference's headline topology — 1 cluster x N stages) must run end to
end under torchrun on CPU/gloo and emit the driver-contract JSON line.
"""
import json
import os
import subprocess
import sys


def test_bench_pp_two_stage(tmp_path):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    port = 29750 + (os.getpid() % 100)
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           os.path.join(os.path.dirname(os.path.dirname(
               os.path.abspath(__file__))), "bench.py"),
           "--cpu", "--parallelism", "pp", "--steps", "3", "--warmup", "1"]
    out = subprocess.run(cmd, cwd=str(tmp_path), env=env,
                         capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["config"]["parallelism"] == "pp2"
    assert rec["n_gpus"] == 2
    assert rec["value"] > 0
    assert rec["scaling"] == "strong"


def test_bench_dp_two_ranks(tmp_path):
    """The driver's exact multi-GPU launch pattern (torchrun, default dp
    mode) on CPU/gloo: weak-scaling JSON with the whole-job aggregate."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    port = 29850 + (os.getpid() % 100)
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           os.path.join(os.path.dirname(os.path.dirname(
               os.path.abspath(__file__))), "bench.py"),
           "--cpu", "--gpus", "2", "--steps", "3", "--warmup", "1"]
    out = subprocess.run(cmd, cwd=str(tmp_path), env=env,
                         capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["n_gpus"] == 2
    assert rec["scaling"] == "weak"
    # whole-job aggregate: global batch = 2 x micro-batch
    assert rec["config"]["global_batch"] == 8


def test_bench_sp_two_ranks(tmp_path):
    """bench.py --parallelism sp: sequence-parallel GPT over 2 ranks
    (ring attention + SP grad all-reduce) on CPU/gloo."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    port = 29650 + (os.getpid() % 100)
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           os.path.join(os.path.dirname(os.path.dirname(
               os.path.abspath(__file__))), "bench.py"),
           "--cpu", "--model", "gpt-nano", "--parallelism", "sp",
           "--steps", "3", "--warmup", "1"]
    out = subprocess.run(cmd, cwd=str(tmp_path), env=env,
                         capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["config"]["parallelism"] == "sp2"
    assert rec["n_gpus"] == 2
    assert rec["value"] > 0
    assert rec["scaling"] == "strong"
