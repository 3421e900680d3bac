"""The driver's bench contract, CPU plumbing: torchrun spawns N ranks;
in via-ray mode rank 0 alone drives ray + TorchTrainer (worker actors are
the ranks) and prints ONE JSON line; sibling ranks exit 0 untouched.
ANTRAY_BENCH_CPU=1 runs the identical launch shape on CPU/gloo with the
tiny model so this exact path is covered before the driver's first
8-GPU scaling run."""
import json
import os
import subprocess
import sys


def test_via_ray_torchrun_cpu_2ranks():
    env = dict(os.environ, ANTRAY_BENCH_CPU="1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29721", "bench.py", "--gpus", "2",
         "--model", "llama-tiny", "--seq", "64", "--batch", "2",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout  # exactly ONE JSON line
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["via"] == "ray-train"
    assert rec["value"] > 0


def test_bare_mode_flag():
    import bench

    a = bench._parse_args(["--bare", "--steps", "4"])
    assert a.bare and a.steps == 4
