"""Bounded run of the 4-way stress mix under RPC chaos delays.

Regression for the actor-restart drain-window race: payloads sitting in
_drain_actor_queue's local batch during a restart were pushed with stale
seq numbers at a fresh executor whose ordering gate waited forever —
ray.get never resolved (reference-parity scenario: ActorTaskSubmitter
resubmit-on-restart, reference src/ray/core_worker/transport/
actor_task_submitter.cc). Runs the harness in a subprocess so the
RAY_testing_asio_delay_us spec is parsed fresh.
"""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(180)
def test_stress_mix_chaos_no_unresolved_gets():
    env = dict(os.environ)
    env["RAY_testing_asio_delay_us"] = "*=200:2000"
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "stress_mix.py"),
         "--seconds", "12", "--seed", "11"],
        capture_output=True, text=True, timeout=170, env=env, cwd=REPO)
    out = proc.stderr[-1200:] + "\n--stdout--\n" + proc.stdout[-2500:]
    assert proc.returncode == 0, f"stress mix failed:\n{out}"
    assert "UNRESOLVED" not in proc.stdout
    assert "HUNG" not in proc.stdout
