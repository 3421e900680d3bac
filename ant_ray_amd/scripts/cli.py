"""`ray` CLI.

Role parity: reference python/ray/scripts/scripts.py (`ray start` :728,
`ray stop` :1290, status/memory/timeline), python/ray/util/state/state_cli.py
(`ray list ...`), and dashboard/modules/job/cli.py (`ray job submit/...`).
Run as `python -m ant_ray_amd.scripts.cli <cmd>` (a `ray` console script when
pip-installed; tools/ray shim in-tree).
"""
from __future__ import annotations

import json
import os
import sys
import time

import click


@click.group()
def cli():
    """ant-ray-amd cluster CLI."""


# ------------------------------------------------------------------ start/stop


@cli.command()
@click.option("--head", is_flag=True, help="start a head node (GCS + raylet)")
@click.option("--address", default=None, help="GCS address to join as worker node")
@click.option("--num-cpus", type=int, default=None)
@click.option("--num-gpus", type=int, default=None)
@click.option("--port", type=int, default=6379)
@click.option("--object-store-memory", type=int, default=None)
@click.option("--block", is_flag=True, help="stay in the foreground")
def start(head, address, num_cpus, num_gpus, port, object_store_memory, block):
    """Start head or worker node processes on this machine."""
    if head:
        from ant_ray_amd._private.node import start_head

        hp = start_head(num_cpus=num_cpus, num_gpus=num_gpus, port=port,
                        object_store_memory=object_store_memory,
                        owner_pid=0)  # ray start --head: detached
        info = hp.info
        click.echo(f"Started head: GCS at {info['gcs_addr']}")
        click.echo(f"session dir: {info['session_dir']}")
        click.echo("To connect: ray.init(address="
                   f"\"{info['gcs_addr']}\") or RAY_ADDRESS={info['gcs_addr']}")
        if block:
            try:
                hp.proc.wait()
            except KeyboardInterrupt:
                hp.terminate()
        else:
            # detach: the head subprocess keeps running after the CLI exits
            hp.proc.stdout.close()
    elif address:
        import subprocess

        cmd = [sys.executable, "-m", "ant_ray_amd._private.raylet",
               "--gcs", address]
        if num_cpus is not None:
            cmd += ["--num-cpus", str(num_cpus)]
        if num_gpus is not None:
            cmd += ["--num-gpus", str(num_gpus)]
        proc = subprocess.Popen(cmd, start_new_session=True)
        click.echo(f"Started worker raylet (pid {proc.pid}) joining {address}")
        if block:
            proc.wait()
    else:
        raise click.UsageError("pass --head or --address")


@cli.command()
@click.option("--quick", is_flag=True, help="1s per metric instead of 2s")
def microbenchmark(quick):
    """Run the core-runtime microbenchmark (parity: `ray microbenchmark`,
    reference ray_perf.py metric definitions)."""
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    cmd = [sys.executable, os.path.join(root, "tools", "microbench.py")]
    if quick:
        cmd.append("--quick")
    raise SystemExit(subprocess.call(cmd))


@cli.command()
def stop():
    """Stop the most recent local cluster (GCS shutdown broadcast)."""
    import ant_ray_amd as ray

    try:
        ray.init(address="auto", ignore_reinit_error=True)
    except ConnectionError:
        click.echo("no running cluster found")
        return
    from ant_ray_amd._private.worker import global_worker

    cw = global_worker.core_worker
    try:
        cw.io.run(cw.gcs.call("shutdown", {}, timeout=5), timeout=6)
        click.echo("cluster shutdown requested")
    except Exception as e:
        click.echo(f"shutdown failed: {e}")


@cli.command()
def memory():
    """Object-store memory usage per node (parity `ray memory`)."""
    import ant_ray_amd as ray
    from ant_ray_amd.util import state as S

    if not ray.is_initialized():
        ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
                 ignore_reinit_error=True)
    rows = S.store_stats()
    if not rows:
        click.echo("no object store stats available")
        return
    for s in rows:
        used = s.get("used_bytes", 0)
        cap = s.get("arena_size", 0) or 1
        click.echo(f"node {str(s.get('node_id'))[:16]}: "
                   f"{used / (1 << 20):.1f}/{cap / (1 << 20):.1f} MiB used, "
                   f"{s.get('num_objects', 0)} objects, "
                   f"{s.get('total_created', 0)} created, "
                   f"{s.get('total_evicted', 0)} evicted, "
                   f"{s.get('spilled_files', 0)} spilled")


@cli.command()
def status():
    """Cluster resource overview (parity `ray status`)."""
    import ant_ray_amd as ray

    if not ray.is_initialized():
        ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
                 ignore_reinit_error=True)
    total = ray.cluster_resources()
    avail = ray.available_resources()
    nodes = ray.nodes()
    click.echo(f"Nodes: {len(nodes)}")
    click.echo("Resources")
    click.echo("  total:    " + json.dumps({k: total[k] for k in sorted(total)}))
    click.echo("  available:" + json.dumps({k: round(avail.get(k, 0), 2) for k in sorted(total)}))
    try:
        from ant_ray_amd.util.state import _gcs_call

        demands = _gcs_call("pending_resource_demands")
        if demands:
            click.echo(f"Pending demands: {len(demands)} "
                       f"(e.g. {json.dumps(demands[0])})")
    except Exception:
        pass


# ----------------------------------------------------------------- state list


@cli.command("list")
@click.argument("kind", type=click.Choice(
    ["actors", "tasks", "nodes", "workers", "jobs", "placement-groups",
     "cluster-events",
     "objects"]))
@click.option("--limit", type=int, default=100)
def list_cmd(kind, limit):
    """List cluster state (parity `ray list ...` state CLI)."""
    from ant_ray_amd.util import state as S

    fn = {
        "actors": S.list_actors, "tasks": S.list_tasks, "nodes": S.list_nodes,
        "workers": S.list_workers, "jobs": S.list_jobs,
        "placement-groups": S.list_placement_groups, "objects": S.list_objects,
        "cluster-events": S.list_cluster_events,
    }[kind]
    rows = fn(limit=limit)
    if not rows:
        click.echo(f"(no {kind})")
        return
    import tabulate

    cols = list(rows[0].keys())
    click.echo(tabulate.tabulate(
        [[str(r.get(c))[:40] for c in cols] for r in rows], headers=cols))


@cli.command()
@click.option("--output", "-o", default="timeline.json")
def timeline(output):
    """Export task events as a Chrome trace (parity `ray timeline`)."""
    from ant_ray_amd.util.state import get_timeline

    trace = get_timeline()
    with open(output, "w") as f:
        json.dump(trace, f)
    click.echo(f"wrote {len(trace)} events to {output} "
               "(open in chrome://tracing or perfetto)")


@cli.command()
@click.argument("kind", type=click.Choice(["tasks", "actors", "objects"]))
def summary(kind):
    from ant_ray_amd.util import state as S

    if kind == "tasks":
        for row in S.summarize_tasks():
            click.echo(f"{row['name'] or '(anon)':<40} "
                       f"{row['state']:<10} {row['count']}")
    elif kind == "actors":
        for row in S.summarize_actors():
            click.echo(f"{row['state']:<20} {row['count']}")
    else:
        for row in S.summarize_objects():
            click.echo(json.dumps(row))


# ----------------------------------------------------------------------- job


def _latest_session():
    """session_dir of the most recent local cluster (head.json)."""
    import json as _json

    base = "/tmp/antray"
    cands = []
    if os.path.isdir(base):
        for d in os.listdir(base):
            hp = os.path.join(base, d, "head.json")
            if os.path.exists(hp):
                cands.append(hp)
    if not cands:
        return None
    with open(sorted(cands)[-1]) as f:
        return _json.load(f).get("session_dir")


@cli.command()
@click.argument("filename", required=False)
@click.option("--tail", type=int, default=100, help="lines from the end")
def logs(filename, tail):
    """List session log files, or tail one (parity `ray logs`)."""
    sd = _latest_session()
    if not sd:
        click.echo("no running cluster found")
        raise SystemExit(1)
    logs_dir = os.path.join(sd, "logs")
    if not os.path.isdir(logs_dir):
        click.echo("no logs directory")
        return
    if not filename:
        for f in sorted(os.listdir(logs_dir)):
            sz = os.path.getsize(os.path.join(logs_dir, f))
            click.echo(f"{f}\t{sz}")
        return
    fp = os.path.join(logs_dir, os.path.basename(filename))
    if not os.path.isfile(fp):
        click.echo(f"no such log: {filename}")
        raise SystemExit(1)
    with open(fp, errors="replace") as f:
        for line in f.read().splitlines()[-tail:]:
            click.echo(line)


@cli.command("health-check")
def health_check():
    """Exit 0 if the local cluster's GCS answers (parity
    `ray health-check`)."""
    import ant_ray_amd as ray

    try:
        ray.init(address="auto", ignore_reinit_error=True)
        ray.cluster_resources()
        click.echo("ok")
    except Exception as e:
        click.echo(f"unhealthy: {e}")
        raise SystemExit(1)


@cli.command()
def stack():
    """Dump python stacks of this node's runtime processes into the
    session logs (parity `ray stack`; SIGUSR1 → faulthandler — py-spy is
    not in this image)."""
    import signal as _signal
    import time as _time

    import psutil

    me = os.getpid()
    hit = []
    for proc in psutil.process_iter(["pid", "cmdline"]):
        try:
            cmd = " ".join(proc.info["cmdline"] or [])
            if proc.info["pid"] != me and (
                    "_private/head.py" in cmd
                    or "default_worker.py" in cmd):
                proc.send_signal(_signal.SIGUSR1)
                hit.append((proc.info["pid"], cmd.split()[-1][-60:]))
        except (psutil.NoSuchProcess, psutil.AccessDenied):
            continue
    if not hit:
        click.echo("no runtime processes found")
        return
    _time.sleep(0.5)
    for pid, what in hit:
        click.echo(f"signaled {pid} ({what})")
    click.echo("stacks dumped to each process's stderr "
               "(see `ray logs`)")


@cli.command("drain-node")
@click.option("--node-id", default=None,
              help="hex node id (default: every alive node)")
@click.option("--timeout-s", type=float, default=30.0)
def drain_node(node_id, timeout_s):
    """Gracefully drain a raylet: running leases finish, queued work
    spills elsewhere (parity `ray drain-node`)."""
    import asyncio

    import ant_ray_amd as ray
    from ant_ray_amd._private import protocol
    from ant_ray_amd._private.worker import global_worker

    ray.init(address="auto", ignore_reinit_error=True)
    cw = global_worker.core_worker
    nodes = cw.io.run(cw.gcs.call("node_table", {}, timeout=10), timeout=15)
    targets = [n for n in nodes if n.get("alive")
               and (node_id is None or n["node_id"].hex() == node_id)]
    if not targets:
        click.echo("no matching alive node")
        raise SystemExit(1)

    async def _drain(addr):
        conn = await protocol.connect(tuple(addr), None, name="drain-cli")
        return await conn.call("drain", {"timeout_s": timeout_s}, timeout=10)

    for n in targets:
        try:
            r = asyncio.run(_drain(n["addr"]))
            click.echo(f"node {n['node_id'].hex()[:16]}: draining={r.get('draining')}")
        except Exception as e:
            click.echo(f"node {n['node_id'].hex()[:16]}: drain failed: {e}")


@cli.command("disable-usage-stats")
def disable_usage_stats():
    """Persist usage-stats opt-out (parity; no data ever leaves this
    air-gapped deployment either way)."""
    os.makedirs(os.path.expanduser("~/.ray"), exist_ok=True)
    with open(os.path.expanduser("~/.ray/usage_stats.json"), "w") as f:
        f.write('{"usage_stats": false}')
    click.echo("usage stats disabled")


@cli.command("enable-usage-stats")
def enable_usage_stats():
    os.makedirs(os.path.expanduser("~/.ray"), exist_ok=True)
    with open(os.path.expanduser("~/.ray/usage_stats.json"), "w") as f:
        f.write('{"usage_stats": true}')
    click.echo("usage stats enabled (local only: no egress)")


@cli.group()
def job():
    """Job submission (parity `ray job ...`)."""


@job.command()
@click.option("--address", default=None)
@click.option("--working-dir", default=None)
@click.option("--submission-id", default=None)
@click.option("--no-wait", is_flag=True)
@click.argument("entrypoint", nargs=-1, required=True)
def submit(address, working_dir, submission_id, no_wait, entrypoint):
    import ant_ray_amd as ray
    from ant_ray_amd.job_submission import JobSubmissionClient

    ray.init(address=address or os.environ.get("RAY_ADDRESS", "auto"),
             ignore_reinit_error=True)
    client = JobSubmissionClient()
    job_id = client.submit_job(entrypoint=" ".join(entrypoint),
                               submission_id=submission_id,
                               runtime_env={"working_dir": working_dir}
                               if working_dir else None)
    click.echo(f"submitted job {job_id}")
    if no_wait:
        return
    while True:
        st = client.get_job_status(job_id)
        if st in ("SUCCEEDED", "FAILED", "STOPPED"):
            break
        time.sleep(1)
    click.echo(f"job {job_id} finished: {st}")
    click.echo(client.get_job_logs(job_id))
    sys.exit(0 if st == "SUCCEEDED" else 1)


@job.command("list")
def job_list():
    import ant_ray_amd as ray
    from ant_ray_amd.job_submission import JobSubmissionClient

    ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
             ignore_reinit_error=True)
    for j in JobSubmissionClient().list_jobs():
        click.echo(json.dumps(j))


@job.command("status")
@click.argument("job_id")
def job_status(job_id):
    import ant_ray_amd as ray
    from ant_ray_amd.job_submission import JobSubmissionClient

    ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
             ignore_reinit_error=True)
    click.echo(JobSubmissionClient().get_job_status(job_id))


@job.command("stop")
@click.argument("job_id")
def job_stop(job_id):
    import ant_ray_amd as ray
    from ant_ray_amd.job_submission import JobSubmissionClient

    ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
             ignore_reinit_error=True)
    ok = JobSubmissionClient().stop_job(job_id)
    click.echo(f"stopped {job_id}" if ok else f"could not stop {job_id}")


@job.command("logs")
@click.argument("job_id")
def job_logs(job_id):
    import ant_ray_amd as ray
    from ant_ray_amd.job_submission import JobSubmissionClient

    ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
             ignore_reinit_error=True)
    click.echo(JobSubmissionClient().get_job_logs(job_id))


def main():
    cli()


if __name__ == "__main__":
    main()


@cli.group()
def vcluster():
    """Virtual clusters (ant-fork parity: partition nodes into named slices)."""


@vcluster.command("create")
@click.argument("vc_id")
@click.option("--node-count", type=int, default=None)
@click.option("--node-ids", default=None, help="comma-separated hex node ids")
@click.option("--divisible", is_flag=True)
def vcluster_create(vc_id, node_count, node_ids, divisible):
    import ant_ray_amd as ray
    from ant_ray_amd.util import virtual_cluster as vc

    ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
             ignore_reinit_error=True)
    view = vc.create_or_update_virtual_cluster(
        vc_id, node_count=node_count,
        node_ids=node_ids.split(",") if node_ids else None,
        divisible=divisible)
    click.echo(json.dumps(view))


@vcluster.command("remove")
@click.argument("vc_id")
def vcluster_remove(vc_id):
    import ant_ray_amd as ray
    from ant_ray_amd.util import virtual_cluster as vc

    ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
             ignore_reinit_error=True)
    click.echo(json.dumps({"removed": vc.remove_virtual_cluster(vc_id)}))


@vcluster.command("list")
def vcluster_list():
    import ant_ray_amd as ray
    from ant_ray_amd.util import virtual_cluster as vc

    ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
             ignore_reinit_error=True)
    for v in vc.list_virtual_clusters():
        click.echo(json.dumps(v))
