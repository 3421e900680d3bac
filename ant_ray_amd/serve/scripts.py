"""`serve` CLI — run/status/shutdown/delete an application.

Role parity: reference python/ray/serve/scripts.py (`serve run
module:app`, `serve status`, `serve shutdown`). The import-path target
follows the reference convention: "pkg.module:app_object" where the
object is a bound Application (Deployment.bind(...)).
"""
from __future__ import annotations

import importlib
import json
import os
import sys
import time

import click


def _load_target(import_path: str):
    if ":" not in import_path:
        raise click.ClickException(
            f"target {import_path!r} must look like 'module.sub:app'")
    mod_name, _, attr = import_path.partition(":")
    sys.path.insert(0, os.getcwd())
    mod = importlib.import_module(mod_name)
    try:
        return getattr(mod, attr)
    except AttributeError:
        raise click.ClickException(
            f"module {mod_name!r} has no attribute {attr!r}") from None


@click.group()
def cli():
    """Manage Ray Serve applications."""


@cli.command()
@click.argument("import_path")
@click.option("--name", default="default", help="Application name.")
@click.option("--route-prefix", default="/", help="HTTP route prefix.")
@click.option("--host", default="127.0.0.1")
@click.option("--port", default=8000, type=int)
@click.option("--blocking/--non-blocking", default=True)
def run(import_path, name, route_prefix, host, port, blocking):
    """Deploy and (by default) block: serve run my_module:app"""
    import ant_ray_amd as ray
    from ant_ray_amd import serve

    if not ray.is_initialized():
        addr = os.environ.get("RAY_ADDRESS")
        ray.init(address=addr) if addr else ray.init()
    app = _load_target(import_path)
    serve.start(http_options={"host": host, "port": port})
    serve.run(app, name=name, route_prefix=route_prefix)
    click.echo(f"application {name!r} deployed at "
               f"http://{host}:{port}{route_prefix}")
    if blocking:
        try:
            while True:
                time.sleep(3600)
        except KeyboardInterrupt:
            click.echo("shutting down")
            serve.shutdown()


@cli.command()
@click.argument("config_file")
def deploy(config_file):
    """Deploy applications from a YAML config (parity: serve deploy).

    Schema (reference multi-app config, reduced):
        applications:
          - name: app1
            route_prefix: /app1
            import_path: my_module:app
    """
    import yaml

    import ant_ray_amd as ray
    from ant_ray_amd import serve

    with open(config_file) as f:
        cfg = yaml.safe_load(f) or {}
    apps = cfg.get("applications") or []
    if not apps:
        raise click.ClickException("config has no applications")
    if not ray.is_initialized():
        addr = os.environ.get("RAY_ADDRESS")
        ray.init(address=addr) if addr else ray.init()
    http = cfg.get("http_options") or {}
    serve.start(http_options=http)
    for app in apps:
        target = _load_target(app["import_path"])
        serve.run(target, name=app.get("name", "default"),
                  route_prefix=app.get("route_prefix", "/"))
        click.echo(f"deployed {app.get('name', 'default')} at "
                   f"{app.get('route_prefix', '/')}")


@cli.command()
def config():
    """Print the currently-deployed applications as a YAML config
    (parity: `serve config` — the inverse of `serve deploy`)."""
    import yaml

    import ant_ray_amd as ray
    from ant_ray_amd import serve

    if not ray.is_initialized():
        try:
            ray.init(address="auto", ignore_reinit_error=True)
        except Exception:
            raise click.ClickException("no running cluster found")
    st = serve.status().get("applications", {})
    apps = []
    for name, info in st.items():
        apps.append({"name": name,
                     "route_prefix": info.get("route_prefix", "/"),
                     "deployments": sorted((info.get("deployments") or {}))})
    click.echo(yaml.safe_dump({"applications": apps}, sort_keys=False))


@cli.command()
@click.argument("import_paths", nargs=-1, required=True)
@click.option("-o", "--output-path", default=None,
              help="write the config YAML here instead of stdout")
def build(import_paths, output_path):
    """Generate a deployable YAML config from application import paths
    (parity: `serve build`)."""
    import yaml

    apps = []
    for i, path in enumerate(import_paths):
        _load_target(path)  # validate it imports and is an Application
        name = f"app{i + 1}" if len(import_paths) > 1 else "default"
        apps.append({"name": name,
                     "route_prefix": "/" if len(import_paths) == 1
                     else f"/{name}",
                     "import_path": path})
    doc = yaml.safe_dump(
        {"http_options": {"host": "127.0.0.1", "port": 8000},
         "applications": apps}, sort_keys=False)
    if output_path:
        with open(output_path, "w") as f:
            f.write(doc)
        click.echo(f"wrote {output_path}")
    else:
        click.echo(doc)


@cli.command()
def status():
    """Show application/deployment status."""
    import ant_ray_amd as ray
    from ant_ray_amd import serve

    if not ray.is_initialized():
        ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
                 ignore_reinit_error=True)
    click.echo(json.dumps(serve.status(), indent=2, default=str))


@cli.command()
@click.argument("name")
def delete(name):
    """Delete one application."""
    import ant_ray_amd as ray
    from ant_ray_amd import serve

    if not ray.is_initialized():
        ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
                 ignore_reinit_error=True)
    serve.delete(name)
    click.echo(f"deleted application {name!r}")


@cli.command()
@click.option("--yes", "-y", is_flag=True, default=False)
def shutdown(yes):
    """Tear down Serve (controller, proxies, all replicas)."""
    import ant_ray_amd as ray
    from ant_ray_amd import serve

    if not yes:
        click.confirm("Shut down Serve and all applications?", abort=True)
    if not ray.is_initialized():
        ray.init(address=os.environ.get("RAY_ADDRESS", "auto"),
                 ignore_reinit_error=True)
    serve.shutdown()
    click.echo("serve shut down")


def main():
    cli()


if __name__ == "__main__":
    main()
