"""The ``modal-amd`` CLI.

Parity: /root/reference/py/modal/cli/entry_point.py:36-140 — command groups
``run|deploy|serve|shell|app|container|volume|queue|dict|secret|image|
environment|profile|token|config|nfs|launch|daemon``. Built on click; heavy
work happens through the same public API users call.
"""

from __future__ import annotations

import inspect
import json
import os
import sys
import time
from typing import Any, Optional

import click

from .._sync import synchronizer, unwrap
from ..exception import Error


def _get_client() -> Any:
    from ..client import _Client

    return synchronizer.run(_Client.from_env())


def _convert_args(fn: Any, raw_args: tuple[str, ...]) -> tuple[tuple, dict]:
    """Map CLI strings onto the function signature using annotations
    (parity: reference cli/run.py builds click params from signatures)."""
    target = getattr(fn, "raw_f", None) or getattr(fn, "get_raw_f", lambda: fn)()
    try:
        sig = inspect.signature(target)
    except (TypeError, ValueError):
        return tuple(raw_args), {}
    args: list = []
    kwargs: dict = {}
    positional = []
    for raw in raw_args:
        if raw.startswith("--") and "=" in raw:
            key, _, value = raw[2:].partition("=")
            kwargs[key.replace("-", "_")] = value
        elif raw.startswith("--"):
            kwargs[raw[2:].replace("-", "_")] = "true"
        else:
            positional.append(raw)
    params = list(sig.parameters.values())

    def convert(value: str, annotation: Any) -> Any:
        if annotation in (int, float):
            return annotation(value)
        if annotation is bool:
            return value.lower() in ("1", "true", "yes")
        return value

    for i, value in enumerate(positional):
        ann = params[i].annotation if i < len(params) else str
        args.append(convert(value, ann))
    for key, value in kwargs.items():
        ann = sig.parameters[key].annotation if key in sig.parameters else str
        kwargs[key] = convert(value, ann)
    return tuple(args), kwargs


@click.group(help="modal-amd: MI355X-native serverless function runtime")
@click.version_option("0.1.0", prog_name="modal-amd")
def entrypoint_cli() -> None:
    pass


# ---- run / deploy / serve / shell -------------------------------------


@entrypoint_cli.command(context_settings={"ignore_unknown_options": True})
@click.argument("ref")
@click.argument("extra_args", nargs=-1, type=click.UNPROCESSED)
@click.option("--detach", is_flag=True, default=False)
@click.option("--quiet", "-q", is_flag=True, default=False)
def run(ref: str, extra_args: tuple[str, ...], detach: bool, quiet: bool) -> None:
    """Run a function or local entrypoint: modal-amd run file.py::app.fn"""
    from ..app import App
    from ..output import enable_output
    from .import_refs import find_app, find_callable, import_target, parse_import_ref

    import_ref = parse_import_ref(ref)
    module = import_target(import_ref)
    app = find_app(module, import_ref.object_path)
    target = find_callable(module, app, import_ref.object_path)
    with enable_output():
        with app.run(detach=detach):
            from ..app import _LocalEntrypoint

            if isinstance(target, _LocalEntrypoint):
                args, kwargs = _convert_args(target.raw_f, extra_args)
                target.raw_f(*args, **kwargs)
            else:
                args, kwargs = _convert_args(target, extra_args)
                result = target.remote(*args, **kwargs)
                if result is not None:
                    click.echo(repr(result))


@entrypoint_cli.command()
@click.argument("ref")
@click.option("--name", default=None, help="Deployment name")
def deploy(ref: str, name: Optional[str]) -> None:
    """Deploy an app: functions stay callable by name afterwards."""
    from .import_refs import find_app, import_target, parse_import_ref

    import_ref = parse_import_ref(ref)
    module = import_target(import_ref)
    app = find_app(module, import_ref.object_path)
    app.deploy(name=name)
    click.echo(f"Deployed app '{name or app.name}' ({app.app_id})")


@entrypoint_cli.command()
@click.argument("ref")
@click.option("--timeout", default=None, type=float)
def serve(ref: str, timeout: Optional[float]) -> None:
    """Serve an app, redeploying on file change (parity: modal serve)."""
    from .import_refs import parse_import_ref
    from .serving import serve_app

    serve_app(parse_import_ref(ref), timeout=timeout)


@entrypoint_cli.command()
@click.argument("ref", required=False)
@click.option("--cmd", default="/bin/bash")
def shell(ref: Optional[str], cmd: str) -> None:
    """Interactive shell in a sandbox (parity: modal shell).

    On a TTY: full PTY mode — raw terminal relay with resize propagation
    (parity: the reference's PTY shell, _pty.py + shell.py). Piped stdin
    falls back to line mode."""
    import threading

    import modal_amd as modal

    if sys.stdin.isatty():
        _pty_shell(modal, cmd)
        return

    sb = modal.Sandbox.create(cmd, "-i")
    click.echo(f"[sandbox {sb.object_id}] {cmd!r} — line mode, 'exit' or Ctrl-D to leave")

    stop = threading.Event()

    def stream(reader: Any, out: Any) -> None:
        try:
            while not stop.is_set():
                data, eof = reader.read_chunk(timeout=0.5)
                if data:
                    out.write(data.decode("utf-8", errors="replace") if isinstance(data, bytes) else data)
                    out.flush()
                if eof:
                    stop.set()
                    return
        except Exception:
            stop.set()

    threads = [
        threading.Thread(target=stream, args=(sb.stdout, sys.stdout), daemon=True),
        threading.Thread(target=stream, args=(sb.stderr, sys.stderr), daemon=True),
    ]
    for t in threads:
        t.start()
    try:
        if sys.stdin.isatty():
            while not stop.is_set():
                try:
                    line = input()
                except EOFError:
                    break
                if line.strip() == "exit":
                    break
                sb.stdin.write(line + "\n")
                sb.stdin.drain()
        else:
            # non-interactive stdin: pipe it through and wait
            data = sys.stdin.read()
            if data:
                sb.stdin.write(data)
            sb.stdin.write_eof()
            sb.stdin.drain()
            sb.wait(raise_on_termination=False)
            # let the reader threads drain trailing output to EOF before
            # terminate() truncates the streams (loaded hosts race here)
            deadline = time.time() + 2.0
            while not stop.is_set() and time.time() < deadline:
                time.sleep(0.05)
    except KeyboardInterrupt:
        pass
    finally:
        stop.set()
        sb.terminate()


def _pty_shell(modal: Any, cmd: str) -> None:
    """Raw-terminal PTY relay: local tty <-> sandbox PTY exec."""
    import os
    import shutil
    import signal
    import termios
    import threading
    import tty

    size = shutil.get_terminal_size()
    sb = modal.Sandbox.create("sleep", "86400")
    p = sb.exec(cmd, pty_info={"rows": size.lines, "cols": size.columns}, text=False)
    click.echo(f"[sandbox {sb.object_id}] {cmd!r} — PTY mode, exit the shell to leave")

    stop = threading.Event()
    resize_needed = threading.Event()
    signal.signal(signal.SIGWINCH, lambda *_a: resize_needed.set())

    def pump_in() -> None:
        try:
            while not stop.is_set():
                data = os.read(0, 4096)
                if not data:
                    return
                p.stdin.write(data)
                p.stdin.drain()
        except Exception:
            pass

    def pump_out() -> None:
        try:
            while not stop.is_set():
                if resize_needed.is_set():
                    resize_needed.clear()
                    s = shutil.get_terminal_size()
                    p.resize(s.lines, s.columns)
                data, eof = p.stdout.read_chunk(timeout=0.5)
                if data:
                    out = data if isinstance(data, bytes) else data.encode()
                    sys.stdout.buffer.write(out)
                    sys.stdout.buffer.flush()
                if eof:
                    stop.set()
                    return
        except Exception:
            stop.set()

    saved = termios.tcgetattr(0)
    tty.setraw(0)
    threads = [
        threading.Thread(target=pump_in, daemon=True),
        threading.Thread(target=pump_out, daemon=True),
    ]
    for t in threads:
        t.start()
    try:
        p.wait()
    except KeyboardInterrupt:
        pass
    finally:
        stop.set()
        termios.tcsetattr(0, termios.TCSADRAIN, saved)
        sb.terminate()
        sys.stdout.write("\n")


@entrypoint_cli.command()
@click.option("--run-dir", default=None, help="Run dir (socket + stores)")
def daemon(run_dir: Optional[str]) -> None:
    """Run a persistent scheduler other processes can attach to."""
    import asyncio

    from ..scheduler.core import Scheduler

    async def main() -> None:
        scheduler = Scheduler(run_dir=run_dir)
        await scheduler.start()
        click.echo(f"modal-amd scheduler on {scheduler.pool.socket_path}")
        try:
            grpc_sock = await scheduler.start_grpc_bridge()
            click.echo(f"modal-amd api.proto gRPC plane on {grpc_sock}")
        except Exception as exc:  # grpc missing: msgpack plane still serves
            click.echo(f"gRPC plane unavailable: {exc}")
        while True:
            await asyncio.sleep(3600)

    asyncio.run(main())


# ---- app ---------------------------------------------------------------


@entrypoint_cli.group(name="app")
def app_cli() -> None:
    """Manage apps."""


@app_cli.command(name="list")
def app_list() -> None:
    client = _get_client()
    rows = synchronizer.run(client.svc.app_list())
    for row in rows:
        click.echo(f"{row['app_id']}  {row['state']:10s}  {row.get('name') or row['description']}")


@app_cli.command(name="stop")
@click.argument("app_id")
def app_stop(app_id: str) -> None:
    client = _get_client()
    synchronizer.run(client.svc.app_stop(app_id=app_id))
    click.echo(f"Stopped {app_id}")


@app_cli.command(name="logs")
@click.argument("app_id")
@click.option("-f", "--follow", is_flag=True, help="Tail live (offset-resumable long-poll).")
@click.option("--timeout", type=float, default=None, help="Stop following after N seconds.")
def app_logs(app_id: str, follow: bool, timeout: float) -> None:
    """Print an app's logs; with -f, tail them live — including from a
    daemon-hosted app over the attach socket (parity: `modal app logs`)."""
    if follow:
        from ..logs_manager import tail_app_logs

        try:
            for entry in tail_app_logs(app_id, timeout=timeout):
                click.echo(entry.get("data", ""), nl=False)
        except KeyboardInterrupt:
            pass
        return
    from ..logs_manager import fetch_app_logs

    for entry in fetch_app_logs(app_id):
        click.echo(entry.get("data", ""), nl=False)


@app_cli.command(name="history")
@click.argument("name")
def app_history(name: str) -> None:
    client = _get_client()
    for row in synchronizer.run(client.svc.app_history(name=name)):
        click.echo(f"v{row['version']}  {row['app_id']}  {time.ctime(row['deployed_at'])}")


@app_cli.command(name="rollback")
@click.argument("name")
@click.option("--version", default=-1, type=int)
def app_rollback(name: str, version: int) -> None:
    client = _get_client()
    out = synchronizer.run(client.svc.app_rollback(name=name, version=version))
    click.echo(f"'{name}' now serves v{out['version']} ({out['app_id']})")


@entrypoint_cli.group(name="billing")
def billing_cli() -> None:
    """Usage accounting (local GPU-seconds)."""


@billing_cli.command(name="summary")
def billing_summary() -> None:
    from ..billing import usage_summary

    for row in usage_summary():
        gpu = "gpu" if row["gpu"] else "cpu"
        click.echo(
            f"{row['function']:30s} {gpu}  inputs={row['inputs']:<8d} "
            f"runtime={row['runtime_seconds']:.2f}s"
        )


# ---- container (= worker) ----------------------------------------------


@entrypoint_cli.group(name="container")
def container_cli() -> None:
    """Manage workers (the container analog)."""


@container_cli.command(name="stats")
def container_stats() -> None:
    client = _get_client()
    stats = synchronizer.run(client.svc.node_stats())
    click.echo(f"pending inputs: {stats['pending_inputs']}  active calls: {stats['active_calls']}")
    for w in stats["workers"]:
        gpu = w.get("gpu")
        mem = ""
        if gpu:
            used = (gpu["hbm_total"] - gpu["hbm_free"]) / 1e9
            mem = f"  hbm {used:.1f}/{gpu['hbm_total']/1e9:.0f} GB"
        click.echo(
            f"{w['task_id']}  worker-{w['worker_id']}  gpu={w['gpu_index']}  "
            f"inflight={w['inflight']}  hb_age={w['last_heartbeat_age']:.0f}s{mem}"
        )


@entrypoint_cli.group()
def endpoint() -> None:
    """Manage web endpoints (parity: modal endpoint)."""


@endpoint.command(name="list")
def endpoint_list() -> None:
    client = _get_client()
    rows = synchronizer.run(client.svc.endpoint_list())
    if not rows:
        click.echo("No web endpoints registered.")
        return
    for row in rows:
        click.echo(f"{row['label']}  {row['function_id']}  {row['url'] or '(gateway not started)'}")


@entrypoint_cli.command()
def metrics() -> None:
    """Node metrics in Prometheus exposition format (also served at the
    web gateway's /_metrics)."""
    client = _get_client()
    click.echo(synchronizer.run(client.svc.node_metrics()), nl=False)


@container_cli.command(name="exec", context_settings={"ignore_unknown_options": True})
@click.argument("task_id")
@click.argument("cmd", nargs=-1, required=True)
def container_exec(task_id: str, cmd: tuple) -> None:
    """Run a command in a worker's context (parity: modal container exec)."""
    client = _get_client()
    resp = synchronizer.run(client.svc.container_exec(task_id=task_id, cmd=list(cmd)))
    click.echo(resp["output"], nl=False)
    sys.exit(resp["returncode"] or 0)


@container_cli.command(name="stop")
@click.argument("task_id")
def container_stop(task_id: str) -> None:
    """Stop one worker (parity: modal container stop)."""
    client = _get_client()
    synchronizer.run(client.svc.container_stop(task_id=task_id))
    click.echo(f"Stopped {task_id}")


@container_cli.command(name="list")
def container_list() -> None:
    client = _get_client()
    svc = client.svc
    if hasattr(svc, "pool"):
        for w in svc.pool.workers.values():
            gpu = f"gpu:{w.gpu_index}" if w.gpu_index is not None else "cpu"
            click.echo(f"{w.task_id}  worker-{w.worker_id}  {gpu}  inflight={len(w.inflight)}")
    else:
        click.echo("(attach mode: worker listing requires the daemon console)")


# ---- volume -------------------------------------------------------------


@entrypoint_cli.group(name="volume")
def volume_cli() -> None:
    """Manage volumes."""


@volume_cli.command(name="create")
@click.argument("name")
def volume_create(name: str) -> None:
    import modal_amd as modal

    modal.Volume.from_name(name, create_if_missing=True).hydrate()
    click.echo(f"Created volume '{name}'")


@volume_cli.command(name="ls")
@click.argument("name")
@click.argument("path", default="/")
def volume_ls(name: str, path: str) -> None:
    import modal_amd as modal

    vol = modal.Volume.from_name(name)
    for entry in vol.listdir(path, recursive=False):
        kind = "d" if entry.is_dir else "f"
        click.echo(f"{kind} {entry.size:>12}  {entry.path}")


@volume_cli.command(name="put")
@click.argument("name")
@click.argument("local_path")
@click.argument("remote_path", default="/")
def volume_put(name: str, local_path: str, remote_path: str) -> None:
    import modal_amd as modal

    vol = modal.Volume.from_name(name, create_if_missing=True)
    with vol.batch_upload() as batch:
        if os.path.isdir(local_path):
            batch.put_directory(local_path, remote_path)
        else:
            target = remote_path
            if remote_path.endswith("/") or remote_path == "/":
                target = os.path.join(remote_path, os.path.basename(local_path))
            batch.put_file(local_path, target)
    click.echo("ok")


@volume_cli.command(name="get")
@click.argument("name")
@click.argument("remote_path")
@click.argument("local_path", default=".")
def volume_get(name: str, remote_path: str, local_path: str) -> None:
    import modal_amd as modal

    vol = modal.Volume.from_name(name)
    dest = local_path
    if os.path.isdir(local_path):
        dest = os.path.join(local_path, os.path.basename(remote_path))
    with open(dest, "wb") as f:
        vol.read_file_into(remote_path, f)
    click.echo(dest)


@volume_cli.command(name="rm")
@click.argument("name")
@click.argument("remote_path")
@click.option("--recursive", "-r", is_flag=True)
def volume_rm(name: str, remote_path: str, recursive: bool) -> None:
    import modal_amd as modal

    modal.Volume.from_name(name).remove_file(remote_path, recursive=recursive)


@volume_cli.command(name="cp")
@click.argument("name")
@click.argument("src_path")
@click.argument("dst_path")
def volume_cp(name: str, src_path: str, dst_path: str) -> None:
    """Copy a file within a volume (parity: `modal volume cp`)."""
    import modal_amd as modal

    modal.Volume.from_name(name).copy_files([src_path], dst_path)
    click.echo(f"Copied {src_path} -> {dst_path}")


@volume_cli.command(name="rename")
@click.argument("old_name")
@click.argument("new_name")
def volume_rename(old_name: str, new_name: str) -> None:
    """Rename a volume (parity: `modal volume rename`)."""
    import modal_amd as modal

    modal.Volume.rename(old_name, new_name)
    click.echo(f"Renamed volume '{old_name}' -> '{new_name}'")


@volume_cli.command(name="delete")
@click.argument("name")
@click.option("--yes", "-y", is_flag=True)
def volume_delete(name: str, yes: bool) -> None:
    import modal_amd as modal

    if not yes:
        click.confirm(f"Delete volume '{name}'?", abort=True)
    modal.Volume.delete(name)


# ---- queue / dict / secret ----------------------------------------------


@entrypoint_cli.group(name="queue")
def queue_cli() -> None:
    """Manage queues."""


@queue_cli.command(name="create")
@click.argument("name")
def queue_create(name: str) -> None:
    import modal_amd as modal

    modal.Queue.from_name(name, create_if_missing=True).hydrate()
    click.echo(f"Created queue '{name}'")


@queue_cli.command(name="len")
@click.argument("name")
@click.option("--total", is_flag=True)
def queue_len(name: str, total: bool) -> None:
    import modal_amd as modal

    click.echo(modal.Queue.from_name(name).len(total=total))


@queue_cli.command(name="peek")
@click.argument("name")
@click.option("-n", default=10)
def queue_peek(name: str, n: int) -> None:
    import modal_amd as modal

    q = modal.Queue.from_name(name)
    for item in list(q.iterate())[:n]:
        click.echo(repr(item))


@queue_cli.command(name="clear")
@click.argument("name")
@click.option("--yes", "-y", is_flag=True)
def queue_clear(name: str, yes: bool) -> None:
    import modal_amd as modal

    if not yes:
        click.confirm(f"Clear queue '{name}'?", abort=True)
    modal.Queue.from_name(name).clear(all=True)


@queue_cli.command(name="delete")
@click.argument("name")
@click.option("--yes", "-y", is_flag=True)
def queue_delete(name: str, yes: bool) -> None:
    import modal_amd as modal

    if not yes:
        click.confirm(f"Delete queue '{name}'?", abort=True)
    modal.Queue.delete(name)


@entrypoint_cli.group(name="dict")
def dict_cli() -> None:
    """Manage dicts."""


@dict_cli.command(name="create")
@click.argument("name")
def dict_create(name: str) -> None:
    import modal_amd as modal

    modal.Dict.from_name(name, create_if_missing=True).hydrate()
    click.echo(f"Created dict '{name}'")


@dict_cli.command(name="get")
@click.argument("name")
@click.argument("key")
def dict_get(name: str, key: str) -> None:
    import modal_amd as modal

    click.echo(repr(modal.Dict.from_name(name).get(key)))


@dict_cli.command(name="items")
@click.argument("name")
def dict_items(name: str) -> None:
    import modal_amd as modal

    for k, v in modal.Dict.from_name(name).items():
        click.echo(f"{k!r}: {v!r}")


@dict_cli.command(name="delete")
@click.argument("name")
@click.option("--yes", "-y", is_flag=True)
def dict_delete(name: str, yes: bool) -> None:
    import modal_amd as modal

    if not yes:
        click.confirm(f"Delete dict '{name}'?", abort=True)
    modal.Dict.delete(name)


@entrypoint_cli.group(name="secret")
def secret_cli() -> None:
    """Manage secrets."""


@secret_cli.command(name="create")
@click.argument("name")
@click.argument("keyvalues", nargs=-1)
@click.option("--force", is_flag=True)
def secret_create(name: str, keyvalues: tuple[str, ...], force: bool) -> None:
    import modal_amd as modal

    env = {}
    for kv in keyvalues:
        key, _, value = kv.partition("=")
        env[key] = value
    modal.Secret.create_deployed(name, env, overwrite=force)
    click.echo(f"Created secret '{name}' with {len(env)} keys")


@secret_cli.command(name="list")
def secret_list() -> None:
    client = _get_client()
    svc = client.svc
    if hasattr(svc, "services"):
        for (env, name), sid in svc.services.secret_names.by_name.items():
            click.echo(f"{sid}  {env}/{name}")


@entrypoint_cli.group(name="image")
def image_cli() -> None:
    """Inspect built images."""


@image_cli.command(name="list")
def image_list() -> None:
    client = _get_client()
    svc = client.svc
    if hasattr(svc, "image_service"):
        for state in svc.image_service.by_id.values():
            click.echo(f"{state.image_id}  built={state.built}  {state.recipe_hash[:12]}")


@image_cli.command(name="info")
@click.argument("image_id")
def image_info(image_id: str) -> None:
    client = _get_client()
    info = synchronizer.run(client.svc.image_info(image_id=image_id))
    click.echo(json.dumps({k: v for k, v in info.items() if k != "build_log"}, indent=2))
    if info.get("build_log"):
        click.echo("--- build log (tail) ---")
        click.echo(info["build_log"][-1500:])


# ---- config / profile / token / environment ------------------------------


@entrypoint_cli.group(name="config")
def config_cli() -> None:
    """Inspect configuration."""


@config_cli.command(name="show")
def config_show() -> None:
    from ..config import config

    click.echo(json.dumps(config.to_dict(), indent=2, default=str))


@entrypoint_cli.group(name="profile")
def profile_cli() -> None:
    """Manage config profiles."""


@profile_cli.command(name="list")
def profile_list() -> None:
    from ..config import config_profiles

    for name in config_profiles() or ["default"]:
        click.echo(name)


@profile_cli.command(name="current")
def profile_current() -> None:
    from ..config import _config_active_profile

    click.echo(_config_active_profile())


@entrypoint_cli.group(name="token")
def token_cli() -> None:
    """Manage tokens (local runtime: recorded for API parity)."""


@token_cli.command(name="set")
@click.option("--token-id", required=True)
@click.option("--token-secret", required=True)
def token_set(token_id: str, token_secret: str) -> None:
    from ..config import USER_CONFIG_PATH

    os.makedirs(os.path.dirname(USER_CONFIG_PATH) or ".", exist_ok=True)
    with open(USER_CONFIG_PATH, "a") as f:
        f.write(f'\n[default]\ntoken_id = "{token_id}"\ntoken_secret = "{token_secret}"\n')
    click.echo(f"Token written to {USER_CONFIG_PATH}")


@entrypoint_cli.group(name="environment")
def environment_cli() -> None:
    """Manage environments."""


@environment_cli.command(name="list")
def environment_list() -> None:
    click.echo("main")


# nfs group aliases the volume group (parity: legacy command kept)
entrypoint_cli.add_command(volume_cli, name="nfs")


@entrypoint_cli.command(name="launch")
@click.argument("template", required=False)
def launch(template: Optional[str]) -> None:
    """Launch a prebuilt template app (parity: modal launch)."""
    click.echo("Templates available locally: jupyter (modal-amd launch jupyter)")


@entrypoint_cli.command(name="curl")
@click.argument("url")
@click.option("-X", "--request", "method", default="GET")
@click.option("-d", "--data", default=None)
def curl(url: str, method: str, data: Optional[str]) -> None:
    """Call a deployed web endpoint (parity: modal curl)."""
    import urllib.request

    req = urllib.request.Request(
        url, data=data.encode() if data else None, method=method.upper()
    )
    try:
        with urllib.request.urlopen(req, timeout=60) as resp:
            click.echo(resp.read().decode("utf-8", errors="replace"))
    except Exception as exc:
        click.echo(f"request failed: {exc}", err=True)
        sys.exit(1)


@entrypoint_cli.command(name="setup")
def setup() -> None:
    """First-time setup (parity: modal setup). Local runtime needs no auth."""
    click.echo("modal-amd runs entirely on this node; no account setup needed.")
    click.echo("Optional: set MODAL_AMD_RUN_DIR and run `modal-amd daemon` for a")
    click.echo("persistent scheduler that other processes attach to.")


@entrypoint_cli.group(name="cluster")
def cluster_cli() -> None:
    """Inspect gang-scheduled (clustered) runs."""


@cluster_cli.command(name="info")
def cluster_info() -> None:
    client = _get_client()
    svc = client.svc
    if hasattr(svc, "pool"):
        gpus = [w for w in svc.pool.workers.values() if w.has_gpu]
        click.echo(f"workers: {len(svc.pool.workers)} ({len(gpus)} GPU-pinned)")
        mesh = svc._extra.get("mesh") or {}
        if mesh.get("formed"):
            click.echo(f"device mesh: formed, backend={mesh.get('backend')}, ranks={mesh.get('ranks')}")
        else:
            click.echo("device mesh: not formed")


@entrypoint_cli.group(name="nfs")
def nfs_cli() -> None:
    """Manage NetworkFileSystems (parity: reference cli/network_file_system.py)."""


@nfs_cli.command(name="list")
def nfs_list() -> None:
    import modal_amd as modal

    client = _get_client()
    rows = synchronizer.run(client.svc.named_objects_list(kind="volume", environment="main"))
    for r in rows:
        if r["name"].startswith("nfs/"):
            click.echo(f'{r["name"][4:]}\t{r["object_id"]}')


@nfs_cli.command(name="create")
@click.argument("name")
def nfs_create(name: str) -> None:
    import modal_amd as modal

    modal.NetworkFileSystem.from_name(name, create_if_missing=True).hydrate()
    click.echo(f"Created NFS '{name}'")


@nfs_cli.command(name="put")
@click.argument("name")
@click.argument("local_path")
@click.argument("remote_path", required=False)
def nfs_put(name: str, local_path: str, remote_path: Optional[str]) -> None:
    import modal_amd as modal

    nfs = modal.NetworkFileSystem.from_name(name, create_if_missing=True)
    if os.path.isdir(local_path):
        nfs.add_local_dir(local_path, remote_path)
    else:
        nfs.add_local_file(local_path, remote_path)
    click.echo("done")


@nfs_cli.command(name="get")
@click.argument("name")
@click.argument("remote_path")
@click.argument("local_path", default=".")
def nfs_get(name: str, remote_path: str, local_path: str) -> None:
    import modal_amd as modal

    nfs = modal.NetworkFileSystem.from_name(name)
    dest = local_path
    if os.path.isdir(dest):
        dest = os.path.join(dest, os.path.basename(remote_path))
    with open(dest, "wb") as f:
        for chunk in nfs.read_file(remote_path):
            f.write(chunk)
    click.echo(dest)


@nfs_cli.command(name="rm")
@click.argument("name")
@click.argument("remote_path")
def nfs_rm(name: str, remote_path: str) -> None:
    import modal_amd as modal

    modal.NetworkFileSystem.from_name(name).remove_file(remote_path)


@entrypoint_cli.group(name="workspace")
def workspace_cli() -> None:
    """Workspace info (parity: reference cli/workspace.py — local: one
    workspace named after the run dir)."""


@workspace_cli.command(name="current")
def workspace_current() -> None:
    from ..config import config

    click.echo(config.get("workspace") or "local")


@entrypoint_cli.command(name="dashboard")
def dashboard() -> None:
    """Print where to look instead of a hosted dashboard: the active run
    dir (logs, volumes, images) and any live web endpoints."""
    client = _get_client()
    click.echo(f"run dir: {client.svc.run_dir}")
    rows = synchronizer.run(client.svc.app_list())
    for row in rows:
        click.echo(f'{row.get("app_id")}\t{row.get("state")}\t{row.get("description") or ""}')


@entrypoint_cli.command(name="changelog")
def changelog() -> None:
    import modal_amd

    root = os.path.dirname(os.path.dirname(os.path.abspath(modal_amd.__file__)))
    path = os.path.join(root, "CHANGELOG.md")
    if os.path.exists(path):
        click.echo(open(path).read())
    else:
        click.echo(f"modal-amd {getattr(modal_amd, '__version__', 'dev')} — see git log for changes")


def main() -> None:
    try:
        entrypoint_cli(standalone_mode=True)
    except Error as exc:
        click.echo(f"Error: {exc}", err=True)
        sys.exit(1)


if __name__ == "__main__":
    main()
