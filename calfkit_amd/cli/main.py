"""The ``ck`` developer CLI.

Capability parity (reference README.md:59-66, CHANGELOG.md:45,47,63,81,82,104):
  ck dev mesh start        zero-setup bundled broker
  ck dev run file:agent    host nodes from a module as a worker (detachable)
  ck dev chat              REPL with live agent picker, follows handoffs
  ck dev status|stop|down  daemon management
"""

from __future__ import annotations

import asyncio
import importlib.util
import os
import signal
import subprocess
import sys
from pathlib import Path

import click

DEV_DIR = Path(".calfkit")
DEFAULT_MESH = os.environ.get("CALFKIT_MESH_URL", "127.0.0.1:7321")


def _pidfile(name: str) -> Path:
    DEV_DIR.mkdir(exist_ok=True)
    return DEV_DIR / f"{name}.pid"


def _alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except OSError:
        return False


def _daemons() -> dict[str, int]:
    out = {}
    if DEV_DIR.exists():
        for p in DEV_DIR.glob("*.pid"):
            try:
                pid = int(p.read_text().split()[0])
            except (ValueError, IndexError):
                continue
            if _alive(pid):
                out[p.stem] = pid
            else:
                p.unlink(missing_ok=True)
    return out


@click.group()
def cli() -> None:
    """calfkit-amd developer tool."""
    from ..utils.logging import setup_logging

    # CALF_LOG_JSON=1 turns every calfkit log record into one JSON line
    # (structured logging; the step ledger rides the same stream)
    setup_logging(json_lines=os.environ.get("CALF_LOG_JSON", "") == "1")


@cli.group()
def dev() -> None:
    """Local development loop."""


# ------------------------------------------------------------------- mesh

@dev.group()
def mesh() -> None:
    """The bundled mesh broker."""


@mesh.command("start")
@click.option("--port", default=7321, show_default=True)
@click.option("--detach/--no-detach", default=False)
def mesh_start(port: int, detach: bool) -> None:
    if detach:
        _spawn_daemon("mesh", [sys.executable, "-m", "calfkit_amd.cli",
                               "dev", "mesh", "start", "--port", str(port)])
        click.echo(f"mesh broker detached on 127.0.0.1:{port}")
        return

    async def run() -> None:
        from ..adapters.mesh import MeshBroker

        broker = await MeshBroker(port=port).start()
        click.echo(f"mesh broker on {broker.address} (ctrl-c to stop)")
        await broker.serve_forever()

    asyncio.run(run())


def _spawn_daemon(name: str, argv: list[str]) -> None:
    DEV_DIR.mkdir(exist_ok=True)
    log = open(DEV_DIR / f"{name}.log", "ab")
    proc = subprocess.Popen(argv, stdout=log, stderr=log,
                            start_new_session=True)
    _pidfile(name).write_text(f"{proc.pid} {' '.join(argv)}\n")


async def _ensure_mesh(bootstrap: str) -> None:
    """Start a local broker when nothing answers at the bootstrap address."""
    from ..adapters.mesh.client import _MeshConnection

    host, _, port = bootstrap.rpartition(":")
    try:
        conn = _MeshConnection(host or "127.0.0.1", int(port))
        await asyncio.wait_for(conn.open(), 2)
        await conn.close()
        return
    except OSError:
        pass
    _spawn_daemon("mesh", [sys.executable, "-m", "calfkit_amd.cli", "dev",
                           "mesh", "start", "--port", port])
    for _ in range(50):
        await asyncio.sleep(0.1)
        try:
            conn = _MeshConnection(host or "127.0.0.1", int(port))
            await asyncio.wait_for(conn.open(), 2)
            await conn.close()
            return
        except OSError:
            continue
    raise click.ClickException("could not start the local mesh broker")


# -------------------------------------------------------------------- run

def _load_nodes(spec: str) -> list:
    """``file.py:agent`` or ``file.py`` (every hostable top-level object)."""
    path, _, name = spec.partition(":")
    module_path = Path(path)
    if not module_path.exists():
        raise click.ClickException(f"no such file: {path}")
    mod_spec = importlib.util.spec_from_file_location(module_path.stem,
                                                      module_path)
    module = importlib.util.module_from_spec(mod_spec)
    sys.modules[module_path.stem] = module
    mod_spec.loader.exec_module(module)
    if name:
        obj = getattr(module, name, None)
        if obj is None:
            raise click.ClickException(f"{path} has no object {name!r}")
        return [obj]
    return [v for v in vars(module).values()
            if not isinstance(v, type)
            and getattr(v, "kind", None) in ("agent", "tool", "toolbox")]


@dev.command("run")
@click.argument("spec")
@click.option("--mesh", "mesh_url", default=DEFAULT_MESH, show_default=True)
@click.option("--detach/--no-detach", default=False)
@click.option("--gpu", type=int, default=None, help="pin this worker to a GPU")
@click.option("--mcp", "mcp_json", type=click.Path(exists=True), default=None,
              help="also host every toolbox from this mcp.json")
def dev_run(spec: str, mesh_url: str, detach: bool, gpu: int | None,
            mcp_json: str | None) -> None:
    """Host the nodes of SPEC (file.py[:name]) as a worker on the mesh."""
    if detach:
        argv = [sys.executable, "-m", "calfkit_amd.cli", "dev", "run", spec,
                "--mesh", mesh_url]
        if gpu is not None:
            argv += ["--gpu", str(gpu)]
        if mcp_json:
            argv += ["--mcp", mcp_json]
        _spawn_daemon(f"worker-{Path(spec.partition(':')[0]).stem}", argv)
        click.echo(f"worker for {spec} detached")
        return

    async def run() -> None:
        await _ensure_mesh(mesh_url)
        from ..adapters.mesh import connect_mesh
        from ..worker import Worker

        nodes = _load_nodes(spec)
        if mcp_json:
            from ..utils.mcp_config import load_mcp_config

            nodes.extend(load_mcp_config(mcp_json))
        if not nodes:
            raise click.ClickException(f"{spec}: nothing hostable found")
        bus, directory = await connect_mesh(mesh_url)
        worker = Worker(*nodes, bus=bus, directory=directory, gpu=gpu)
        click.echo(f"hosting {[getattr(n, 'name', '?') for n in nodes]} "
                   f"on {mesh_url}")
        await worker.run()

    asyncio.run(run())


# ------------------------------------------------------------------- chat

@dev.command("chat")
@click.option("--mesh", "mesh_url", default=DEFAULT_MESH, show_default=True)
@click.option("--agent", "agent_name", default="", help="skip the picker")
def dev_chat(mesh_url: str, agent_name: str) -> None:
    """Interactive REPL against mesh agents; follows handoffs across turns."""

    async def run() -> None:
        from ..adapters.mesh import connect_mesh
        from ..client import Client

        await _ensure_mesh(mesh_url)
        bus, directory = await connect_mesh(mesh_url)
        client = await Client.connect(bus, directory)
        view = await client.mesh()
        agents = sorted(a.node for a in view.agents())
        target = agent_name
        if not target:
            if not agents:
                raise click.ClickException("no agents on the mesh — "
                                           "`ck dev run <file>:<agent>` first")
            click.echo("agents on the mesh:")
            for i, a in enumerate(agents):
                click.echo(f"  [{i}] {a}")
            idx = click.prompt("pick", type=int, default=0)
            target = agents[idx]
        thread_id = None
        click.echo(f"chatting with {target!r} (ctrl-d to exit)")
        while True:
            try:
                line = input(f"{target}> ")
            except EOFError:
                break
            if not line.strip():
                continue
            tid = thread_id or __import__("uuid").uuid4().hex

            async def stream() -> None:
                # live step events while the turn runs (text deltas, tool
                # calls, handoffs) — the reference's wait-progress
                async for chunk in client.firehose(tid):
                    if chunk.event == "text_delta":
                        click.echo(chunk.text, nl=False)
                    elif chunk.event == "tool_call":
                        click.echo(f"\n[{chunk.node} -> "
                                   f"{chunk.data.get('tool_name')}"
                                   f"({chunk.data.get('arguments')})]")
                    elif chunk.event == "handoff":
                        click.echo(f"\n[handoff -> {chunk.data.get('to')}]")

            watcher = asyncio.ensure_future(stream())
            try:
                result = await client.execute(target, line, thread_id=tid,
                                              timeout=600)
            finally:
                watcher.cancel()
            thread_id = result.thread_id or tid  # handoffs keep the thread
            click.echo("")  # newline after the streamed text

    asyncio.run(run())


# ----------------------------------------------------------------- daemons

@dev.command("status")
def dev_status() -> None:
    ds = _daemons()
    if not ds:
        click.echo("no calfkit daemons running")
        return
    for name, pid in sorted(ds.items()):
        click.echo(f"{name:24s} pid={pid}")


@dev.command("stop")
@click.argument("name")
def dev_stop(name: str) -> None:
    ds = _daemons()
    if name not in ds:
        raise click.ClickException(f"no daemon named {name!r}")
    os.kill(ds[name], signal.SIGTERM)
    _pidfile(name).unlink(missing_ok=True)
    click.echo(f"stopped {name}")


@dev.command("down")
def dev_down() -> None:
    for name, pid in _daemons().items():
        try:
            os.kill(pid, signal.SIGTERM)
        except OSError:
            pass
        _pidfile(name).unlink(missing_ok=True)
        click.echo(f"stopped {name}")


def run() -> None:  # console-script entry
    cli(prog_name="ck")


if __name__ == "__main__":
    run()
