"""`kt` CLI (reference parity: python_client/kubetorch/cli.py — the same
command surface, MI355X control plane underneath)."""
import importlib
import importlib.util
import json
import os
import sys

import typer
from rich.console import Console
from rich.table import Table

app = typer.Typer(help="kubetorch_amd: MI355X-native serverless ML dispatch")


@app.callback(invoke_without_command=True)
def _root(ctx: typer.Context,
          version: bool = typer.Option(False, "--version", "-V")):
    if version:
        import kubetorch_amd

        console.print(f"kubetorch_amd {kubetorch_amd.__version__}")
        raise typer.Exit()
    if ctx.invoked_subcommand is None:
        console.print(ctx.get_help())
        raise typer.Exit()
console = Console()

secrets_app = typer.Typer(help="manage secrets")
volumes_app = typer.Typer(help="manage volumes")
server_app = typer.Typer(help="run framework services")
app.add_typer(secrets_app, name="secrets")
app.add_typer(volumes_app, name="volumes")
app.add_typer(server_app, name="server")


def _load_module_file(path):
    path = os.path.abspath(path)
    sys.path.insert(0, os.path.dirname(path))
    spec = importlib.util.spec_from_file_location(
        os.path.splitext(os.path.basename(path))[0], path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


@app.command()
def check():
    """Cluster/controller doctor: connectivity, driver, GPU visibility."""
    from kubetorch_amd.globals import controller_client

    try:
        cc = controller_client()
        import httpx

        h = httpx.get(cc.base_url + "/health", timeout=10).json()
        console.print(f"[green]controller ok[/green] at {cc.base_url} "
                      f"(driver: {h.get('driver')})")
    except Exception as e:
        console.print(f"[red]controller unreachable:[/red] {e}")
        raise typer.Exit(1)
    try:
        import torch

        n = torch.cuda.device_count() if torch.cuda.is_available() else 0
        console.print(f"local GPUs visible: {n}")
    except Exception:
        console.print("torch not importable locally")
    from kubetorch_amd.ops import hip_available

    console.print(f"gfx950 HIP extension built: {hip_available()}")
    import os as _os

    store = _os.environ.get("KT_STORE_URL")
    if store:
        try:
            import httpx

            httpx.get(f"{store}/health", timeout=5).raise_for_status()
            console.print(f"[green]data store ok[/green] at {store}")
        except Exception as e:
            console.print(f"[yellow]data store unreachable:[/yellow] {e}")
    else:
        console.print("data store: local-dir mode (KT_STORE_URL unset)")


@app.command()
def config(key: str = typer.Argument(None), value: str = typer.Argument(None),
           persist: bool = typer.Option(False, "--persist")):
    """Show or set client config."""
    from kubetorch_amd.config import config as cfg

    if key is None:
        console.print_json(json.dumps(cfg.as_dict()))
    elif value is None:
        console.print(str(cfg.get(key)))
    else:
        cfg.set(key, value, persist=persist)
        console.print(f"set {key}={value}")


@app.command()
def deploy(target: str,
           workers: int = typer.Option(None, help="override workers")):
    """Deploy decorated modules from a file: `kt deploy train.py[:fn]`."""
    path, _, symbol = target.partition(":")
    mod = _load_module_file(path)
    from kubetorch_amd.resources.decorators import PartialModule

    partials = (
        {symbol: getattr(mod, symbol)} if symbol
        else {k: v for k, v in vars(mod).items()
              if isinstance(v, PartialModule)}
    )
    if not partials:
        console.print("[red]no @kt.compute-decorated callables found[/red]")
        raise typer.Exit(1)
    mods = []
    for name, pm in partials.items():
        if not isinstance(pm, PartialModule):
            console.print(f"[red]{name} is not decorated[/red]")
            raise typer.Exit(1)
        m = pm.build_module()
        if workers and m.compute.distributed_config:
            m.compute.distributed_config["workers"] = workers
            m.compute.replicas = workers
        mods.append(m)
    # parallel deploy (reference: kt deploy launches modules concurrently)
    from concurrent.futures import ThreadPoolExecutor

    with ThreadPoolExecutor(max_workers=min(8, len(mods))) as ex:
        futs = {ex.submit(m.to, m.compute): m for m in mods}
        errs = []
        for fut, m in futs.items():
            try:
                fut.result()
                console.print(f"[green]deployed[/green] {m.name}")
            except Exception as e:  # noqa: BLE001
                errs.append((m.name, e))
                console.print(f"[red]failed[/red] {m.name}: {e}")
    if errs:
        raise typer.Exit(1)


@app.command()
def call(service: str, method: str = typer.Argument(None),
         args_json: str = typer.Option("[]", "--args"),
         kwargs_json: str = typer.Option("{}", "--kwargs")):
    """Call a deployed service: `kt call my-fn --args '[1,2]'`."""
    from kubetorch_amd.client.http_client import HTTPClient
    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import controller_client, service_url

    w = controller_client().get_workload(service, cfg.namespace)
    if not w:
        console.print(f"[red]no workload {service!r}[/red]")
        raise typer.Exit(1)
    pods = w.get("pods") or []
    name = w["metadata"].get("callable_name", service)
    client = HTTPClient(service_url(service, cfg.namespace, pods), name)
    result = client.call(args=json.loads(args_json),
                         kwargs=json.loads(kwargs_json), method=method,
                         serialization="json")
    console.print_json(json.dumps({"result": result}))


@app.command("list")
def list_cmd(namespace: str = typer.Option(None, "-n")):
    """List workloads."""
    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import controller_client

    ns = namespace or cfg.namespace
    ws = controller_client().list_workloads(ns).get("workloads", [])
    t = Table("name", "namespace", "kind", "launch_id", "module")
    for w in ws:
        t.add_row(w["name"], w["namespace"],
                  w.get("service_config", {}).get("kind", "?"),
                  str(w.get("launch_id")),
                  w.get("metadata", {}).get("callable_name", ""))
    console.print(t)


@app.command()
def describe(service: str, namespace: str = typer.Option(None, "-n")):
    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import controller_client

    ns = namespace or cfg.namespace
    w = controller_client().get_workload(service, ns)
    if not w:
        console.print("[red]not found[/red]")
        raise typer.Exit(1)
    console.print_json(json.dumps(w, default=str))
    try:
        evs = controller_client().service_events(service, ns)[-10:]
    except Exception:
        evs = []
    if evs:
        console.print("\n[bold]recent events[/bold]")
        for e in evs:
            tag = "!" if e.get("type") == "Warning" else "·"
            console.print(f"  {tag} {e.get('reason', '')} "
                          f"{e.get('pod') or ''}: {e.get('message', '')}")


@app.command()
def teardown(service: str = typer.Argument(None),
             prefix: str = typer.Option(None, "-p"),
             namespace: str = typer.Option(None, "-n")):
    """Tear down a workload (or all with a name prefix)."""
    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import controller_client

    ns = namespace or cfg.namespace
    cc = controller_client()
    if prefix:
        names = [w["name"] for w in cc.list_workloads(ns).get("workloads", [])
                 if w["name"].startswith(prefix)]
    elif service:
        names = [service]
    else:
        console.print("[red]pass a service or -p prefix[/red]")
        raise typer.Exit(1)
    for n in names:
        cc.delete_workload(n, ns)
        console.print(f"deleted {n}")


@app.command()
def logs(service: str, namespace: str = typer.Option(None, "-n"),
         limit: int = typer.Option(100),
         follow: bool = typer.Option(False, "--follow", "-f",
                                     help="keep polling for new lines")):
    import time as _time

    from kubetorch_amd.client.http_client import HTTPClient
    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import controller_client, service_url

    ns = namespace or cfg.namespace
    w = controller_client().get_workload(service, ns)
    if not w:
        raise typer.Exit(1)
    client = HTTPClient(service_url(service, ns, w.get("pods")), service)
    since = 0
    while True:
        entries = client.logs(since=since, limit=limit)
        for e in entries:
            console.print(f"[dim]{e['source']}[/dim] {e['line']}")
            since = max(since, e.get("seq", since) + 1)
        if not follow:
            break
        _time.sleep(1.0)


@app.command()
def run(command: str, name: str = typer.Option("app"),
        cpus: str = typer.Option(None), gpus: int = typer.Option(0),
        follow: bool = typer.Option(False, "--follow", "-f",
                                    help="block until the app exits, then "
                                         "tear it down"),
        timeout: int = typer.Option(3600)):
    """Deploy an arbitrary command as an app: `kt run 'python serve.py'`."""
    import kubetorch_amd as kt

    a = kt.app(command, name=name)
    a.to(kt.Compute(cpus=cpus, gpus=gpus))
    console.print(f"[green]running[/green] {a.name}")
    if follow:
        # wait() streams the app's stdout/stderr while blocking (app-mode
        # pods run the user command directly — no HTTP server to query
        # after exit)
        done = a.wait(timeout=timeout, printer=console.print)
        a.teardown()
        if not done:
            console.print(f"[red]timed out after {timeout}s[/red]")
            raise typer.Exit(1)
        console.print("[green]app finished[/green]")


@app.command()
def apply(manifest_path: str, namespace: str = typer.Option(None, "-n")):
    """Apply a raw manifest through the controller."""
    import yaml

    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import controller_client

    with open(manifest_path) as f:
        manifest = yaml.safe_load(f)
    name = manifest["metadata"]["name"]
    resp = controller_client().deploy(
        name=name, namespace=namespace or cfg.namespace, manifest=manifest)
    console.print_json(json.dumps(resp))


@app.command()
def put(key: str, src: str):
    from kubetorch_amd.data_store import commands as ds

    console.print_json(json.dumps(ds.put(key, src=src)))


@app.command()
def get(key: str, dest: str = typer.Argument(None)):
    from kubetorch_amd.data_store import commands as ds

    console.print(str(ds.get(key, dest=dest)))


@app.command()
def ls(prefix: str = typer.Argument("")):
    from kubetorch_amd.data_store import commands as ds

    t = Table("key", "size")
    for e in ds.ls(prefix):
        t.add_row(e["key"], str(e["size"]))
    console.print(t)


@app.command()
def rm(key: str):
    from kubetorch_amd.data_store import commands as ds

    ds.rm(key)
    console.print(f"removed {key}")


@app.command()
def workload(service: str, namespace: str = typer.Option(None, "-n")):
    describe(service, namespace)


@app.command()
def debug(service: str, port: int = typer.Option(None),
          namespace: str = typer.Option(None, "-n")):
    """Attach to a remote breakpoint() (pdb over WebSocket)."""
    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import controller_client
    from kubetorch_amd.serving.pdb_ws import attach

    ns = namespace or cfg.namespace
    w = controller_client().get_workload(service, ns)
    if not w:
        raise typer.Exit(1)
    pods = w.get("pods") or []
    host = pods[0].split(":")[0] if pods else "127.0.0.1"
    attach(host, port or 4444)


@app.command()
def bench(steps: int = typer.Option(8), warmup: int = typer.Option(3),
          gpus: int = typer.Option(1), batch: int = typer.Option(4),
          seq: int = typer.Option(4096),
          model: str = typer.Option("llama3-8b")):
    """Run the flagship Llama-3-8B DDP benchmark THROUGH the dispatch
    stack (deploy -> SPMD fan-out -> per-rank training loop) and print
    the tokens/s JSON. `python bench.py` is the direct (launcher-free)
    variant; the two agree within noise (profiles/r02_launcher.json)."""
    import json as _json

    import kubetorch_amd as kt
    from kubetorch_amd.models.benchmark import bench_entry

    f = kt.fn(bench_entry).to(
        kt.Compute(gpus=gpus).distribute("pytorch", workers=1,
                                         num_proc=max(1, gpus)))
    try:
        results = f(steps=steps, warmup=warmup, batch=batch, seq=seq,
                    model=model, kt_timeout=3600)
        if not isinstance(results, list):
            results = [results]
        result = next(r for r in results if r)
        console.print_json(_json.dumps(result))
    finally:
        f.teardown()


@app.command("tunnel")
def tunnel(service: str, port: int = typer.Argument(8080),
           local_port: int = typer.Option(0, "--local-port"),
           namespace: str = typer.Option("default", "-n")):
    """Bridge a local TCP port to an in-cluster service through the
    controller's public port (firewall traversal; no kubectl needed)."""
    from kubetorch_amd.client.tunnel import TcpTunnel

    t = TcpTunnel(service, port, namespace=namespace,
                  local_port=local_port).start()
    console.print(f"[green]tunnel up[/green]: 127.0.0.1:{t.local_port} -> "
                  f"{service}:{port} (ctrl-c to stop)")
    import time as _t

    try:
        while True:
            _t.sleep(3600)
    except KeyboardInterrupt:
        t.stop()


@app.command()
def dashboard(namespace: str = typer.Option(None, "-n"),
              port: int = typer.Option(9090)):
    """Open the cluster metrics dashboard (reference parity: kt dashboard).
    On Kubernetes: port-forward the bundled Prometheus
    (kubetorch-amd-metrics) and print its URL. On the local driver: print
    the live pods' /metrics endpoints."""
    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import controller_client

    ns = namespace or cfg.get("install_namespace") or cfg.namespace
    cc = controller_client()
    import httpx

    try:
        h = httpx.get(cc.base_url + "/health", timeout=10).json()
    except Exception as e:
        console.print(f"[red]controller unreachable:[/red] {e}")
        raise typer.Exit(1)
    if h.get("driver") == "local":
        for w in cc.list_workloads(cfg.namespace).get("workloads", []):
            full = cc.get_workload(w["name"], cfg.namespace) or {}
            for pod in full.get("pods") or []:
                console.print(f"{w['name']}: http://{pod}/metrics")
        console.print("local driver: per-pod Prometheus text above "
                      "(no cluster dashboard)")
        return
    from kubetorch_amd.globals import PortForward

    pf = PortForward("svc/kubetorch-amd-metrics", ns, port, port).start()
    console.print(f"[green]metrics dashboard[/green]: "
                  f"http://127.0.0.1:{port} (ctrl-c to stop)")
    try:
        pf.proc.wait()
    except KeyboardInterrupt:
        pf.stop()


@app.command("port-forward")
def port_forward(target: str, ports: str,
                 namespace: str = typer.Option(None, "-n")):
    """kubectl port-forward with health wait: `kt port-forward svc/x 8080:8081`."""
    from kubetorch_amd.config import config as cfg
    from kubetorch_amd.globals import PortForward

    local, _, remote = ports.partition(":")
    pf = PortForward(target, namespace or cfg.namespace, int(local),
                     int(remote or local)).start()
    console.print(f"forwarding 127.0.0.1:{local} -> {target}:{remote or local}"
                  " (ctrl-c to stop)")
    try:
        pf.proc.wait()
    except KeyboardInterrupt:
        pf.stop()


@app.command()
def ssh(service: str, namespace: str = typer.Option(None, "-n")):
    """Exec an interactive shell in the service's first pod (kubectl)."""
    import shutil
    import subprocess

    from kubetorch_amd.config import config as cfg

    if shutil.which("kubectl") is None:
        console.print("[red]kubectl not found; `kt ssh` needs cluster access."
                      " In local mode use `kt call`/`kt logs`.[/red]")
        raise typer.Exit(1)
    from kubetorch_amd import constants as C

    ns = namespace or cfg.namespace
    out = subprocess.run(
        ["kubectl", "-n", ns, "get", "pods", "-l",
         f"{C.SERVICE_LABEL}={service}", "-o", "name"],
        capture_output=True, text=True)
    pods = out.stdout.split()
    if not pods:
        console.print("[red]no pods found[/red]")
        raise typer.Exit(1)
    subprocess.run(["kubectl", "-n", ns, "exec", "-it", pods[0], "--", "bash"])


@app.command()
def notebook(service: str = typer.Option("notebook"),
             port: int = typer.Option(8888)):
    """Deploy a remote Jupyter server and port-forward to it."""
    import kubetorch_amd as kt

    a = kt.app(
        f"python -m jupyter lab --ip=0.0.0.0 --port={port} --allow-root "
        f"--NotebookApp.token=''",
        name=service, port=port)
    a.to(kt.Compute(cpus=2))
    console.print(f"[green]notebook deployed[/green]: {a.name} "
                  f"(use `kt port-forward` to reach :{port})")


@secrets_app.command("create")
def secrets_create(name: str, provider: str = typer.Option(None),
                   values_json: str = typer.Option("{}", "--values"),
                   namespace: str = typer.Option("default", "-n")):
    from kubetorch_amd.globals import controller_client
    from kubetorch_amd.resources.secret import Secret

    s = Secret(name, values=json.loads(values_json), provider=provider)
    controller_client().put_secret(s, namespace)
    console.print(f"[green]secret {s.k8s_name} created[/green] "
                  f"(keys: {sorted(s.values)})")


@secrets_app.command("list")
def secrets_list(namespace: str = typer.Option("default", "-n")):
    from kubetorch_amd.globals import controller_client

    for s in controller_client().list_secrets(namespace):
        console.print(f"{s['name']}  ({s['k8s_name']}, keys: {s['keys']})")


@secrets_app.command("delete")
def secrets_delete(name: str,
                   namespace: str = typer.Option("default", "-n")):
    from kubetorch_amd.globals import controller_client

    controller_client().delete_secret(name, namespace)
    console.print(f"[green]secret {name} deleted[/green]")


@volumes_app.command("create")
def volumes_create(name: str, size: str = typer.Option("10Gi"),
                   existing_pv: str = typer.Option(None, "--pv"),
                   namespace: str = typer.Option("default", "-n")):
    from kubetorch_amd.globals import controller_client
    from kubetorch_amd.resources.volume import Volume

    v = Volume(name, size=size, existing_pv=existing_pv)
    controller_client().put_volume(v, namespace)
    console.print(f"[green]pvc {v.claim_name} created[/green]")


@volumes_app.command("list")
def volumes_list(namespace: str = typer.Option("default", "-n")):
    from kubetorch_amd.globals import controller_client

    for v in controller_client().list_volumes(namespace):
        console.print(f"{v['name']}  {v.get('spec', {})}")


@volumes_app.command("delete")
def volumes_delete(name: str, namespace: str = typer.Option("default", "-n")):
    from kubetorch_amd.globals import controller_client

    controller_client().delete_volume(name, namespace)
    console.print(f"[green]pvc {name} deleted[/green]")


@server_app.command("start")
def server_start(port: int = typer.Option(None),
                 kind: str = typer.Option("worker",
                                          help="worker|controller|store|gpu-data")):
    """Run a framework service in the foreground (pod/BYO entrypoint)."""
    if kind == "worker":
        from kubetorch_amd.serving import http_server

        sys.argv = ["http_server"] + (["--port", str(port)] if port else [])
        http_server.main()
    elif kind == "controller":
        from kubetorch_amd.controller import app as controller

        sys.argv = ["controller"] + (["--port", str(port)] if port else [])
        controller.main()
    elif kind == "store":
        from kubetorch_amd.data_store import server as store

        sys.argv = ["store"] + (["--port", str(port)] if port else [])
        store.main()
    elif kind == "gpu-data":
        from kubetorch_amd.data_store import pod_data_server

        sys.argv = ["pod_data_server"]
        pod_data_server.main()


def main():
    app()


if __name__ == "__main__":
    main()
