"""deepflow-ctl — operator CLI (reference: cli/ctl/*.go).

Talks to a running all-in-one server over HTTP.

  python -m deepflow_amd.cli query "SELECT ..."         DF-SQL query
  python -m deepflow_amd.cli tables|tags|metrics        metadata discovery
  python -m deepflow_amd.cli agent list                 registered agents
  python -m deepflow_amd.cli agent rebalance            redistribute agents
  python -m deepflow_amd.cli stats                      self-telemetry
  python -m deepflow_amd.cli trace <trace_id>           trace tree
  python -m deepflow_amd.cli promql '<expr>'            instant PromQL
"""
from __future__ import annotations

import json
import sys

import click
import requests


@click.group()
@click.option("--server", default="http://127.0.0.1:20416",
              envvar="DEEPFLOW_SERVER", help="querier base URL")
@click.pass_context
def cli(ctx, server):
    ctx.obj = {"server": server.rstrip("/")}


def _print_table(result: dict) -> None:
    cols = result.get("columns", [])
    vals = result.get("values", [])
    if not cols:
        click.echo("(empty)")
        return
    widths = [max(len(str(c)), *(len(str(r[i])) for r in vals)) if vals
              else len(str(c)) for i, c in enumerate(cols)]
    click.echo("  ".join(str(c).ljust(w) for c, w in zip(cols, widths)))
    for r in vals:
        click.echo("  ".join(str(x).ljust(w) for x, w in zip(r, widths)))


@cli.command()
@click.argument("sql")
@click.option("--json", "as_json", is_flag=True)
@click.pass_context
def query(ctx, sql, as_json):
    """Run a DF-SQL query."""
    r = requests.post(f"{ctx.obj['server']}/v1/query/", json={"sql": sql},
                      timeout=60)
    body = r.json()
    if body.get("OPT_STATUS") != "SUCCESS":
        click.echo(f"error: {body.get('DESCRIPTION')}", err=True)
        sys.exit(1)
    if as_json:
        click.echo(json.dumps(body["result"]))
    else:
        _print_table(body["result"])


@cli.command()
@click.pass_context
def tables(ctx):
    ctx.invoke(query, sql="show tables", as_json=False)


@cli.command()
@click.option("--table", default="l7_flow_log")
@click.pass_context
def tags(ctx, table):
    ctx.invoke(query, sql=f"show tags from {table}", as_json=False)


@cli.command()
@click.option("--table", default="l7_flow_log")
@click.pass_context
def metrics(ctx, table):
    ctx.invoke(query, sql=f"show metrics from {table}", as_json=False)


@cli.group()
def agent():
    """Agent management."""


@agent.command("list")
@click.pass_context
def agent_list(ctx):
    r = requests.get(f"{ctx.obj['server']}/v1/agents/", timeout=30)
    rows = r.json()
    if not rows:
        click.echo("(no agents)")
        return
    _print_table({"columns": list(rows[0].keys()),
                  "values": [list(a.values()) for a in rows]})


@agent.command("rebalance")
@click.pass_context
def agent_rebalance(ctx):
    r = requests.post(f"{ctx.obj['server']}/v1/rebalance/", timeout=30)
    click.echo(json.dumps(r.json()))


@agent.command("run")
@click.option("--server", "ingest_server", default="127.0.0.1:20033",
              help="ingest host:port (trident framed protocol)")
@click.option("--iface", default="lo", help="capture interface")
@click.option("--vtap-id", default=1, type=int)
@click.option("--ring/--no-ring", default=True,
              help="TPACKET_V3 block ring vs per-packet AF_PACKET")
@click.option("--ebpf/--no-ebpf", default=True,
              help="attach the eBPF socket tracer when permitted")
@click.option("--flush-interval", default=1.0, type=float)
def agent_run(ingest_server, iface, vtap_id, ring, ebpf, flush_interval):
    """Run the collection agent: capture -> FlowMap/L7 parse -> framed
    sender (the deployment entry the manifests launch)."""
    import time as _t
    from deepflow_amd.agent import Agent
    host, port = ingest_server.rsplit(":", 1)
    a = Agent(vtap_id=vtap_id, server=(host, int(port)))
    cap = None
    try:
        if ring:
            from deepflow_amd.agent.capture import RingCapture
            cap = RingCapture(a, iface=iface)
        else:
            from deepflow_amd.agent.capture import CaptureWorker
            cap = CaptureWorker(a, iface=iface)
        cap.start()
        click.echo(f"capture on {iface} "
                   f"({'tpacket_v3 ring' if ring else 'af_packet'})")
    except (PermissionError, OSError) as e:
        click.echo(f"capture unavailable ({e}); running without packets")
    tracer = a.start_ebpf() if ebpf else None
    click.echo(f"ebpf socket tracer: {'on' if tracer else 'unavailable'}")
    click.echo(f"sending to {ingest_server}; ctrl-c to stop")
    try:
        while True:
            _t.sleep(flush_interval)
            try:
                a.flush_to_server(_t.time_ns(), compress=True)
            except OSError as e:
                click.echo(f"send failed: {e}; retrying")
    except KeyboardInterrupt:
        pass
    finally:
        if cap is not None:
            cap.stop()
        a.stop_ebpf()
        a.close()


@agent.command("config")
@click.argument("group", default="default")
@click.option("--set", "kv", multiple=True,
              help="key=value overrides to push")
@click.pass_context
def agent_config(ctx, group, kv):
    """Show or update an agent group's config (reference:
    deepflow-ctl agent-group-config)."""
    base = f"{ctx.obj['server']}/v1/agent-group-config/{group}"
    if kv:
        body = {}
        for item in kv:
            k, _, v = item.partition("=")
            body[k] = yaml_value(v)
        r = requests.post(base, json=body, timeout=30)
    else:
        r = requests.get(base, timeout=30)
    click.echo(json.dumps(r.json(), indent=2))


def yaml_value(v: str):
    for cast in (int, float):
        try:
            return cast(v)
        except ValueError:
            pass
    return {"true": True, "false": False}.get(v.lower(), v)


@cli.group()
def domain():
    """Cloud-platform domains (reference: deepflow-ctl domain)."""


@domain.command("list")
@click.pass_context
def domain_list(ctx):
    r = requests.get(f"{ctx.obj['server']}/v1/domains/", timeout=30)
    rows = r.json()
    if not rows:
        click.echo("(no domains)")
        return
    _print_table({"columns": list(rows[0].keys()),
                  "values": [list(d.values()) for d in rows]})


@domain.command("add")
@click.argument("name")
@click.option("--type", "dtype", default="filereader")
@click.option("--config", default="{}")
@click.pass_context
def domain_add(ctx, name, dtype, config):
    r = requests.post(f"{ctx.obj['server']}/v1/domains/",
                      json={"name": name, "type": dtype,
                            "config": json.loads(config)}, timeout=30)
    click.echo(json.dumps(r.json()))


@cli.command()
@click.pass_context
def genesis(ctx):
    """Dump the genesis (auto-discovered platform) inventory."""
    r = requests.get(f"{ctx.obj['server']}/v1/genesis/", timeout=30)
    click.echo(json.dumps(r.json(), indent=2))


@cli.command()
@click.pass_context
def stats(ctx):
    """Self-telemetry counters."""
    r = requests.get(f"{ctx.obj['server']}/v1/stats", timeout=30)
    for s in r.json():
        tags = ",".join(f"{k}={v}" for k, v in
                        zip(s.get("tag_names", []), s.get("tag_values", [])))
        for name, val in zip(s["metrics_float_names"],
                             s["metrics_float_values"]):
            click.echo(f"{s['name']}{{{tags}}} {name}={val}")


@cli.command()
@click.argument("trace_id")
@click.pass_context
def trace(ctx, trace_id):
    """Distributed trace tree."""
    r = requests.get(f"{ctx.obj['server']}/v1/tracing/{trace_id}", timeout=60)
    body = r.json()

    def walk(idx, depth):
        n = body["spans"][idx]
        click.echo("  " * depth +
                   f"{n['service'] or '?'} {n['resource'] or ''} "
                   f"[{n['duration_ns'] / 1e6:.2f} ms]")
        for c in n["children"]:
            walk(c, depth + 1)

    for root in body["roots"]:
        walk(root, 0)
    click.echo(f"({body['span_count']} spans)")


@cli.command()
@click.argument("expr")
@click.option("--time", "t", default=None)
@click.pass_context
def promql(ctx, expr, t):
    """Instant PromQL query."""
    params = {"query": expr}
    if t:
        params["time"] = t
    r = requests.get(f"{ctx.obj['server']}/prom/api/v1/query", params=params,
                     timeout=60)
    click.echo(json.dumps(r.json(), indent=2))


def main():
    cli()


if __name__ == "__main__":
    main()


@cli.command()
@click.argument("cmd")
@click.option("--port", type=int, required=True,
              help="server debug-bus UDP port")
def debug(cmd, port):
    """Query the server's UDP debug bus (stats/store/queues/agents)."""
    from .utils.debug_bus import debug_call
    click.echo(json.dumps(debug_call(port, cmd), indent=2, default=str))
