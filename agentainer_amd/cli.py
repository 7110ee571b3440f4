"""CLI — the reference's command surface as a thin HTTP client.

Mirrors the cobra command tree (reference cmd/agentainer/main.go:266-281):
server, deploy, start, stop, restart, pause, resume, remove, logs, list,
invoke, requests, health, metrics, audit, plus `backup
create|restore|list|delete|export`. Unlike the reference — whose CLI built
images and backups client-side against Docker/Redis directly
(main.go:377-458, 1452-1590) — every verb here goes through the REST API
(the unification SURVEY.md §1 recommends).

`deploy --config file.yaml` multi-agent deployments use the
AgentDeployment schema (main.go:1007-1110 analog).
"""

from __future__ import annotations

import json
import os
import sys
from typing import Any, Dict, Optional

import click
import httpx

from .config import DEFAULT_TOKEN, load_config, load_deployment

DEFAULT_URL = os.environ.get("AGENTAINER_URL", "http://127.0.0.1:8081")


class Client:
    def __init__(self, url: str, token: str):
        self.url = url.rstrip("/")
        self.token = token

    def call(self, method: str, path: str, body: Optional[Dict[str, Any]] = None,
             params: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
        headers = {"Authorization": f"Bearer {self.token}"}
        try:
            r = httpx.request(method, self.url + path, json=body, params=params,
                              headers=headers, timeout=120.0)
        except httpx.ConnectError:
            click.echo(f"error: cannot reach server at {self.url} "
                       f"(is `agentainer server` running?)", err=True)
            sys.exit(2)
        try:
            payload = r.json()
        except json.JSONDecodeError:
            payload = {"success": False, "message": r.text}
        if r.status_code >= 400 and not payload.get("success"):
            click.echo(f"error ({r.status_code}): {payload.get('message')}", err=True)
            sys.exit(1)
        return payload


pass_client = click.make_pass_decorator(Client)


@click.group()
@click.option("--url", default=DEFAULT_URL, help="server URL")
@click.option("--token", default=None, help="API token")
@click.pass_context
def cli(ctx, url: str, token: Optional[str]):
    """agentainer-amd: MI355X-native LLM-agent runtime."""
    if token is None:
        token = os.environ.get("AGENTAINER_TOKEN") or \
            load_config().get("security", "api_token", DEFAULT_TOKEN)
    ctx.obj = Client(url, token)


def _table(rows, cols):
    if not rows:
        click.echo("(none)")
        return
    widths = [max(len(str(c)), max((len(str(r.get(c, ""))) for r in rows), default=0))
              for c in cols]
    click.echo("  ".join(c.upper().ljust(w) for c, w in zip(cols, widths)))
    for r in rows:
        click.echo("  ".join(str(r.get(c, "")).ljust(w) for c, w in zip(cols, widths)))


# ---------- server ----------

@cli.command()
@click.option("--host", default=None)
@click.option("--port", default=None, type=int)
@click.option("--config", "config_path", default=None, help="config.yaml path")
@click.option("--engine-device", default=None,
              help="override engine device: auto|cuda|cpu|echo")
def server(host, port, config_path, engine_device):
    """Run the runtime server (control plane + inference engine)."""
    from .api.server import run_server
    from .service import Runtime

    cfg = load_config(config_path)
    if engine_device:
        cfg.data["engine"]["device"] = engine_device
    rt = Runtime(cfg)
    run_server(rt, host=host, port=port)


# ---------- deploy ----------

@cli.command()
@click.argument("model", required=False)
@click.option("--name", default=None)
@click.option("--config", "config_path", default=None,
              help="AgentDeployment YAML for multi-agent deploy")
@click.option("--dtype", default="bf16")
@click.option("--tp-degree", default=1, type=int)
@click.option("--kv-budget", default="0", help="per-agent KV budget, e.g. 2G")
@click.option("--max-context", default=8192, type=int)
@click.option("--env", "env_kv", multiple=True, help="KEY=VALUE (repeatable)")
@click.option("--auto-restart", is_flag=True)
@click.option("--token", "agent_token", default=None, help="per-agent token")
@click.option("--system-prompt", default="")
@pass_client
def deploy(client: Client, model, name, config_path, dtype, tp_degree, kv_budget,
           max_context, env_kv, auto_restart, agent_token, system_prompt):
    """Deploy agent(s): MODEL is a family id (e.g. llama3-8b) or weights path."""
    from .config.deployment import parse_memory

    if config_path:
        specs = load_deployment(config_path)
        for spec in specs:
            payload = spec.to_dict()
            resp = client.call("POST", "/agents", payload)
            a = resp.get("data") or {}
            click.echo(f"deployed {a.get('name')} -> {a.get('id')}")
        return
    if not model:
        raise click.UsageError("MODEL argument or --config required")
    env = dict(kv.split("=", 1) for kv in env_kv)
    resp = client.call("POST", "/agents", {
        "name": name or model.replace("/", "-"),
        "model": model, "dtype": dtype, "tp_degree": tp_degree,
        "kv_budget": parse_memory(kv_budget), "max_context": max_context,
        "env": env, "auto_restart": auto_restart, "token": agent_token,
        "system_prompt": system_prompt,
    })
    a = resp.get("data") or {}
    click.echo(f"deployed {a.get('name')} -> {a.get('id')}")


# ---------- lifecycle ----------

def _lifecycle_cmd(op: str, helptext: str):
    @cli.command(name=op, help=helptext)
    @click.argument("agent_id")
    @pass_client
    def _cmd(client: Client, agent_id: str):
        resp = client.call("POST", f"/agents/{agent_id}/{op}")
        click.echo(resp.get("message") or "ok")
    return _cmd


_lifecycle_cmd("start", "Start an agent (attach shard + allocate KV).")
_lifecycle_cmd("stop", "Stop an agent (drain + offload KV to host).")
_lifecycle_cmd("restart", "Restart an agent.")
_lifecycle_cmd("pause", "Pause admission; KV stays resident.")
_lifecycle_cmd("resume", "Resume a paused/stopped/failed agent (KV restore).")


@cli.command()
@click.argument("agent_id")
@pass_client
def remove(client: Client, agent_id: str):
    """Remove an agent and purge its request queues."""
    resp = client.call("DELETE", f"/agents/{agent_id}")
    click.echo(resp.get("message") or "ok")


@cli.command(name="list")
@pass_client
def list_cmd(client: Client):
    """List agents."""
    resp = client.call("GET", "/agents")
    rows = [{"id": a["id"], "name": a["name"], "model": a["model"],
             "status": a["status"], "tp": a.get("tp_degree", 1)}
            for a in resp.get("data") or []]
    _table(rows, ["id", "name", "model", "status", "tp"])


@cli.command()
@click.argument("agent_id", required=False, default="")
@click.option("--limit", default=100, type=int)
@click.option("--follow", "-f", is_flag=True,
              help="stream live log lines (SSE); ctrl-c to stop")
@pass_client
def logs(client: Client, agent_id: str, limit: int, follow: bool):
    """Show an agent's engine log (or, with -f, tail the live stream)."""
    if follow:
        params = {"agent_id": agent_id} if agent_id else {}
        with httpx.stream("GET", f"{client.url}/logs/stream", params=params,
                          headers={"Authorization": f"Bearer {client.token}"},
                          timeout=None) as r:
            if r.status_code != 200:
                click.echo(f"error ({r.status_code})", err=True)
                sys.exit(1)
            for line in r.iter_lines():
                if line.startswith("data: "):
                    click.echo(line[len("data: "):])
        return
    if not agent_id:
        click.echo("error: AGENT_ID required unless --follow", err=True)
        sys.exit(2)
    resp = client.call("GET", f"/agents/{agent_id}/logs", params={"limit": limit})
    for e in resp.get("data") or []:
        click.echo(json.dumps(e))


@cli.command()
@click.argument("agent_id")
@click.option("--message", "-m", required=True)
@click.option("--stream", is_flag=True, help="stream tokens as they decode (SSE)")
@pass_client
def invoke(client: Client, agent_id: str, message: str, stream: bool):
    """Send a chat message to an agent (authenticated dispatch)."""
    if stream:
        with httpx.stream("POST", f"{client.url}/agent/{agent_id}/chat",
                          json={"message": message, "stream": True},
                          timeout=120.0) as r:
            if r.status_code != 200:
                click.echo(f"error ({r.status_code})", err=True)
                sys.exit(1)
            for line in r.iter_lines():
                if not line.startswith("data: "):
                    continue
                ev = json.loads(line[len("data: "):])
                if ev.get("done"):
                    click.echo("")  # newline after the streamed text
                elif "error" in ev:
                    click.echo(f"\nerror: {ev['error']}", err=True)
                else:
                    click.echo(ev.get("text", ""), nl=False)
        return
    resp = client.call("POST", f"/agents/{agent_id}/invoke",
                       {"path": "/chat", "method": "POST",
                        "body": {"message": message}})
    data = resp.get("data")
    if isinstance(data, dict) and "response" in data:
        click.echo(data["response"])
    else:
        click.echo(json.dumps(data, indent=2))


@cli.command()
@click.argument("agent_id")
@click.option("--status", default="", help="pending|completed|failed")
@pass_client
def requests(client: Client, agent_id: str, status: str):
    """Show an agent's request queues (WAL)."""
    resp = client.call("GET", f"/agents/{agent_id}/requests",
                       params={"status": status} if status else None)
    data = resp.get("data") or {}
    for queue, reqs in data.items():
        click.echo(f"== {queue} ({len(reqs)})")
        rows = [{"id": r["id"], "path": r["path"], "retries": r["retry_count"],
                 "created": r["created_at"]} for r in reqs]
        _table(rows, ["id", "path", "retries", "created"])


@cli.command()
@click.argument("agent_id", required=False)
@pass_client
def health(client: Client, agent_id: Optional[str]):
    """Show health status for one agent or all."""
    if agent_id:
        resp = client.call("GET", f"/agents/{agent_id}/health")
        click.echo(json.dumps(resp.get("data"), indent=2))
    else:
        resp = client.call("GET", "/health/agents")
        data = resp.get("data") or {}
        rows = [{"id": aid, "healthy": st.get("healthy"),
                 "failures": st.get("consecutive_failures")}
                for aid, st in data.items()]
        _table(rows, ["id", "healthy", "failures"])


@cli.command()
@click.argument("agent_id", required=False)
@click.option("--history", is_flag=True)
@click.option("--duration", default=3600.0, type=float, help="history window (s)")
@click.option("--device", "device_", is_flag=True,
              help="device-level metrics (HBM, xGMI link counters)")
@pass_client
def metrics(client: Client, agent_id, history: bool, duration: float,
            device_: bool):
    """Show agent metrics (tokens/s, req/s, p50/p99, KV usage), or
    device-level HBM/xGMI metrics with --device."""
    if device_:
        resp = client.call("GET", "/metrics/device")
        click.echo(json.dumps(resp.get("data"), indent=2))
        return
    if not agent_id:
        raise click.UsageError("AGENT_ID required (or use --device)")
    if history:
        resp = client.call("GET", f"/agents/{agent_id}/metrics/history",
                           params={"duration_s": duration})
        for row in resp.get("data") or []:
            click.echo(json.dumps(row))
    else:
        resp = client.call("GET", f"/agents/{agent_id}/metrics")
        click.echo(json.dumps(resp.get("data"), indent=2))


@cli.command()
@click.option("--user", default="")
@click.option("--action", default="")
@click.option("--resource", default="")
@click.option("--limit", default=100, type=int)
@pass_client
def audit(client: Client, user, action, resource, limit):
    """Show the audit trail."""
    resp = client.call("GET", "/audit", params={
        "user": user, "action": action, "resource": resource, "limit": limit})
    for e in resp.get("data") or []:
        click.echo(json.dumps(e))


# ---------- backup ----------

@cli.group()
def backup():
    """Backup/restore agent configs + conversation state."""


@backup.command("create")
@click.option("--name", required=True)
@click.option("--description", default="")
@click.option("--agent", "agents", multiple=True,
              help="limit to specific agent ids (repeatable)")
@pass_client
def backup_create(client: Client, name, description, agents):
    body = {"name": name, "description": description}
    if agents:
        body["agent_ids"] = list(agents)
    resp = client.call("POST", "/backups", body)
    click.echo(json.dumps(resp.get("data")))


@backup.command("list")
@pass_client
def backup_list(client: Client):
    resp = client.call("GET", "/backups")
    _table(resp.get("data") or [], ["id", "name", "n_agents", "created_at"])


@backup.command("restore")
@click.argument("backup_id")
@pass_client
def backup_restore(client: Client, backup_id):
    resp = client.call("POST", f"/backups/{backup_id}/restore")
    for a in resp.get("data") or []:
        click.echo(f"restored {a['name']} -> {a['id']}")


@backup.command("delete")
@click.argument("backup_id")
@pass_client
def backup_delete(client: Client, backup_id):
    resp = client.call("DELETE", f"/backups/{backup_id}")
    click.echo(resp.get("message") or "ok")


@backup.command("export")
@click.argument("backup_id")
@click.option("--output", "-o", required=True, type=click.Path())
@pass_client
def backup_export(client: Client, backup_id, output):
    r = httpx.get(client.url + f"/backups/{backup_id}/export",
                  headers={"Authorization": f"Bearer {client.token}"},
                  timeout=120.0)
    r.raise_for_status()
    with open(output, "wb") as f:
        f.write(r.content)
    click.echo(f"exported {backup_id} -> {output}")


@backup.command("import")
@click.argument("bundle", type=click.Path(exists=True))
@pass_client
def backup_import(client: Client, bundle):
    with open(bundle, "rb") as f:
        raw = f.read()
    r = httpx.post(client.url + "/backups/import", content=raw,
                   headers={"Authorization": f"Bearer {client.token}"},
                   timeout=120.0)
    r.raise_for_status()
    click.echo(json.dumps(r.json().get("data")))


def main():
    cli()


if __name__ == "__main__":
    main()
