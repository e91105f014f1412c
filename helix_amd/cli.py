"""helix-amd CLI (parity with the reference's `helix` cobra CLI,
api/cmd/helix root.go:45-72: serve, apply, chat, knowledge, secret,
session, model, runner, version, test).
"""
from __future__ import annotations

import asyncio
import json
import os
import sys

import typer

app = typer.Typer(name="helix-amd", help="MI355X-native private GenAI stack")


def _api(ctx_url: str = "") -> tuple[str, dict]:
    url = ctx_url or os.environ.get("HELIX_URL", "http://localhost:8080")
    key = os.environ.get("HELIX_API_KEY", "admin-key")
    return url, {"Authorization": f"Bearer {key}"}


@app.command()
def serve(host: str = typer.Option(None), port: int = typer.Option(None),
          local_runner: bool = typer.Option(False, "--local-runner"),
          store_path: str = typer.Option(None)):
    """Start the control plane (optionally with an in-process GPU runner)."""
    import uvicorn
    from helix_amd.server.app import create_app
    from helix_amd.server.config import load_config
    cfg = load_config()
    if host:
        cfg.web.host = host
    if port:
        cfg.web.port = port
    if local_runner:
        cfg.runner_plane.local_runner = True
    if store_path:
        cfg.store.path = store_path
    api = create_app(cfg)
    uvicorn.run(api, host=cfg.web.host, port=cfg.web.port, log_level="info")


@app.command()
def runner(api_url: str = typer.Option("http://localhost:8080"),
           runner_id: str = typer.Option("runner-0"),
           port: int = typer.Option(8090),
           advertise: str = typer.Option(""),
           device: str = typer.Option("cuda:0"),
           tunnel: bool = typer.Option(False, "--tunnel",
                                       help="reverse-dial (NAT-friendly): "
                                            "no inbound port needed"),
           preload: str = typer.Option("", help="comma-separated models")):
    """Start a GPU runner: OpenAI-compatible server + heartbeat."""
    import threading

    import uvicorn
    from helix_amd.runner.heartbeat import heartbeat_loop
    from helix_amd.runner.http import create_runner_app
    from helix_amd.runner.service import RunnerService
    from helix_amd.server.config import load_config
    cfg = load_config()
    svc = RunnerService(device=device)
    for m in filter(None, preload.split(",")):
        svc.ensure_loaded(m.strip())
    api = create_runner_app(svc, runner_id)
    if tunnel:
        addr = f"tunnel:{runner_id}"

        def run_tunnel():
            from helix_amd.server.tunnel import tunnel_loop
            asyncio.run(tunnel_loop(api_url, cfg.runner_plane.runner_token,
                                    runner_id, svc))
        threading.Thread(target=run_tunnel, daemon=True).start()
    else:
        addr = advertise or f"http://{_local_ip()}:{port}"

    def beat():
        asyncio.run(heartbeat_loop(
            api_url, cfg.runner_plane.runner_token, runner_id, addr, svc,
            interval=cfg.runner_plane.heartbeat_interval_s))
    threading.Thread(target=beat, daemon=True).start()

    def poll_assignment():
        from helix_amd.runner.assignment import assignment_loop
        asyncio.run(assignment_loop(
            api_url, cfg.runner_plane.runner_token, runner_id, svc))
    threading.Thread(target=poll_assignment, daemon=True).start()
    try:
        uvicorn.run(api, host="0.0.0.0", port=port, log_level="info")
    finally:
        # graceful drain on SIGTERM/SIGINT (rolling restarts): reject
        # new work, let in-flight sequences finish, then free the GPU
        typer.echo("draining in-flight requests...")
        svc.drain(timeout=30)
        svc.shutdown()


def _local_ip() -> str:
    import socket
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        s.connect(("10.255.255.255", 1))
        return s.getsockname()[0]
    except Exception:
        return "127.0.0.1"
    finally:
        s.close()


@app.command()
def apply(file: str = typer.Option(..., "-f", "--file"),
          url: str = typer.Option("", "--url"),
          global_: bool = typer.Option(False, "--global")):
    """Create/update an app from helix.yaml (reference `helix apply`)."""
    import httpx
    from helix_amd.server.apps import load_app_file
    cfg = load_app_file(file)
    api, headers = _api(url)
    # update-if-exists by name
    with httpx.Client(timeout=30) as http:
        existing = http.get(f"{api}/api/v1/apps", headers=headers).json()
        match = next((a for a in existing
                      if a.get("config", {}).get("name") == cfg.name), None)
        if match:
            r = http.put(f"{api}/api/v1/apps/{match['id']}",
                         json={"config": cfg.model_dump(by_alias=True),
                               "global": global_}, headers=headers)
        else:
            r = http.post(f"{api}/api/v1/apps",
                          json={"config": cfg.model_dump(by_alias=True),
                                "global": global_}, headers=headers)
        r.raise_for_status()
        doc = r.json()
        typer.echo(f"applied app {doc['id']} ({cfg.name})")


@app.command()
def chat(message: str = typer.Argument(...),
         model: str = typer.Option("", "--model"),
         app_id: str = typer.Option("", "--app"),
         url: str = typer.Option("", "--url"),
         stream: bool = typer.Option(True)):
    """One-shot chat against /v1/chat/completions."""
    import httpx
    api, headers = _api(url)
    body = {"messages": [{"role": "user", "content": message}],
            "stream": stream}
    if model:
        body["model"] = model
    if app_id:
        body["app_id"] = app_id
    with httpx.Client(timeout=300) as http:
        if stream:
            with http.stream("POST", f"{api}/v1/chat/completions",
                             json=body, headers=headers) as r:
                r.raise_for_status()
                for line in r.iter_lines():
                    if not line.startswith("data: ") or \
                            line[6:].strip() == "[DONE]":
                        continue
                    chunk = json.loads(line[6:])
                    if chunk.get("choices"):
                        sys.stdout.write(
                            chunk["choices"][0].get("delta", {}).get(
                                "content") or "")
                        sys.stdout.flush()
                sys.stdout.write("\n")
        else:
            r = http.post(f"{api}/v1/chat/completions", json=body,
                          headers=headers)
            r.raise_for_status()
            typer.echo(r.json()["choices"][0]["message"]["content"])


knowledge_app = typer.Typer(help="Manage knowledge sources")
app.add_typer(knowledge_app, name="knowledge")


@knowledge_app.command("list")
def knowledge_list(url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    rows = httpx.get(f"{api}/api/v1/knowledge", headers=headers).json()
    for k in rows:
        typer.echo(f"{k['id']}  {k['name']:<24} {k['state']:<10} "
                   f"v{k.get('version', 0)} chunks={k.get('chunks', 0)}")


@knowledge_app.command("create")
def knowledge_create(name: str,
                     text: str = typer.Option("", help="inline text"),
                     path: str = typer.Option("", help="filestore path"),
                     s3: str = typer.Option("", help="bucket/prefix"),
                     url: str = typer.Option("", "--web-url"),
                     refresh_schedule: str = typer.Option("")):
    """Create a knowledge source (text / filestore / s3 / web)."""
    if text:
        source = {"text": text}
    elif path:
        source = {"filestore": {"path": path}}
    elif s3:
        bucket, _, prefix = s3.partition("/")
        source = {"s3": {"bucket": bucket, "prefix": prefix}}
    elif url:
        source = {"web": {"urls": [url]}}
    else:
        typer.echo("one of --text/--path/--s3/--web-url required")
        raise typer.Exit(2)
    doc = _client().create_knowledge(name, source, refresh_schedule)
    typer.echo(f"{doc['id']}  {doc['state']}")


@knowledge_app.command("refresh")
def knowledge_refresh(kid: str, url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    r = httpx.post(f"{api}/api/v1/knowledge/{kid}/refresh", headers=headers)
    typer.echo(r.json())


secret_app = typer.Typer(help="Manage secrets")
app.add_typer(secret_app, name="secret")


@secret_app.command("set")
def secret_set(name: str, value: str, url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    r = httpx.post(f"{api}/api/v1/secrets", json={"name": name,
                                                  "value": value},
                   headers=headers)
    typer.echo(r.json())


@secret_app.command("list")
def secret_list(url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    for s in httpx.get(f"{api}/api/v1/secrets", headers=headers).json():
        typer.echo(f"{s['name']}")


session_app = typer.Typer(help="Sessions")
app.add_typer(session_app, name="session")


@session_app.command("list")
def session_list(url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    for s in httpx.get(f"{api}/api/v1/sessions", headers=headers).json():
        typer.echo(f"{s['id']}  {s['name'][:40]}")


@session_app.command("export")
def session_export(session_id: str,
                   out: str = typer.Option("", "-o")):
    """Export a session with its interactions as JSON (docs say this
    exists; backup/portability role)."""
    c = _client()
    doc = c.get_session(session_id)
    blob = json.dumps(doc, indent=2, default=str)
    if out:
        with open(out, "w") as fh:
            fh.write(blob)
        typer.echo(f"wrote {out}")
    else:
        typer.echo(blob)


@session_app.command("delete")
def session_delete(session_id: str):
    _client().delete_session(session_id)
    typer.echo("deleted")


model_app = typer.Typer(help="Model catalog & local models")
app.add_typer(model_app, name="model")

app_app = typer.Typer(help="Manage apps/agents")
app.add_typer(app_app, name="app")


@app_app.command("list")
def app_list(url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    for a in httpx.get(f"{api}/api/v1/apps", headers=headers).json():
        cfg = (a.get("config") or {}).get("helix") or a.get("config") or {}
        n = len(cfg.get("assistants") or [])
        typer.echo(f"{a['id']:<30} {cfg.get('name', ''):<24} "
                   f"{n} assistant(s)")


@app_app.command("get")
def app_get(app_id: str, url: str = typer.Option("", "--url")):
    import json as _json

    import httpx
    api, headers = _api(url)
    r = httpx.get(f"{api}/api/v1/apps/{app_id}", headers=headers)
    typer.echo(_json.dumps(r.json(), indent=2))


@app_app.command("delete")
def app_delete(app_id: str, url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    r = httpx.delete(f"{api}/api/v1/apps/{app_id}", headers=headers)
    typer.echo("deleted" if r.status_code == 200 else f"error: {r.text}")


@model_app.command("list")
def model_list(url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    for m in httpx.get(f"{api}/api/v1/helix-models", headers=headers).json():
        typer.echo(f"{m['id']:<16} ctx={m.get('context_length', '?'):<7} "
                   f"{m.get('runtime', '')}")


@model_app.command("load")
def model_load(model: str, url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    r = httpx.post(f"{api}/api/v1/local-models/{model}/load",
                   headers=headers, timeout=900)
    typer.echo(r.json())


@model_app.command("unload")
def model_unload(model: str, url: str = typer.Option("", "--url")):
    import httpx
    api, headers = _api(url)
    r = httpx.post(f"{api}/api/v1/local-models/{model}/unload",
                   headers=headers)
    typer.echo(r.json())


@app.command()
def backup(url: str = typer.Option("", "--url"),
           path: str = typer.Option("", "--path",
                                    help="server-side destination")):
    """Online control-plane store backup (admin)."""
    import httpx
    api, headers = _api(url)
    body = {"path": path} if path else {}
    r = httpx.post(f"{api}/api/v1/admin/backup", json=body,
                   headers=headers, timeout=120)
    typer.echo(r.json())


@app.command()
def version():
    from helix_amd import __version__
    typer.echo(f"helix_amd {__version__}")


@app.command()
def test(file: str = typer.Option("helix.yaml", "-f", "--file"),
         url: str = typer.Option("", "--url")):
    """Run helix.yaml assistant tests (reference `helix test`): each test
    step sends the prompt and an LLM judge checks the expected response."""
    import httpx
    from helix_amd.server.apps import load_app_file
    cfg = load_app_file(file)
    api, headers = _api(url)
    failures = 0
    with httpx.Client(timeout=300) as http:
        for asst in cfg.assistants:
            for t in asst.tests:
                for step in t.steps:
                    prompt = step.get("prompt", "")
                    expected = step.get("expected_output", "")
                    r = http.post(f"{api}/v1/chat/completions", json={
                        "model": asst.model,
                        "messages": [{"role": "user", "content": prompt}]},
                        headers=headers)
                    answer = r.json()["choices"][0]["message"]["content"]
                    judge = http.post(f"{api}/v1/chat/completions", json={
                        "model": asst.model,
                        "messages": [{"role": "user", "content":
                                      f"Does this answer satisfy the "
                                      f"expectation?\nExpectation: "
                                      f"{expected}\nAnswer: {answer}\n"
                                      f"Reply YES or NO."}]},
                        headers=headers)
                    verdict = judge.json()["choices"][0]["message"][
                        "content"].strip().upper()
                    ok = verdict.startswith("YES")
                    failures += 0 if ok else 1
                    typer.echo(f"[{'PASS' if ok else 'FAIL'}] {t.name}: "
                               f"{prompt[:40]}")
    raise typer.Exit(1 if failures else 0)


@app.command()
def status(url: str = typer.Option("http://localhost:8080"),
           key: str = typer.Option("admin-key", envvar="HELIX_API_KEY")):
    """Cluster overview: runners, loaded models, store counts."""
    import httpx
    H = {"Authorization": f"Bearer {key}"}
    try:
        runners = httpx.get(f"{url}/api/v1/admin/runners", headers=H,
                            timeout=10).json()
    except Exception as e:
        typer.echo(f"control plane unreachable at {url}: {e}")
        raise typer.Exit(1)
    typer.echo(f"runners: {len(runners)}")
    for r in runners:
        models = ", ".join(m.get("model_id", str(m))
                           for m in r.get("models", []))
        typer.echo(f"  {r.get('id')}: {r.get('status', 'ready')} "
                   f"[{models}]")
    try:
        stats = httpx.get(f"{url}/debug/stats", headers=H,
                          timeout=10).json()
        typer.echo(f"store: {stats.get('store_counts')}")
        typer.echo(f"rss: {stats.get('rss_bytes', 0) >> 20} MiB, "
                   f"threads: {stats.get('num_threads')}")
    except Exception:
        pass
    try:
        models = httpx.get(f"{url}/v1/models", headers=H,
                           timeout=10).json()
        typer.echo("models: " + ", ".join(
            m["id"] for m in models.get("data", [])[:20]))
    except Exception:
        pass


@app.command()
def doctor():
    """Environment check for an MI355X node: ROCm, GPU arch, extension,
    RCCL prerequisites."""
    import shutil
    ok = True

    def check(name, cond, hint=""):
        nonlocal ok
        mark = "ok " if cond else "FAIL"
        typer.echo(f"[{mark}] {name}" + (f" — {hint}" if (hint and not
                                                          cond) else ""))
        ok = ok and bool(cond)

    import torch
    check("python/torch", True)
    typer.echo(f"       torch {torch.__version__}")
    check("hipcc on PATH", shutil.which("hipcc") is not None,
          "install ROCm / add /opt/rocm/bin to PATH")
    import helix_amd.ops as ops
    check("helix_amd._C extension built", ops.have_native(),
          "PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace")
    has_gpu = torch.cuda.is_available()
    # informational: a control-plane-only node is valid without a GPU
    typer.echo(f"[{'ok ' if has_gpu else 'n/a'}] GPU visible" +
               ("" if has_gpu else " — control-plane-only mode"))
    if has_gpu:
        props = torch.cuda.get_device_properties(0)
        arch = getattr(props, "gcnArchName", "?")
        typer.echo(f"       {props.name} ({arch}), "
                   f"{props.total_memory >> 30} GiB")
        check("gfx950 (MI355X)", "gfx950" in str(arch),
              "kernels are compiled for gfx950 only")
        import os as _os
        check("HSA_ENABLE_IPC_MODE_LEGACY=0",
              _os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY") == "0",
              "export HSA_ENABLE_IPC_MODE_LEGACY=0 for multi-process "
              "RCCL (dmabuf IPC)")
        import torch.distributed as dist
        check("torch.distributed nccl (RCCL)",
              dist.is_nccl_available())
    raise typer.Exit(0 if ok else 1)


@app.command("export-gguf")
def export_gguf(preset: str = typer.Option("llama3-8b"),
                ckpt: str = typer.Option("", help="safetensors dir "
                                         "(random-init if empty)"),
                out: str = typer.Option(..., "-o", "--out"),
                seed: int = typer.Option(0)):
    """Export a model as a GGUF v3 checkpoint (llama.cpp tensor names,
    Q/K rows permuted for GGML rope — interops with the llama.cpp
    ecosystem; see engine/gguf.py)."""
    import torch
    from helix_amd.engine import gguf as g
    from helix_amd.models.llama import LlamaForCausalLM, PRESETS

    cfg = PRESETS[preset]
    model = LlamaForCausalLM(cfg)
    if ckpt:
        from helix_amd.engine.weights import load_llama_weights
        load_llama_weights(model, ckpt, use_async=False)
    else:
        model.init_random(seed)

    def permute(w, n_head):
        o, i = w.shape
        return (w.reshape(n_head, 2, o // n_head // 2, i)
                .swapaxes(1, 2).reshape(o, i))

    sd = dict(model.named_parameters())
    q, kv, inter = cfg.q_size, cfg.kv_size, cfg.intermediate_size
    tensors = {
        "token_embd.weight": sd["embed_tokens.weight"].data,
        "output_norm.weight": sd["final_norm_w"].data,
        "output.weight": sd["lm_head.weight"].data,
    }
    for i in range(cfg.num_layers):
        qkv = sd[f"layers.{i}.attn.qkv_proj.weight"].data
        gu = sd[f"layers.{i}.mlp.gate_up_proj.weight"].data
        tensors[f"blk.{i}.attn_q.weight"] = permute(qkv[:q], cfg.num_heads)
        tensors[f"blk.{i}.attn_k.weight"] = permute(qkv[q:q + kv],
                                                    cfg.num_kv_heads)
        tensors[f"blk.{i}.attn_v.weight"] = qkv[q + kv:]
        tensors[f"blk.{i}.attn_output.weight"] = \
            sd[f"layers.{i}.attn.o_proj.weight"].data
        tensors[f"blk.{i}.ffn_gate.weight"] = gu[:inter]
        tensors[f"blk.{i}.ffn_up.weight"] = gu[inter:]
        tensors[f"blk.{i}.ffn_down.weight"] = \
            sd[f"layers.{i}.mlp.down_proj.weight"].data
        tensors[f"blk.{i}.attn_norm.weight"] = \
            sd[f"layers.{i}.input_norm_w"].data
        tensors[f"blk.{i}.ffn_norm.weight"] = \
            sd[f"layers.{i}.post_norm_w"].data
    meta = {
        "general.architecture": "llama",
        "general.name": cfg.name,
        "llama.block_count": cfg.num_layers,
        "llama.embedding_length": cfg.hidden_size,
        "llama.feed_forward_length": cfg.intermediate_size,
        "llama.attention.head_count": cfg.num_heads,
        "llama.attention.head_count_kv": cfg.num_kv_heads,
        "llama.context_length": cfg.max_position,
        "llama.rope.freq_base": float(cfg.rope_base),
    }
    g.write_gguf(out, meta, tensors)
    est = g.estimate_gguf_bytes(out)
    typer.echo(f"wrote {out}: {len(tensors)} tensors, "
               f"{est['weights'] >> 20} MiB")


# ---------------------------------------------------------------------------
# Round-2 subcommand families (reference root.go:45-72 has ~24: org,
# project, spectask, mcp, evals, member, team ... — these map onto the
# HelixClient library, client.py).

def _client():
    from helix_amd.client import HelixClient
    return HelixClient()


org_app = typer.Typer(help="Organizations and teams")
app.add_typer(org_app, name="org")


@org_app.command("list")
def org_list():
    for o in _client().list_organizations():
        typer.echo(f"{o['id']}  {o.get('name', '')}")


@org_app.command("create")
def org_create(name: str):
    o = _client().create_organization(name)
    typer.echo(o["id"])


@org_app.command("add-member")
def org_add_member(org_id: str, user_id: str,
                   role: str = typer.Option("member")):
    _client().add_org_member(org_id, user_id, role)
    typer.echo("added")


@org_app.command("teams")
def org_teams(org_id: str):
    for t in _client().list_teams(org_id):
        typer.echo(f"{t['id']}  {t.get('name', '')}")


@org_app.command("create-team")
def org_create_team(org_id: str, name: str):
    typer.echo(_client().create_team(org_id, name)["id"])


project_app = typer.Typer(help="Projects and spec-driven tasks")
app.add_typer(project_app, name="project")


@project_app.command("list")
def project_list():
    for p in _client().list_projects():
        typer.echo(f"{p['id']}  {p.get('name', '')}")


@project_app.command("create")
def project_create(name: str):
    typer.echo(_client().create_project(name)["id"])


@project_app.command("tasks")
def project_tasks(project_id: str):
    for t in _client().list_tasks(project_id):
        typer.echo(f"{t['id']}  [{t.get('state', '?')}] "
                   f"{t.get('title', '')}")


spectask_app = typer.Typer(help="Spec-task kanban operations")
app.add_typer(spectask_app, name="spectask")


@spectask_app.command("create")
def spectask_create(project_id: str, title: str,
                    description: str = typer.Option("")):
    typer.echo(_client().create_task(project_id, title,
                                     description)["id"])


@spectask_app.command("transition")
def spectask_transition(task_id: str, state: str):
    t = _client().transition_task(task_id, state)
    typer.echo(t.get("state", ""))


@spectask_app.command("plan")
def spectask_plan(task_id: str):
    t = _client().plan_task(task_id)
    typer.echo(t.get("state", ""))


@spectask_app.command("implement")
def spectask_implement(task_id: str):
    t = _client().implement_task(task_id)
    typer.echo(f"{t.get('state', '')} branch={t.get('branch', '')}")


mcp_app = typer.Typer(help="MCP gateway operations")
app.add_typer(mcp_app, name="mcp")


@mcp_app.command("tools")
def mcp_tools(app_id: str):
    """List the MCP tools an app exposes (tools/list JSON-RPC)."""
    c = _client()
    out = c.request("POST", f"/api/v1/mcp/{app_id}",
                    body={"jsonrpc": "2.0", "id": 1,
                          "method": "tools/list", "params": {}})
    for t in (out.get("result", {}) or {}).get("tools", []):
        typer.echo(f"{t['name']}: {t.get('description', '')[:80]}")


@mcp_app.command("call")
def mcp_call(app_id: str, tool: str,
             args: str = typer.Option("{}", help="JSON arguments")):
    c = _client()
    out = c.request("POST", f"/api/v1/mcp/{app_id}",
                    body={"jsonrpc": "2.0", "id": 1,
                          "method": "tools/call",
                          "params": {"name": tool,
                                     "arguments": json.loads(args)}})
    typer.echo(json.dumps(out.get("result", out), indent=2))


evals_app = typer.Typer(help="Evaluation suites")
app.add_typer(evals_app, name="evals")


@evals_app.command("create")
def evals_create(app_id: str, name: str,
                 cases_file: str = typer.Option(..., "-f")):
    with open(cases_file) as fh:
        cases = json.load(fh)
    s = _client().create_evaluation_suite(app_id, name, cases)
    typer.echo(s["id"])


@evals_app.command("run")
def evals_run(suite_id: str):
    r = _client().run_evaluation_suite(suite_id)
    typer.echo(json.dumps(r, indent=2))


@evals_app.command("show")
def evals_show(run_id: str):
    typer.echo(json.dumps(_client().get_evaluation_run(run_id),
                          indent=2))


sandbox_app = typer.Typer(help="Sandboxed workspaces")
app.add_typer(sandbox_app, name="sandbox")


@sandbox_app.command("create")
def sandbox_create(name: str = typer.Option("")):
    typer.echo(_client().create_sandbox(name)["id"])


@sandbox_app.command("list")
def sandbox_list():
    for s in _client().list_sandboxes():
        typer.echo(f"{s['id']}  {s.get('name', '')}  "
                   f"{s.get('state', '')}")


@sandbox_app.command("exec")
def sandbox_exec(sandbox_id: str, command: str,
                 timeout_s: float = typer.Option(60.0)):
    r = _client().sandbox_exec(sandbox_id, command, timeout_s)
    if r["stdout"]:
        typer.echo(r["stdout"], nl=False)
    if r["stderr"]:
        typer.echo(r["stderr"], nl=False, err=True)
    raise typer.Exit(code=0 if r["exit_code"] == 0 else 1)


@sandbox_app.command("rm")
def sandbox_rm(sandbox_id: str):
    _client().delete_sandbox(sandbox_id)
    typer.echo("deleted")


fs_app = typer.Typer(help="Filestore operations")
app.add_typer(fs_app, name="fs")


@fs_app.command("list")
def fs_list(path: str = typer.Argument("")):
    for f in _client().filestore_list(path):
        kind = "d" if f.get("dir") else "f"
        typer.echo(f"{kind}  {f.get('size', 0):>10}  {f['name']}")


@fs_app.command("upload")
def fs_upload(local: str, remote: str = typer.Argument("")):
    with open(local, "rb") as fh:
        _client().filestore_upload(remote or local.split("/")[-1],
                                   fh.read())
    typer.echo("uploaded")


@fs_app.command("rm")
def fs_rm(path: str):
    _client().filestore_delete(path)
    typer.echo("deleted")


user_app = typer.Typer(help="User administration")
app.add_typer(user_app, name="user")


@user_app.command("create")
def user_create(username: str, admin: bool = typer.Option(False)):
    c = _client()
    out = c.request("POST", "/api/v1/users",
                    body={"username": username, "admin": admin})
    typer.echo(f"{out['id']}  api_key={out.get('api_key', '')}")


@user_app.command("list")
def user_list():
    for u in _client().request("GET", "/api/v1/users"):
        typer.echo(f"{u['id']}  {u.get('username', '')}"
                   f"{'  [admin]' if u.get('admin') else ''}")


provider_app = typer.Typer(help="Provider endpoints")
app.add_typer(provider_app, name="provider")


@provider_app.command("list")
def provider_list():
    out = _client().request("GET", "/api/v1/provider-endpoints")
    for pvd in out:
        typer.echo(f"{pvd.get('provider', '')}  "
                   f"{pvd.get('base_url', '')}")


@provider_app.command("add")
def provider_add(provider: str, base_url: str,
                 api_key: str = typer.Option("")):
    _client().request("POST", "/api/v1/provider-endpoints",
                      body={"provider": provider, "base_url": base_url,
                            "api_key": api_key})
    typer.echo("added")


image_app = typer.Typer(help="Image generation (diffusion models)")
app.add_typer(image_app, name="image")


@image_app.command("generate")
def image_generate(prompt: str,
                   model: str = typer.Option("flux-lite"),
                   out: str = typer.Option("out.png", "--out", "-o"),
                   n: int = typer.Option(1),
                   size: str = typer.Option(""),
                   steps: int = typer.Option(0),
                   seed: int = typer.Option(-1)):
    """Generate image(s) and write PNG files."""
    import base64
    resp = _client().images_generate(
        prompt, model=model, n=n, size=size, steps=steps,
        seed=None if seed < 0 else seed)
    for i, item in enumerate(resp.get("data", [])):
        path = out if n == 1 else \
            out.replace(".png", f"-{i}.png") if out.endswith(".png") \
            else f"{out}-{i}.png"
        with open(path, "wb") as f:
            f.write(base64.b64decode(item["b64_json"]))
        typer.echo(path)


billing_app = typer.Typer(help="Billing and usage")
app.add_typer(billing_app, name="billing")


@billing_app.command("show")
def billing_show():
    typer.echo(json.dumps(_client().billing(), indent=2))


@billing_app.command("topup")
def billing_topup(amount_usd: float):
    c = _client()
    out = c.request("POST", "/api/v1/billing/topup-session",
                    body={"amount_usd": amount_usd})
    typer.echo(out.get("url", ""))


if __name__ == "__main__":
    app()
