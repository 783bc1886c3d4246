"""Dashboard — single-page cluster/jobs/services view served by the API
server (reference: sky/dashboard/, a 47k-LoC Next.js app; here a
server-rendered page with auto-refresh keeps the surface without the
node toolchain)."""
from __future__ import annotations

import html
import time
from typing import Any, Dict, List


class _Raw(str):
    """Cell rendered without escaping (pre-escaped HTML, e.g. links)."""


def _link(href: str, text: str) -> "_Raw":
    return _Raw(f'<a href="{html.escape(href)}" style="color:#e8443a">'
                f"{html.escape(str(text))}</a>")


def _cell(c) -> str:
    return str(c) if isinstance(c, _Raw) else html.escape(str(c))


def _table(title: str, headers: List[str], rows: List[List[Any]]) -> str:
    if not rows:
        body = f"<tr><td colspan={len(headers)} class=empty>none</td></tr>"
    else:
        body = "".join(
            "<tr>" + "".join(f"<td>{_cell(c)}</td>" for c in row)
            + "</tr>" for row in rows)
    head = "".join(f"<th>{h}</th>" for h in headers)
    return (f"<h2>{title}</h2><table><thead><tr>{head}</tr></thead>"
            f"<tbody>{body}</tbody></table>")


def _sparkline(values, width=560, height=60) -> str:
    """Inline SVG sparkline (no JS, no chart deps)."""
    if not values or max(values) == 0:
        return "<p class=empty>no activity yet</p>"
    mx = max(values)
    n = len(values)
    bar_w = max(2, width // n - 1)
    bars = []
    for i, v in enumerate(values):
        h = int(v / mx * (height - 4))
        bars.append(
            f'<rect x="{i * (bar_w + 1)}" y="{height - h}" '
            f'width="{bar_w}" height="{h}" fill="#e8443a"/>')
    return (f'<svg width="{width}" height="{height + 14}">'
            + "".join(bars)
            + f'<text x="0" y="{height + 12}" fill="#666" '
              f'font-size="10">last {n} minutes; peak {mx}/min</text>'
            "</svg>")


def render() -> str:
    from skypilot_amd import global_state
    from skypilot_amd.jobs import state as jobs_state
    from skypilot_amd.serve import serve_state
    from skypilot_amd.server import requests_db as rdb
    from skypilot_amd.utils.gpu_topology import detect_gpus

    clusters = []
    for c in global_state.list_clusters():
        h = c["handle"]
        clusters.append([
            _link(f"/dashboard/cluster/{c['name']}", c["name"]),
            c["status"],
            f"{h.get('num_nodes', 1)}x{h.get('gpus_per_node', 0)} GPU",
            ",".join(map(str, h.get("gpu_ids", []))) or "-",
            time.strftime("%m-%d %H:%M",
                          time.localtime(c.get("launched_at") or 0)),
        ])
    jobs = [[j["job_id"], j.get("name") or "-", j["status"],
             j.get("recovery_count", 0), j.get("cluster_name") or "-"]
            for j in jobs_state.list_jobs()]
    services = []
    for s in serve_state.list_services():
        if s is None:
            continue
        reps = serve_state.list_replicas(s["name"])
        ready = sum(1 for r in reps if r["status"] == "READY")
        services.append([s["name"], s["status"], f"{ready}/{len(reps)}",
                         f"http://127.0.0.1:{s['lb_port']}"])
    gpus_taken: Dict[int, str] = {}
    for c in global_state.list_clusters():
        if c["status"] == "UP":
            for g in c["handle"].get("gpu_ids", []):
                gpus_taken[g] = c["name"]
    gpus = [[g.index, g.name, f"{g.memory_gb} GB", g.numa_node,
             gpus_taken.get(g.index, "-")] for g in detect_gpus()]

    # worker pools + storage + cost (reference dashboard: infra tabs)
    pool_rows = []
    try:
        from skypilot_amd.jobs import pools as jobs_pools
        for prec in jobs_pools.status():
            busy = sum(1 for w_ in prec["workers"]
                       if w_["status"] == "BUSY")
            pool_rows.append([prec["name"], len(prec["workers"]), busy,
                              ",".join(w_["cluster_name"]
                                       for w_ in prec["workers"][:4])])
    except Exception:  # noqa: BLE001
        pass
    storage_rows = []
    try:
        from skypilot_amd.data import storage as storage_lib
        for srec in storage_lib.list_storage():
            storage_rows.append([srec["name"], srec["store_type"],
                                 srec["source"] or "-"])
    except Exception:  # noqa: BLE001
        pass
    cost_rows = []
    try:
        from skypilot_amd import core as _core
        for crec in _core.cost_report()[:8]:
            cost_rows.append([crec["name"],
                              "live" if crec.get("live") else "done",
                              f"{crec.get('gpu_hours', 0):.2f}",
                              f"{crec.get('duration_hours', 0):.2f}"])
    except Exception:  # noqa: BLE001
        pass

    # recent API requests (reference dashboard: requests table)
    reqs = []
    for r in rdb.list_requests(limit=15):
        dur = ""
        if r.get("started_at"):
            end = r.get("finished_at") or time.time()
            dur = f"{end - r['started_at']:.1f}s"
        reqs.append([r["request_id"][:8], r["name"],
                     r.get("user") or "-", r["status"], dur,
                     time.strftime("%H:%M:%S",
                                   time.localtime(r["created_at"]))])

    # requests-per-minute time series (reference dashboard charts)
    now = time.time()
    buckets = [0] * 30
    for r in rdb.list_requests(limit=1000):
        age_min = int((now - (r["created_at"] or now)) / 60)
        if 0 <= age_min < 30:
            buckets[29 - age_min] += 1
    spark = _sparkline(buckets)

    # teardown history (reference dashboard: cluster history view)
    hist = [[h["name"],
             time.strftime("%m-%d %H:%M",
                           time.localtime(h.get("launched_at") or 0)),
             time.strftime("%m-%d %H:%M",
                           time.localtime(h.get("torn_down_at") or 0)),
             f"{((h.get('torn_down_at') or 0) - (h.get('launched_at') or 0)) / 60:.0f} min"]
            for h in global_state.list_cluster_history(limit=10)]

    # users + workspaces (reference dashboard: users / workspaces tabs)
    user_rows = []
    try:
        from skypilot_amd import users as users_lib
        user_rows = [[u["name"], u["role"]]
                     for u in users_lib.list_users()]
    except Exception:  # noqa: BLE001
        pass
    ws_rows = []
    try:
        seen = {}
        for c in global_state.list_clusters():
            w = c.get("workspace") or "default"
            seen[w] = seen.get(w, 0) + 1
        from skypilot_amd import config as sky_config
        for wname, wcfg in (sky_config.get_nested(["workspaces"], {})
                            or {}).items():
            seen.setdefault(wname, 0)
        ws_rows = [[w, n, "private" if (sky_config.get_nested(
            ["workspaces", w, "private"], False)) else "open"]
            for w, n in sorted(seen.items())]
    except Exception:  # noqa: BLE001
        pass

    return f"""<!doctype html><html><head><title>skypilot-amd</title>
<meta http-equiv="refresh" content="5">
<style>
body {{ font-family: ui-monospace, monospace; margin: 2em; background:#111;
        color:#ddd; }}
h1 {{ color:#e8443a; }} h2 {{ color:#ccc; margin-top:1.4em; }}
table {{ border-collapse: collapse; min-width: 48em; }}
th, td {{ border:1px solid #333; padding:4px 10px; text-align:left; }}
th {{ background:#222; color:#e8443a; }}
.empty {{ color:#666; }}
</style></head><body>
<h1>skypilot-amd <small style="color:#666">MI355X pool</small></h1>
{_table("Clusters", ["name", "status", "shape", "gpus", "launched"],
        clusters)}
{_table("Managed jobs", ["id", "name", "status", "recoveries", "cluster"],
        jobs)}
{_table("Services", ["name", "status", "ready", "endpoint"], services)}
{_table("Pool GPUs", ["idx", "model", "HBM", "numa", "used by"], gpus)}
{_table("Worker pools", ["name", "workers", "busy", "clusters"],
        pool_rows)}
{_table("Storage", ["name", "type", "source"], storage_rows)}
{_table("Cost report (GPU-hours)", ["cluster", "state", "gpu-hours",
        "hours"], cost_rows)}
<h2>Request activity</h2>{spark}
{_table("Recent API requests",
        ["id", "type", "user", "status", "duration", "at"], reqs)}
{_table("Cluster history", ["name", "launched", "torn down", "lifetime"],
        hist)}
{_table("Users", ["name", "role"], user_rows)}
{_table("Workspaces", ["name", "clusters", "access"], ws_rows)}
<p style="color:#555">auto-refreshes every 5s · {time.strftime("%H:%M:%S")}
</p></body></html>"""


_STYLE = """
<style>
body { font-family: ui-monospace, monospace; margin: 2em; background:#111;
       color:#ddd; }
h1 { color:#e8443a; } h2 { color:#ccc; margin-top:1.4em; }
table { border-collapse: collapse; min-width: 48em; }
th, td { border:1px solid #333; padding:4px 10px; text-align:left; }
th { background:#222; color:#e8443a; }
.empty { color:#666; } a { color:#e8443a; }
pre { background:#1a1a1a; padding:1em; overflow-x:auto; }
</style>"""


def render_cluster(name: str) -> str:
    """Per-cluster drill-down: handle, job queue, events, agent health
    (reference: the dashboard's cluster detail page)."""
    from skypilot_amd import core, global_state
    rec = global_state.get_cluster(name)
    if rec is None:
        return (f"<!doctype html><html><head>{_STYLE}</head><body>"
                f"<h1>no cluster {html.escape(name)!s}</h1>"
                '<p><a href="/dashboard">back</a></p></body></html>')
    h = rec.get("handle") or {}
    handle_rows = [[k, h[k]] for k in sorted(h)
                   if k not in ("agent_token",)]  # never render secrets
    try:
        jobs = [[j["job_id"], j.get("name") or "-", j["status"],
                 ",".join(map(str, (j.get("spec") or {}).get("gpu_ids",
                                                             []))) or "-",
                 _link(f"/dashboard/cluster/{name}/job/{j['job_id']}",
                       "logs")]
                for j in core.queue(name)]
    except Exception as e:  # noqa: BLE001 — agent may be down
        jobs = [["-", "-", f"unreachable: {e}", "-", "-"]]
    events = [[time.strftime("%m-%d %H:%M:%S",
                             time.localtime(e.get("ts") or 0)),
               e.get("event"), e.get("detail") or "-"]
              for e in core.cluster_events(name)[-25:]]
    agent = "no agent"
    if h.get("agent_port"):
        from skypilot_amd.agent.client import AgentClient
        c = AgentClient(h["agent_port"], token=h.get("agent_token"),
                        timeout=3.0)
        try:
            agent = "healthy" if c.healthy() else "UNREACHABLE"
        finally:
            c.close()
    return f"""<!doctype html><html><head>
<title>{html.escape(name)} — skypilot-amd</title>{_STYLE}</head><body>
<h1>{html.escape(name)} <small style="color:#666">{rec["status"]} ·
agent {agent}</small></h1>
<p><a href="/dashboard">← all clusters</a></p>
{_table("Handle", ["key", "value"], handle_rows)}
{_table("Job queue", ["id", "name", "status", "gpus", ""], jobs)}
{_table("Events", ["at", "event", "detail"], events)}
</body></html>"""


def render_job_logs(cluster_name: str, job_id: int, tail: int = 400) -> str:
    """Job log view (reference: dashboard job logs page)."""
    from skypilot_amd import core
    lines: List[str] = []
    try:
        for chunk in core.tail_logs(cluster_name, job_id, follow=False):
            lines.append(chunk.decode(errors="replace"))
            if sum(len(x) for x in lines) > 1 << 20:
                break
    except Exception as e:  # noqa: BLE001
        lines = [f"(logs unavailable: {e})"]
    text = "".join(lines)
    tail_text = "\n".join(text.splitlines()[-tail:])
    return f"""<!doctype html><html><head>
<title>job {job_id} — {html.escape(cluster_name)}</title>{_STYLE}</head>
<body><h1>job {job_id} on {html.escape(cluster_name)}</h1>
<p><a href="/dashboard/cluster/{html.escape(cluster_name)}">← cluster</a></p>
<pre>{html.escape(tail_text)}</pre></body></html>"""
