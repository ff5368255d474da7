"""Metrics dashboard: HTTP server + SQLite store + in-browser chart.

Reference: dolphin/dashboard/DashboardLauncher.java:31-63 copies and spawns
resources/dashboard/dashboard.py (Flask + SQLite + plotly; schema.sql has
worker/server metric tables); DashboardConnector.java:44 HTTP-POSTs metric
JSON. Here the server is stdlib http.server + sqlite3 (no web framework
needed), the chart is a self-contained inline-SVG page, and the connector
is the same POST protocol.

  server = DashboardServer(port=0).start()      # returns bound port
  DashboardConnector(f"http://127.0.0.1:{port}").send(job_id, rank, metrics)
"""

from __future__ import annotations

import json
import sqlite3
import threading
import urllib.request
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

_SCHEMA = """
CREATE TABLE IF NOT EXISTS metrics (
  id INTEGER PRIMARY KEY AUTOINCREMENT,
  time REAL, job_id TEXT, rank INTEGER, kind TEXT, payload TEXT
);
"""

_PAGE = """<!doctype html><html><head><title>harmony_amd dashboard</title>
<style>body{font-family:monospace;margin:2em}svg{border:1px solid #ccc}
.m{margin-bottom:1.5em}a{margin-right:1em}</style></head><body>
<h2>harmony_amd — job metrics</h2><div id="jobs"></div>
<div id="root">loading...</div>
<script>
const sel=new URLSearchParams(location.search).get('job');
// job ids arrive via unauthenticated POST /metrics: escape before any
// innerHTML/href interpolation (stored-XSS hardening)
const esc=s=>String(s).replace(/[&<>"']/g,c=>({'&':'&amp;','<':'&lt;',
  '>':'&gt;','"':'&quot;',"'":'&#39;'}[c]));
fetch('/jobs').then(r=>r.json()).then(jobs=>{
  document.getElementById('jobs').innerHTML='jobs: '+
    ['<a href="/">all</a>'].concat(jobs.map(j=>
      `<a href="/?job=${encodeURIComponent(j.job_id)}">${esc(j.job_id)}</a>`+
      ` (${j.reports|0} reports, `+
      `${new Date(j.t0*1000).toISOString().slice(0,19)})`)).join(' | ');
});
fetch('/data'+(sel?'?job='+sel:'')).then(r=>r.json()).then(rows=>{
  const byJob={};
  rows.forEach(r=>{(byJob[r.job_id] ||= []).push(r);});
  const chart=(pts,color)=>{
    const W=600,H=120,mx=Math.max(...pts,1e-12);
    const poly=pts.map((v,i)=>`${i*W/Math.max(pts.length-1,1)},${H-v/mx*H}`).join(' ');
    return `<svg width=${W} height=${H}><polyline fill=none stroke=${color}
      stroke-width=2 points="${poly}"/></svg> max ${mx.toFixed(2)}`;
  };
  let html='';
  for (const [job,rs] of Object.entries(byJob)) {
    const get=k=>rs.map(r=>{const p=JSON.parse(r.payload);return p[k]||0;});
    const rate=get('data_processing_rate'), et=get('epoch_time_sec');
    html+=`<div class=m><b>${esc(job)}</b> (${rs.length} reports)<br>
      rate ex/s: ${chart(rate,'steelblue')}<br>`+
      (et.some(v=>v>0)?`epoch sec: ${chart(et,'indianred')}`:'')+`</div>`;
  }
  document.getElementById('root').innerHTML=html||'no metrics yet';
});
</script></body></html>"""


class _Handler(BaseHTTPRequestHandler):
    db_path = ""

    def log_message(self, *a):  # quiet
        pass

    def _db(self):
        con = sqlite3.connect(self.db_path)
        con.execute(_SCHEMA)
        return con

    def do_GET(self):
        if self.path.startswith("/jobs"):
            # historical browsing: every job ever reported to this db file
            con = self._db()
            rows = con.execute(
                "SELECT job_id, COUNT(*), MIN(time), MAX(time) FROM metrics "
                "GROUP BY job_id ORDER BY MIN(id)").fetchall()
            body = json.dumps([
                {"job_id": j, "reports": n, "t0": t0, "t1": t1}
                for j, n, t0, t1 in rows]).encode()
            ct = "application/json"
        elif self.path.startswith("/data"):
            from urllib.parse import parse_qs, urlparse

            q = parse_qs(urlparse(self.path).query)
            con = self._db()
            if "job" in q:
                rows = con.execute(
                    "SELECT time, job_id, rank, kind, payload FROM metrics "
                    "WHERE job_id = ? ORDER BY id", (q["job"][0],)).fetchall()
            else:
                rows = con.execute(
                    "SELECT time, job_id, rank, kind, payload FROM metrics "
                    "ORDER BY id").fetchall()
            body = json.dumps([
                {"time": t, "job_id": j, "rank": r, "kind": k, "payload": p}
                for t, j, r, k, p in rows]).encode()
            ct = "application/json"
        else:
            body = _PAGE.encode()
            ct = "text/html"
        self.send_response(200)
        self.send_header("Content-Type", ct)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_POST(self):
        n = int(self.headers.get("Content-Length", 0))
        rec = json.loads(self.rfile.read(n))
        con = self._db()
        con.execute("INSERT INTO metrics (time, job_id, rank, kind, payload) "
                    "VALUES (?,?,?,?,?)",
                    (rec.get("time", 0.0), rec.get("job_id", ""),
                     rec.get("rank", 0), rec.get("kind", "worker"),
                     json.dumps(rec.get("payload", {}))))
        con.commit()
        self.send_response(200)
        self.send_header("Content-Length", "2")
        self.end_headers()
        self.wfile.write(b"ok")


class DashboardServer:
    def __init__(self, port: int = 0, db_path: str = "/tmp/harmony_dashboard.db"):
        self.port = port
        self.db_path = db_path
        self._srv: Optional[ThreadingHTTPServer] = None

    def start(self) -> int:
        handler = type("H", (_Handler,), {"db_path": self.db_path})
        self._srv = ThreadingHTTPServer(("127.0.0.1", self.port), handler)
        self.port = self._srv.server_address[1]
        threading.Thread(target=self._srv.serve_forever, daemon=True).start()
        return self.port

    def stop(self) -> None:
        if self._srv:
            self._srv.shutdown()


class DashboardConnector:
    """HTTP-POST metric sender (reference DashboardConnector.java:44)."""

    def __init__(self, url: str):
        self.url = url.rstrip("/")

    def send(self, job_id: str, rank: int, payload: dict,
             kind: str = "worker", t: float = 0.0) -> bool:
        try:
            data = json.dumps({"job_id": job_id, "rank": rank, "kind": kind,
                               "time": t, "payload": payload}).encode()
            req = urllib.request.Request(self.url + "/metrics", data=data,
                                         headers={"Content-Type":
                                                  "application/json"})
            with urllib.request.urlopen(req, timeout=5) as resp:
                return resp.status == 200
        except Exception:  # noqa: BLE001
            return False
