"""Web dashboard display driver (reference: NiceGUI driver +
nicegui_sections, display_drivers/nicegui.py:535).

The image has no NiceGUI, so this is a self-contained FastAPI/uvicorn app:
``/`` serves a single-page dashboard that polls ``/api/live`` (the same
step-time pipeline + section loaders every other surface uses) and renders
the verdict, per-rank phase table, phase-share bar, memory and node health.
"""

from __future__ import annotations

import logging
import threading
from typing import Optional

from traceml_amd.aggregator.display.base import DisplayDriver

logger = logging.getLogger(__name__)

_PAGE = """<!DOCTYPE html><html><head><meta charset="utf-8">
<title>traceml-amd dashboard</title>
<style>
body{font-family:-apple-system,'Segoe UI',Roboto,sans-serif;background:#111;
     color:#eee;margin:1.5rem auto;max-width:1100px}
h1{font-size:1.2rem;color:#7ab8ff}
.verdict{padding:.8rem 1rem;border-radius:8px;font-weight:600;margin:.6rem 0}
.crit{background:#7a1f1f}.warn{background:#7a5a1f}.info{background:#1f4b7a}
table{border-collapse:collapse;width:100%;font-size:.85rem;margin:.6rem 0}
th,td{border:1px solid #333;padding:4px 8px;text-align:right}
th:first-child,td:first-child{text-align:left}
.bar{display:flex;height:26px;border-radius:4px;overflow:hidden;margin:.4rem 0}
.bar div{height:100%}
.legend{font-size:.75rem;color:#aaa}
h2{font-size:1rem;color:#9ad;margin-top:1.4rem}
.dim{color:#888;font-size:.75rem}
</style></head><body>
<h1>traceml-amd <span class="dim">MI355X training-step profiler</span></h1>
<div id="content">loading…</div>
<script>
const COLORS={input:'#e07b39',h2d:'#8e44ad',forward:'#2d7dd2',
  backward:'#1b998b',optimizer:'#97cc04',ddp_comm:'#d05ce3',residual:'#777'};
async function tick(){
  try{
    const r=await fetch('/api/live');const d=await r.json();
    let html='';
    const diag=d.step_time.diagnosis||{};
    html+=`<div class="verdict ${diag.severity||'info'}">${diag.status||'…'} — ${diag.summary||''}</div>`;
    const shares=d.step_time.shares||{};
    let bar='<div class="bar">';let legend='';
    for(const [k,v] of Object.entries(shares)){
      if(v&&COLORS[k]){bar+=`<div style="width:${Math.min(100,v*100)}%;background:${COLORS[k]}" title="${k} ${(v*100).toFixed(1)}%"></div>`;
        legend+=`<span style="color:${COLORS[k]}">■</span> ${k} ${(v*100).toFixed(0)}%  `;}}
    html+=bar+'</div><div class="legend">'+legend+'</div>';
    const ranks=d.step_time.ranks||{};
    const metrics=['step_time_ms','input_wait_ms','h2d_ms','forward_ms',
                   'backward_ms','optimizer_ms','ddp_comm_ms','residual_ms'];
    if(Object.keys(ranks).length){
      html+='<h2>Step time (ms) by rank</h2><table><tr><th>metric</th>';
      for(const r of Object.keys(ranks)) html+=`<th>r${r}</th>`;
      html+='</tr>';
      for(const m of metrics){
        let any=false,row=`<tr><td>${m.replace('_ms','')}</td>`;
        for(const r of Object.keys(ranks)){
          const v=ranks[r][m];row+=`<td>${v==null?'—':v.toFixed(1)}</td>`;
          if(v!=null)any=true;}
        if(any)html+=row+'</tr>';}
      html+='</table>';
      html+=`<div class="dim">${d.step_time.steps_analyzed} aligned steps · ${d.step_time.clock} clock · ${d.step_time.strategy}</div>`;
    }
    if(d.memory&&Object.keys(d.memory).length){
      html+='<h2>Peak memory</h2><table><tr><th>rank</th><th>allocated</th><th>reserved</th><th style="text-align:left">of capacity</th></tr>';
      for(const [r,m] of Object.entries(d.memory)){
        const cap=m.capacity||288*2**30;
        const pct=m.reserved?Math.min(100,m.reserved/cap*100):0;
        const barColor=pct>92?'#d9534f':(pct>80?'#f0ad4e':'#1b998b');
        html+=`<tr><td>r${r}</td><td>${m.alloc==null?'—':(m.alloc/2**30).toFixed(1)+' GiB'}</td><td>${m.reserved==null?'—':(m.reserved/2**30).toFixed(1)+' GiB'}</td>`+
          `<td style="text-align:left;min-width:180px"><div class="bar" style="height:12px;background:#222"><div style="width:${pct}%;background:${barColor}"></div></div><span class="dim">${pct.toFixed(0)}% of ${(cap/2**30).toFixed(0)} GiB</span></td></tr>`;}
      html+='</table>';}
    if(d.history&&Object.keys(d.history).length){
      const ranksH=Object.keys(d.history);
      let allPts=[];for(const r of ranksH)allPts=allPts.concat(d.history[r]);
      if(allPts.length>4){
        const xs=allPts.map(p=>p[0]),ys=allPts.map(p=>p[1]);
        const x0=Math.min(...xs),x1=Math.max(...xs),y1=Math.max(...ys)*1.05||1;
        const W=920,H=110;
        const colors=['#7ab8ff','#1b998b','#e07b39','#d05ce3','#97cc04','#f0ad4e','#e85d75','#9ad'];
        let svg=`<svg width="${W}" height="${H+18}" xmlns="http://www.w3.org/2000/svg">`;
        svg+=`<line x1="0" y1="${H}" x2="${W}" y2="${H}" stroke="#333"/>`;
        ranksH.forEach((r,i)=>{
          const pts=d.history[r].map(p=>
            `${((p[0]-x0)/Math.max(1,x1-x0)*W).toFixed(1)},${(H-p[1]/y1*H).toFixed(1)}`).join(' ');
          svg+=`<polyline points="${pts}" fill="none" stroke="${colors[i%8]}" stroke-width="1.5"/>`;});
        svg+=`<text x="4" y="12" font-size="10" fill="#888">step time ms (0–${y1.toFixed(1)}), steps ${x0}–${x1}</text>`;
        svg+=ranksH.map((r,i)=>`<text x="${70+i*46}" y="${H+14}" font-size="10" fill="${colors[i%8]}">r${r}</text>`).join('');
        svg+='</svg>';
        html+='<h2>Step-time history</h2>'+svg;
      }
    }
    if(d.comm&&d.comm.ranks&&d.comm.ranks.length>1){
      html+='<h2>RCCL rank stats (xGMI all-gather)</h2><table><tr><th>rank</th><th>step</th><th>input ms</th><th>bwd ms</th><th>ddp comm ms</th></tr>';
      for(const r of d.comm.ranks)
        html+=`<tr><td>r${r.rank}</td><td>${r.step}</td><td>${r.input_ms.toFixed(1)}</td><td>${r.backward_ms.toFixed(1)}</td><td>${r.ddp_comm_ms.toFixed(1)}</td></tr>`;
      html+='</table>';}
    if(d.issues&&d.issues.length){
      html+='<h2>Findings</h2>';
      for(const i of d.issues.slice(0,8)){
        if(i.kind==='NORMAL'||i.kind==='BALANCED'||i.kind==='NO_DATA'||i.kind==='NO_GPU') continue;
        html+=`<div class="verdict ${i.severity}" style="font-weight:400;padding:.4rem .8rem;margin:.3rem 0">`+
          `<b>${i.status}</b> <span class="dim">[${i.section}]</span> ${i.summary}</div>`;}
    }
    if(d.stdout&&d.stdout.length){
      html+='<h2>Console (rank 0)</h2><div style="background:#000;border-radius:6px;padding:.6rem;font-family:monospace;font-size:.75rem">';
      for(const l of d.stdout)
        html+=`<div style="color:${l.stream==='stderr'?'#e08':'#9d9'}">${l.line.replace(/</g,'&lt;')}</div>`;
      html+='</div>';}
    if(d.system&&d.system.gpus&&Object.keys(d.system.gpus).length){
      html+='<h2>GPUs (amdsmi)</h2><table><tr><th>gpu</th><th>util %</th><th>VRAM GiB</th><th>temp °C</th><th>power W</th></tr>';
      for(const [g,m] of Object.entries(d.system.gpus))
        html+=`<tr><td>${g}</td><td>${m.util==null?'—':m.util.toFixed(0)}</td><td>${m.mem_used==null?'—':(m.mem_used/2**30).toFixed(0)}</td><td>${m.temp==null?'—':m.temp.toFixed(0)}</td><td>${m.power==null?'—':m.power.toFixed(0)}</td></tr>`;
      html+='</table>';}
    document.getElementById('content').innerHTML=html;
  }catch(e){document.getElementById('content').innerHTML='<div class="dim">aggregator not ready…</div>';}
}
setInterval(tick,1500);tick();
</script></body></html>"""


def _live_payload(db_path: str, session=None) -> dict:
    from traceml_amd.renderers import live_view

    return live_view(db_path, session=session)


class DashboardDisplayDriver(DisplayDriver):
    def __init__(self, port: int = 8765) -> None:
        self.port = port
        self._server = None
        self._thread: Optional[threading.Thread] = None
        self._db_path: Optional[str] = None
        self._session = None

    def start(self) -> None:
        try:
            import uvicorn
            from fastapi import FastAPI
            from fastapi.responses import HTMLResponse, JSONResponse
        except Exception:
            logger.warning(
                "traceml_amd: fastapi/uvicorn unavailable, dashboard disabled"
            )
            return

        app = FastAPI()

        @app.get("/")
        def index():
            return HTMLResponse(_PAGE)

        @app.get("/api/live")
        def live():
            if self._db_path is None:
                return JSONResponse({}, status_code=503)
            try:
                if self._session is None:
                    from traceml_amd.steptime.pipeline import LiveStepTimeSession

                    self._session = LiveStepTimeSession(self._db_path)
                return JSONResponse(_live_payload(self._db_path, self._session))
            except Exception as exc:
                return JSONResponse({"error": repr(exc)}, status_code=500)

        config = uvicorn.Config(
            app, host="0.0.0.0", port=self.port, log_level="error"
        )
        self._server = uvicorn.Server(config)
        self._thread = threading.Thread(
            target=self._server.run, name="traceml-dashboard", daemon=True
        )
        self._thread.start()
        print(f"[TraceML-AMD] dashboard at http://localhost:{self.port}", flush=True)

    def render_tick(self, db_path: str) -> None:
        self._db_path = db_path  # the page polls; nothing to push

    def stop(self) -> None:
        if self._server is not None:
            self._server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout=3.0)
