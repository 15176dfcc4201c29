"""Web dashboard display driver (reference: NiceGUI driver +
nicegui_sections, display_drivers/nicegui.py:535 + ~1,900 LoC of section
widgets).

The image has no NiceGUI, so this is a self-contained FastAPI/uvicorn app:
``/`` serves a single-page dashboard that polls ``/api/live`` and renders
the per-section view models from ``traceml_amd.renderers`` — the SAME
payloads the CLI and HTML report use, so surfaces cannot drift apart.
Sections (mirroring the reference's section set): hero step-time (verdict,
stacked phase-share bar, per-rank phase table with cohort badges, skew
callout, history chart), per-rank memory cards (capacity bars, trend,
overhang), node-health GPU cards (amdsmi util/VRAM/temp/power with bands),
process table (RSS/CPU/GPU mem/self-overhead), RCCL/xGMI comm card
(gather latency + per-rank skew), findings and the rank-0 console.
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
     color:#eee;margin:1.5rem auto;max-width:1160px;padding:0 1rem}
h1{font-size:1.2rem;color:#7ab8ff}
h2{font-size:1rem;color:#9ad;margin:1.4rem 0 .4rem}
.verdict{padding:.8rem 1rem;border-radius:8px;font-weight:600;margin:.6rem 0}
.crit{background:#7a1f1f}.warn{background:#7a5a1f}.info{background:#1f4b7a}
.ok{background:#1f4b2a}
table{border-collapse:collapse;width:100%;font-size:.85rem;margin:.6rem 0}
th,td{border:1px solid #333;padding:4px 8px;text-align:right}
th:first-child,td:first-child{text-align:left}
.bar{display:flex;height:26px;border-radius:4px;overflow:hidden;margin:.4rem 0}
.bar div{height:100%}
.legend{font-size:.75rem;color:#aaa}
.dim{color:#888;font-size:.75rem}
.cards{display:flex;flex-wrap:wrap;gap:.6rem}
.card{background:#1a1a22;border:1px solid #333;border-radius:8px;
      padding:.6rem .8rem;min-width:200px;flex:1}
.card .t{font-size:.8rem;color:#9ad;margin-bottom:.3rem}
.card .v{font-size:1.05rem;font-weight:600}
.meter{height:10px;background:#222;border-radius:4px;overflow:hidden;margin:.3rem 0}
.meter div{height:100%}
.b-ok{color:#7c6}.b-warn{color:#f0ad4e}.b-crit{color:#e55}.b-low{color:#f0ad4e}
.b-moderate{color:#dd6}
.badge{font-size:.68rem;border-radius:4px;padding:1px 5px;margin-left:4px}
.badge.slow{background:#7a1f1f}.badge.fast{background:#1f4b7a}
.badge.typical{background:#333}
</style></head><body>
<h1>traceml-amd <span class="dim">MI355X training-step profiler</span></h1>
<div id="content">loading…</div>
<script>
const PHASE_COLORS={input:'#e07b39',h2d:'#8e44ad',forward:'#2d7dd2',
  backward:'#1b998b',optimizer:'#97cc04',ddp_comm:'#d05ce3',residual:'#777'};
const BAND={ok:'#1b998b',warn:'#f0ad4e',crit:'#d9534f',low:'#f0ad4e',
  moderate:'#dd6'};
const esc=s=>String(s).replace(/&/g,'&amp;').replace(/</g,'&lt;')
  .replace(/>/g,'&gt;').replace(/"/g,'&quot;');
const fm=(v,d=1)=>v==null?'—':Number(v).toFixed(d);

function meter(frac,bandName){
  const pct=frac==null?0:Math.min(100,frac*100);
  const color=BAND[bandName]||'#2d7dd2';
  return `<div class="meter"><div style="width:${pct}%;background:${color}"></div></div>`;
}

function stepTimeSection(st){
  if(!st) return '';
  let html='';
  const diag=st.diagnosis||{};
  html+=`<div class="verdict ${diag.severity||'info'}">${esc(diag.status||'…')} — ${esc(diag.summary||'')}</div>`;
  if(diag.action) html+=`<div class="dim">→ ${esc(diag.action)}</div>`;
  if(st.shares&&st.shares.length){
    let bar='<div class="bar">',legend='';
    for(const s of st.shares){
      const c=PHASE_COLORS[s.phase]||'#777';
      bar+=`<div style="width:${Math.min(100,s.fraction*100)}%;background:${c}" title="${s.phase} ${(s.fraction*100).toFixed(1)}%"></div>`;
      legend+=`<span style="color:${c}">■</span> ${s.phase} ${(s.fraction*100).toFixed(0)}%  `;
    }
    html+=bar+'</div><div class="legend">'+legend+'</div>';
  }
  if(st.skew&&st.skew.skew_fraction>0.05){
    html+=`<div class="dim">rank skew: r${st.skew.worst_rank} at ${fm(st.skew.worst_ms)} ms vs median ${fm(st.skew.median_ms)} ms (+${(st.skew.skew_fraction*100).toFixed(0)}%)</div>`;
  }
  if(st.rows&&st.rows.length&&st.ranks.length){
    html+='<h2>Step time (ms) by rank</h2><table><tr><th>metric</th>';
    for(const r of st.ranks){
      const cohort=st.cohorts&&st.cohorts[r];
      const badge=cohort&&cohort!=='typical'?`<span class="badge ${cohort}">${cohort}</span>`:'';
      html+=`<th>r${r}${badge}</th>`;
    }
    html+='</tr>';
    for(const row of st.rows){
      let tr=`<tr><td>${esc(row.label)}</td>`;
      for(const r of st.ranks){
        const cell=row.cells[r]||{};
        tr+=`<td>${cell.text==null?'—':cell.text}</td>`;
      }
      html+=tr+'</tr>';
    }
    html+='</table>';
  }
  const f=st.footer||{};
  html+=`<div class="dim">${f.steps_analyzed||0} aligned steps · ${f.clock||'?'} clock · strategy ${f.strategy||'?'}</div>`;
  html+=historyChart(st.history);
  return html;
}

function historyChart(history){
  if(!history) return '';
  const ranks=Object.keys(history);
  let allPts=[];for(const r of ranks)allPts=allPts.concat(history[r]);
  if(allPts.length<5) return '';
  const xs=allPts.map(p=>p[0]),ys=allPts.map(p=>p[1]);
  const x0=Math.min(...xs),x1=Math.max(...xs),y1=Math.max(...ys)*1.05||1;
  const W=980,H=110;
  const colors=['#7ab8ff','#1b998b','#e07b39','#d05ce3','#97cc04','#f0ad4e','#e85d75','#9ad'];
  let svg=`<svg width="${W}" height="${H+18}" xmlns="http://www.w3.org/2000/svg">`;
  svg+=`<line x1="0" y1="${H}" x2="${W}" y2="${H}" stroke="#333"/>`;
  ranks.forEach((r,i)=>{
    const pts=history[r].map(p=>
      `${((p[0]-x0)/Math.max(1,x1-x0)*W).toFixed(1)},${(H-p[1]/y1*H).toFixed(1)}`).join(' ');
    svg+=`<polyline points="${pts}" fill="none" stroke="${colors[i%8]}" stroke-width="1.5"/>`;});
  svg+=`<text x="4" y="12" font-size="10" fill="#888">step time ms (0–${y1.toFixed(1)}), steps ${x0}–${x1}</text>`;
  svg+=ranks.map((r,i)=>`<text x="${70+i*46}" y="${H+14}" font-size="10" fill="${colors[i%8]}">r${r}</text>`).join('');
  return '<h2>Step-time history</h2>'+svg+'</svg>';
}

function memorySpark(spark){
  if(!spark||spark.length<4) return '';
  const W=170,H=26;
  const xs=spark.map(p=>p[0]),ys=spark.map(p=>p[1]);
  const x0=Math.min(...xs),x1=Math.max(...xs);
  const y0=Math.min(...ys),y1=Math.max(...ys);
  const span=Math.max(1,y1-y0);
  const pts=spark.map(p=>
    `${((p[0]-x0)/Math.max(1,x1-x0)*W).toFixed(1)},${(H-(p[1]-y0)/span*H).toFixed(1)}`
  ).join(' ');
  const rising=(y1-y0)/Math.max(1,y0)>0.02;
  return `<svg width="${W}" height="${H}" xmlns="http://www.w3.org/2000/svg">`+
    `<polyline points="${pts}" fill="none" stroke="${rising?'#f0ad4e':'#1b998b'}" stroke-width="1.2"/></svg>`;
}

function memorySection(sm){
  if(!sm||!sm.available||!sm.cards.length) return '';
  let html='<h2>Peak memory (HIP caching allocator)</h2><div class="cards">';
  for(const c of sm.cards){
    const trend=c.trend_bytes_per_step;
    const trendTxt=trend==null?'':(trend>1024?
      `<div class="dim">trend +${(trend/1048576).toFixed(2)} MiB/step</div>`:'');
    const spark=memorySpark(c.spark);
    const over=c.overhang_ratio!=null&&c.overhang_ratio>=2?
      `<div class="b-warn dim">reserved ${c.overhang_ratio.toFixed(1)}x allocated</div>`:'';
    html+=`<div class="card"><div class="t">rank ${c.rank}</div>
      <div class="v">${c.peak_alloc_gib==null?'—':c.peak_alloc_gib+' GiB'} <span class="dim">alloc</span></div>
      <div class="dim">${c.peak_reserved_gib==null?'—':c.peak_reserved_gib+' GiB reserved of '+(c.capacity_gib||'?')+' GiB</div>'}
      ${meter(c.pressure_fraction,c.pressure_band||'ok')}
      <div class="dim">${c.pressure_fraction==null?'':(c.pressure_fraction*100).toFixed(0)+'% of capacity · '+c.steps_observed+' steps'}</div>
      ${spark}${trendTxt}${over}</div>`;
  }
  return html+'</div>';
}

function systemSection(sys){
  if(!sys||!sys.available) return '';
  let html='<h2>Node health (amdsmi)</h2><div class="cards">';
  const h=sys.host||{};
  html+=`<div class="card"><div class="t">host</div>
    <div>cpu <span class="b-${h.cpu_band||'ok'}">${fm(h.cpu_percent,0)}%</span> ·
    ram <span class="b-${h.ram_band||'ok'}">${fm(h.ram_percent,0)}%</span>
    <span class="dim">of ${h.ram_total_gib||'?'} GiB</span></div></div>`;
  for(const g of sys.gpus||[]){
    html+=`<div class="card"><div class="t">GPU ${g.gpu}</div>
      <div>util <span class="b-${g.util_band||'ok'}">${fm(g.util_percent,0)}%</span> ·
      <span class="b-${g.temp_band||'ok'}">${fm(g.temp_c,0)}°C</span> ·
      <span class="b-${g.power_band||'ok'}">${fm(g.power_w,0)} W</span>
      <span class="dim">/ ${fm(g.power_cap_w,0)} W</span></div>
      ${meter(g.mem_fraction,g.mem_band||'ok')}
      <div class="dim">VRAM ${g.mem_used_gib==null?'—':g.mem_used_gib+' / '+(g.mem_total_gib||'?')+' GiB'}</div></div>`;
  }
  return html+'</div>';
}

function processSection(pr){
  if(!pr||!pr.available) return '';
  let html='<h2>Processes</h2><table><tr><th>rank</th><th>host</th><th>RSS GiB</th><th>cpu %</th><th>GPU alloc</th><th>GPU reserved</th><th>overhang</th><th>self µs/step</th></tr>';
  for(const r of pr.rows){
    html+=`<tr><td>r${r.rank}</td><td>${esc(r.hostname||'—')}</td>
      <td class="b-${r.rss_band||'ok'}">${r.rss_gib||'—'}</td>
      <td>${fm(r.cpu_percent,0)}</td>
      <td class="b-${r.gpu_band||'ok'}">${r.gpu_alloc_gib||'—'}</td>
      <td>${r.gpu_reserved_gib||'—'}</td>
      <td class="${r.overhang_flag?'b-warn':''}">${r.overhang_ratio==null?'—':r.overhang_ratio.toFixed(1)+'x'}</td>
      <td>${fm(r.self_overhead_us,0)}</td></tr>`;
  }
  return html+'</table>';
}

function commSection(cm){
  if(!cm||!cm.available) return '';
  let html='<h2>RCCL rank stats (xGMI all-gather)</h2>';
  html+=`<div class="dim">gather latency ${fm(cm.gather_latency_ms,2)} ms (mean ${fm(cm.gather_latency_ms_mean,2)} ms) · world ${cm.world_size}`;
  if(cm.step_skew) html+=` · step spread ${fm(cm.step_skew.spread_ms)} ms`;
  if(cm.slowest_rank!=null) html+=` · slowest r${cm.slowest_rank}`;
  html+='</div>';
  html+='<table><tr><th>rank</th><th>step</th><th>input ms</th><th>fwd ms</th><th>bwd ms</th><th>opt ms</th><th>step ms</th><th>ddp comm ms</th><th>peak GiB</th></tr>';
  for(const r of cm.rows){
    html+=`<tr><td>r${r.rank}</td><td>${r.step==null?'—':r.step}</td>
      <td>${fm(r.input_ms)}</td><td>${fm(r.forward_ms)}</td><td>${fm(r.backward_ms)}</td>
      <td>${fm(r.optimizer_ms)}</td><td>${fm(r.step_ms)}</td><td>${fm(r.ddp_comm_ms)}</td>
      <td>${r.peak_alloc_bytes==null?'—':(r.peak_alloc_bytes/2**30).toFixed(1)}</td></tr>`;
  }
  return html+'</table>';
}

function findings(d){
  if(!d.issues||!d.issues.length) return '';
  const neutral=new Set(['NORMAL','BALANCED','NO_DATA','NO_GPU','WARMUP']);
  let html='';
  for(const i of d.issues.slice(0,8)){
    if(neutral.has(i.kind)) continue;
    html+=`<div class="verdict ${i.severity}" style="font-weight:400;padding:.4rem .8rem;margin:.3rem 0">`+
      `<b>${esc(i.status)}</b> <span class="dim">[${i.section}]</span> ${esc(i.summary)}</div>`;
  }
  return html?'<h2>Findings</h2>'+html:'';
}

function consoleSection(d){
  if(!d.stdout||!d.stdout.length) return '';
  let html='<h2>Console (rank 0)</h2><div style="background:#000;border-radius:6px;padding:.6rem;font-family:monospace;font-size:.75rem">';
  for(const l of d.stdout)
    html+=`<div style="color:${l.stream==='stderr'?'#e08':'#9d9'}">${esc(l.line)}</div>`;
  return html+'</div>';
}

async function tick(){
  try{
    const r=await fetch('/api/live');const d=await r.json();
    const s=d.sections||{};
    let html='';
    html+=stepTimeSection(s.step_time);
    // model-health banner: surface a memory-domain finding that outranks
    // the step-time verdict (combined model-diagnostics card)
    if(s.model&&s.model.diagnosis&&s.model.diagnosis.evidence&&
       s.model.diagnosis.evidence.domain==='step_memory'&&
       (s.model.diagnosis.severity==='warn'||s.model.diagnosis.severity==='crit')){
      const m=s.model.diagnosis;
      html+=`<div class="verdict ${m.severity}" style="font-weight:500">${esc(m.status)} <span class="dim">[model health]</span> ${esc(m.summary)}</div>`;
    }
    html+=memorySection(s.step_memory);
    html+=commSection(s.comm);
    html+=findings(d);
    html+=systemSection(s.system);
    html+=processSection(s.process);
    html+=consoleSection(d);
    if(d.freshness&&d.freshness!=='live')
      html+=`<div class="dim">freshness: ${d.freshness}</div>`;
    document.getElementById('content').innerHTML=html||'<div class="dim">no telemetry yet…</div>';
  }catch(e){document.getElementById('content').innerHTML='<div class="dim">aggregator not ready…</div>';}
}
setInterval(tick,1500);tick();
</script></body></html>"""


def _live_payload(db_path: str, session=None) -> dict:
    from traceml_amd.renderers import live_view

    return live_view(db_path, session=session)


def build_app(get_db_path, get_session=lambda: None):
    """FastAPI app factory (separate from the uvicorn server so the route
    contract is testable with a plain TestClient)."""
    from fastapi import FastAPI
    from fastapi.responses import HTMLResponse, JSONResponse

    app = FastAPI()

    @app.get("/")
    def index():
        return HTMLResponse(_PAGE)

    # FastAPI runs sync endpoints in a threadpool: two concurrent /api/live
    # polls would otherwise race on the shared LiveStepTimeSession cursor
    poll_lock = threading.Lock()

    @app.get("/api/live")
    def live():
        db_path = get_db_path()
        if db_path is None:
            return JSONResponse({}, status_code=503)
        try:
            with poll_lock:
                return JSONResponse(_live_payload(db_path, get_session()))
        except Exception as exc:
            return JSONResponse({"error": repr(exc)}, status_code=500)

    return app


class DashboardDisplayDriver(DisplayDriver):
    def __init__(self, port: int = 8765) -> None:
        self.port = port
        self._server = None
        self._thread: Optional[threading.Thread] = None
        self._db_path: Optional[str] = None
        self._session = None

    def _get_session(self):
        if self._session is None and self._db_path is not None:
            from traceml_amd.steptime.pipeline import LiveStepTimeSession

            self._session = LiveStepTimeSession(self._db_path)
        return self._session

    def start(self) -> None:
        try:
            import uvicorn

            app = build_app(lambda: self._db_path, self._get_session)
        except Exception:
            logger.warning(
                "traceml_amd: fastapi/uvicorn unavailable, dashboard disabled"
            )
            return

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
