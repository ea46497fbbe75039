"""Web console — the L7 analog of the reference's React app
(console/src/features: overview / runtime / configuration / components /
events / rollouts / settings).

No build toolchain exists in this environment, so instead of a
Vite/React bundle this is a build-free single-file SPA (hash routing,
fetch + EventSource) speaking to the same node-API and hub endpoints with
bearer-token auth and SSE Last-Event-ID resume. Feature parity map:

  overview      → /api/v1/system/status + stream summaries
  runtime       → stream table, per-stream metrics detail, lifecycle ops,
                  operations log (console/src/features/runtime.tsx)
  configuration → editor with validate / diff / apply, version list +
                  rollback (features/configuration.tsx)
  components    → registry browser with JSON schema + example
                  (features/components.tsx)
  events        → recent events + SSE live tail with Last-Event-ID resume
                  (features/events.tsx)
  nodes/rollouts→ hub fleet views incl. rollout create/pause/resume/
                  cancel/rollback (features/rollouts.tsx)
  settings      → endpoint + token, persisted in localStorage
                  (features/settings.tsx)
"""

CONSOLE_HTML = r"""<!doctype html>
<html><head><meta charset="utf-8"><title>arkflow_amd console</title>
<style>
 body{font-family:system-ui,sans-serif;margin:0;background:#0c0f14;color:#dde}
 header{padding:10px 16px;background:#151a23;display:flex;gap:12px;
        align-items:center;flex-wrap:wrap}
 header h1{font-size:16px;margin:0;color:#7fd}
 nav a{color:#9ab;font-size:14px;cursor:pointer;padding:6px 10px;
       text-decoration:none}
 nav a.act{color:#7fd;border-bottom:2px solid #7fd}
 input,textarea,select{background:#1a2030;border:1px solid #345;color:#dde;
       padding:4px 8px;border-radius:4px;font-family:inherit}
 textarea{width:100%;min-height:260px;font-family:ui-monospace,monospace;
          font-size:12px}
 main{padding:16px;max-width:1100px}
 table{border-collapse:collapse;width:100%;font-size:13px}
 th,td{border-bottom:1px solid #2a3345;padding:6px 8px;text-align:left}
 tr.sel{background:#182238}
 .pill{padding:2px 8px;border-radius:10px;font-size:12px}
 .running,.succeeded,.online{background:#0a4;color:#fff}
 .stopped,.created{background:#555;color:#fff}
 .failed,.offline,.cancelled{background:#a22;color:#fff}
 .paused,.pending,.dispatched{background:#a80;color:#fff}
 button.op{background:#26324a;border:1px solid #456;color:#cde;
           border-radius:4px;cursor:pointer;margin-right:4px;padding:3px 8px}
 button.op:hover{background:#31405e}
 pre{background:#10141c;padding:10px;border-radius:6px;overflow:auto;
     font-size:12px}
 .cards{display:flex;gap:12px;flex-wrap:wrap;margin-bottom:14px}
 .card{background:#151a23;border:1px solid #2a3345;border-radius:8px;
       padding:12px 18px;min-width:130px}
 .card .v{font-size:22px;color:#7fd}.card .k{font-size:12px;color:#9ab}
 #log{max-height:55vh;overflow:auto;font-size:12px}
 #log div{border-bottom:1px solid #1c2435;padding:3px 0}
 .row{display:flex;gap:16px;align-items:flex-start}
 .row>div{flex:1}
 .msg{color:#fa6}.ok{color:#6fa}
 h3{color:#9cf;font-size:14px;margin:14px 0 6px}
</style></head><body>
<header><h1>arkflow_amd</h1><nav id="nav"></nav></header>
<main id="main">loading…</main>
<script>
"use strict";
const TABS=["overview","runtime","configuration","components","events",
            "nodes","rollouts","settings"];
const $=(s)=>document.querySelector(s);
const store={get t(){return localStorage.getItem("af_token")||""},
             set t(v){localStorage.setItem("af_token",v)},
             get base(){return localStorage.getItem("af_base")||""},
             set base(v){localStorage.setItem("af_base",v)}};
const hdrs=(extra)=>{const h=extra||{};
  if(store.t)h["Authorization"]="Bearer "+store.t;return h;};
async function get(p){const r=await fetch(store.base+p,{headers:hdrs()});
  if(!r.ok)throw new Error(r.status+" "+p);return r.json();}
async function post(p,body){const r=await fetch(store.base+p,{method:"POST",
  headers:hdrs(body?{"Content-Type":"application/json"}:{}),
  body:body?JSON.stringify(body):undefined});
  let j=null;try{j=await r.json()}catch(e){}
  if(!r.ok)throw new Error((j&&(j.detail||j.error))||r.status);
  return j;}
const esc=(s)=>String(s??"").replace(/[&<>"]/g,
  c=>({"&":"&amp;","<":"&lt;",">":"&gt;",'"':"&quot;"}[c]));
const pill=(s)=>`<span class="pill ${esc(s)}">${esc(s)}</span>`;
let tab=location.hash.replace("#/","")||"overview";
let selStream=null,es=null,lastId=0;
function nav(){$("#nav").innerHTML=TABS.map(t=>
 `<a href="#/${t}" class="${t===tab?'act':''}">${t}</a>`).join("");}
window.addEventListener("hashchange",()=>{tab=location.hash.replace("#/","")
 ||"overview";if(es&&tab!=="events"){es.close();es=null;}nav();render();});

// ---- views ----------------------------------------------------------------
async function vOverview(m){
 const s=await get("/api/v1/system/status");
 let streams=[];try{streams=await get("/api/v1/streams")}catch(e){}
 const agg=(k)=>streams.reduce((a,r)=>a+(r.metrics?.[k]||0),0);
 m.innerHTML=`<div class="cards">
  <div class="card"><div class="v">${esc(s.engine_state||s.state||"—")}</div>
    <div class="k">engine</div></div>
  <div class="card"><div class="v">${streams.length}</div>
    <div class="k">streams</div></div>
  <div class="card"><div class="v">${streams.filter(r=>r.state==="running")
    .length}</div><div class="k">running</div></div>
  <div class="card"><div class="v">${agg("input_messages")}</div>
    <div class="k">msgs in</div></div>
  <div class="card"><div class="v">${agg("output_messages")}</div>
    <div class="k">msgs out</div></div>
  <div class="card"><div class="v">${agg("processing_errors")+
    agg("output_errors")}</div><div class="k">errors</div></div></div>
  <pre>${esc(JSON.stringify(s,null,2))}</pre>`;
}
async function vRuntime(m){
 const rows=await get("/api/v1/streams");
 let detail="";
 if(selStream){
  try{
   const d=await get(`/api/v1/streams/${selStream}`);
   const mx=await get(`/api/v1/streams/${selStream}/metrics`);
   let ops=[];try{ops=await get("/api/v1/operations")}catch(e){}
   detail=`<h3>${esc(selStream)}</h3><div class="row">
    <div><pre>${esc(JSON.stringify(d,null,2))}</pre></div>
    <div><pre>${esc(JSON.stringify(mx,null,2))}</pre>
     <h3>operations</h3><pre>${esc(JSON.stringify(
       ops.filter(o=>!o.stream_id||o.stream_id===selStream).slice(0,12),
       null,2))}</pre></div></div>`;
  }catch(e){detail=`<pre>error: ${esc(e.message)}</pre>`}
 }
 m.innerHTML=`<table><tr><th>id</th><th>state</th><th>conv</th>
   <th>in</th><th>out</th><th>errors</th><th>restarts</th><th>ops</th></tr>`+
  rows.map(r=>`<tr class="${r.id===selStream?'sel':''}">
   <td><a href="#" onclick="selS('${esc(r.id)}');return false">${esc(r.id)}
   </a></td><td>${pill(r.state)}</td><td>${esc(r.convergence)}</td>
   <td>${r.metrics.input_messages}</td><td>${r.metrics.output_messages}</td>
   <td>${r.metrics.processing_errors+r.metrics.output_errors}</td>
   <td>${r.metrics.restarts||0}</td>
   <td><button class="op" onclick="op('${esc(r.id)}','start')">start</button>
       <button class="op" onclick="op('${esc(r.id)}','stop')">stop</button>
       <button class="op" onclick="op('${esc(r.id)}','restart')">restart
       </button></td></tr>`).join("")+"</table>"+detail;
}
async function vConfig(m){
 let versions=[];try{versions=await get("/api/v1/configuration/versions")}
 catch(e){}
 m.innerHTML=`<div class="row"><div>
   <h3>editor</h3>
   <textarea id="cfg" spellcheck="false"></textarea><br>
   <button class="op" onclick="cfgLoad()">load current</button>
   <button class="op" onclick="cfgDo('validate')">validate</button>
   <button class="op" onclick="cfgDo('diff')">diff</button>
   <button class="op" onclick="cfgDo('apply')">apply</button>
   <span id="cfgmsg"></span><pre id="cfgout"></pre></div>
  <div><h3>versions</h3><table><tr><th>version</th><th>time</th><th></th>
   </tr>${versions.map(v=>`<tr><td>${esc(v.version)}</td>
   <td>${esc(v.created_at||v.ts||"")}</td>
   <td><button class="op" onclick="cfgRollback('${esc(v.version)}')">
   rollback</button></td></tr>`).join("")}</table></div></div>`;
 cfgLoad();
}
window.cfgLoad=async()=>{try{
 const c=await get("/api/v1/configuration");
 $("#cfg").value=JSON.stringify(c,null,2);}catch(e){
 $("#cfgmsg").innerHTML=`<span class="msg">${esc(e.message)}</span>`}};
window.cfgDo=async(what)=>{
 let body;try{body=JSON.parse($("#cfg").value)}catch(e){
  $("#cfgmsg").innerHTML=`<span class="msg">bad JSON: ${esc(e.message)}
  </span>`;return;}
 try{const r=await post(`/api/v1/configuration/${what}`,body);
  $("#cfgout").textContent=JSON.stringify(r,null,2);
  $("#cfgmsg").innerHTML=`<span class="ok">${what} ok</span>`;
 }catch(e){$("#cfgmsg").innerHTML=`<span class="msg">${esc(e.message)}
  </span>`}};
window.cfgRollback=async(v)=>{try{
 const r=await post(`/api/v1/configuration/rollback/${v}`);
 $("#cfgout").textContent=JSON.stringify(r,null,2);render();}catch(e){
 $("#cfgmsg").innerHTML=`<span class="msg">${esc(e.message)}</span>`}};
async function vComponents(m){
 const cs=await get("/api/v1/components");
 const kinds=[...new Set(cs.map(c=>c.kind))];
 const kind=window._ckind||kinds[0];
 const sel=window._cname;
 let schema="";
 if(sel){try{const d=await get(`/api/v1/components/${kind}/${sel}`);
  schema=`<h3>${esc(kind)} / ${esc(sel)}</h3>
   <pre>${esc(JSON.stringify(d,null,2))}</pre>`}catch(e){}}
 m.innerHTML=`<select onchange="window._ckind=this.value;
   window._cname=null;render()">${kinds.map(k=>
   `<option ${k===kind?"selected":""}>${esc(k)}</option>`).join("")}</select>
  <div class="row"><div><table><tr><th>name</th><th>description</th></tr>
  ${cs.filter(c=>c.kind===kind).map(c=>`<tr>
   <td><a href="#" onclick="window._cname='${esc(c.name)}';render();
    return false">${esc(c.name)}</a></td>
   <td>${esc(c.description||"")}</td></tr>`).join("")}
  </table></div><div>${schema}</div></div>`;
}
async function vEvents(m){
 let recent=[];try{recent=await get("/api/v1/events?limit=50")}catch(e){}
 m.innerHTML=`<h3>live (SSE, resumes from id ${lastId})</h3><div id="log">
  ${recent.slice().reverse().map(ev=>`<div>${esc(JSON.stringify(ev))}
  </div>`).join("")}</div>`;
 startSse();
}
async function vNodes(m){
 const ns=await get("/nodes");
 m.innerHTML=`<table><tr><th>node</th><th>status</th><th>lease</th>
  <th>report</th></tr>`+ns.map(n=>{
  const st=n.online?"online":"offline";
  const rep=n.last_report?JSON.stringify(n.last_report.status||{}):"—";
  return `<tr><td>${esc(n.node_id)}</td><td>${pill(st)}</td>
   <td>${new Date(n.lease_expires*1000).toLocaleTimeString()}</td>
   <td><pre style="margin:0">${esc(rep)}</pre></td></tr>`}).join("")+
  "</table><h3>intents</h3><pre id='ints'>…</pre>";
 try{$("#ints").textContent=JSON.stringify(
   await get("/intents"),null,2)}catch(e){$("#ints").textContent=e.message}
}
async function vRollouts(m){
 let rs=[];try{rs=await get("/rollouts")}catch(e){
  m.innerHTML=`<pre>hub-only view (${esc(e.message)})</pre>`;return;}
 m.innerHTML=`<h3>create</h3>
  <input id="ro_nodes" placeholder="nodes (comma-sep)" size="30">
  <textarea id="ro_cfg" placeholder='config JSON'
   style="min-height:80px"></textarea>
  <button class="op" onclick="roCreate()">create rollout</button>
  <span id="romsg"></span>
  <h3>rollouts</h3><table><tr><th>id</th><th>state</th><th>progress</th>
  <th>ops</th></tr>`+rs.map(r=>{
  const nodes=JSON.parse(r.nodes||"[]");
  return `<tr><td>${esc(r.rollout_id)}</td><td>${pill(r.state)}</td>
  <td>${r.position}/${nodes.length}</td>
  <td>${["step","pause","resume","cancel","rollback"].map(a=>
   `<button class="op" onclick="roAct('${esc(r.rollout_id)}','${a}')">
   ${a}</button>`).join("")}</td></tr>`}).join("")+"</table>";
}
window.roCreate=async()=>{try{
 let cfg={};const t=$("#ro_cfg").value.trim();if(t)cfg=JSON.parse(t);
 await post("/rollouts",{config:cfg,
  nodes:$("#ro_nodes").value.split(",").map(s=>s.trim()).filter(Boolean)});
 render();}catch(e){$("#romsg").innerHTML=
  `<span class="msg">${esc(e.message)}</span>`}};
window.roAct=async(id,a)=>{try{
 await post(a==="step"?`/rollouts/${id}/step`:`/rollouts/${id}/${a}`);
 render();}catch(e){alert(e.message)}};
async function vSettings(m){
 m.innerHTML=`<h3>settings</h3>
  <p>API base <input id="s_base" value="${esc(store.base)}" size="30"
   placeholder="(same origin)"></p>
  <p>Bearer token <input id="s_tok" value="${esc(store.t)}" size="30"></p>
  <button class="op" onclick="store.base=$('#s_base').value;
   store.t=$('#s_tok').value;render()">save</button>`;
}
const VIEWS={overview:vOverview,runtime:vRuntime,configuration:vConfig,
 components:vComponents,events:vEvents,nodes:vNodes,rollouts:vRollouts,
 settings:vSettings};
async function render(){const m=$("#main");
 try{await (VIEWS[tab]||vOverview)(m);}catch(e){
  m.innerHTML=`<pre>error: ${esc(e.message)}</pre>`;}}
window.selS=(id)=>{selStream=id===selStream?null:id;render();};
window.op=async(id,o)=>{try{await post(`/api/v1/streams/${id}/${o}`)}
 catch(e){alert(e.message)}render();};
function startSse(){
 if(es)es.close();
 es=new EventSource(store.base+`/api/v1/events/stream`);
 es.onmessage=(ev)=>{lastId=ev.lastEventId||lastId;
  const d=document.createElement("div");d.textContent=ev.data;
  $("#log")?.prepend(d);};
}
nav();render();
setInterval(()=>{if(tab!=="events"&&tab!=="configuration"&&
 tab!=="settings")render();},4000);
</script></body></html>
"""


def mount_console(app) -> None:
    """Serve the console at / on a FastAPI app (node or hub)."""
    from fastapi.responses import HTMLResponse

    @app.get("/", response_class=HTMLResponse, include_in_schema=False)
    async def console():
        return CONSOLE_HTML
