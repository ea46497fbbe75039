"""Minimal web console — the L7 analog of the reference's React app
(console/src: overview/runtime/configuration/components/events/rollouts).
A single static page (no build toolchain in this environment) served by the
hub and the node API, talking to the same endpoints with token auth and SSE
with Last-Event-ID resume.
"""

CONSOLE_HTML = """<!doctype html>
<html><head><meta charset="utf-8"><title>arkflow_amd console</title>
<style>
 body{font-family:system-ui,sans-serif;margin:0;background:#0c0f14;color:#dde}
 header{padding:10px 16px;background:#151a23;display:flex;gap:16px;
        align-items:center}
 header h1{font-size:16px;margin:0;color:#7fd}
 nav button{background:none;border:0;color:#9ab;font-size:14px;cursor:pointer;
            padding:6px 10px}
 nav button.act{color:#7fd;border-bottom:2px solid #7fd}
 input{background:#1a2030;border:1px solid #345;color:#dde;padding:4px 8px;
       border-radius:4px}
 main{padding:16px}
 table{border-collapse:collapse;width:100%;font-size:13px}
 th,td{border-bottom:1px solid #2a3345;padding:6px 8px;text-align:left}
 .pill{padding:2px 8px;border-radius:10px;font-size:12px}
 .running{background:#0a4;color:#fff}.stopped{background:#555;color:#fff}
 .failed{background:#a22;color:#fff}
 button.op{background:#26324a;border:1px solid #456;color:#cde;
           border-radius:4px;cursor:pointer;margin-right:4px}
 pre{background:#10141c;padding:10px;border-radius:6px;overflow:auto;
     font-size:12px}
 #log{max-height:50vh;overflow:auto}
</style></head><body>
<header><h1>arkflow_amd</h1>
<nav id="nav"></nav>
<span style="flex:1"></span>
<input id="token" placeholder="API token" size="18">
</header>
<main id="main">loading…</main>
<script>
const tabs = ["overview","streams","components","events","nodes","rollouts"];
let tab = "overview";
const $ = (s)=>document.querySelector(s);
const hdrs = ()=>{const t=$("#token").value;
  return t?{"Authorization":"Bearer "+t}:{};};
async function get(p){const r=await fetch(p,{headers:hdrs()});
  if(!r.ok)throw new Error(r.status);return r.json();}
async function post(p){const r=await fetch(p,{method:"POST",headers:hdrs()});
  return r.json();}
function nav(){$("#nav").innerHTML=tabs.map(t=>
  `<button class="${t===tab?'act':''}" onclick="setTab('${t}')">${t}</button>`
 ).join("");}
window.setTab=(t)=>{tab=t;nav();render();};
function pill(s){return `<span class="pill ${s}">${s}</span>`;}
async function render(){
 const m=$("#main");
 try{
 if(tab==="overview"){
   const s=await get("/api/v1/system/status");
   m.innerHTML=`<pre>${JSON.stringify(s,null,2)}</pre>`;
 }else if(tab==="streams"){
   const rows=await get("/api/v1/streams");
   m.innerHTML=`<table><tr><th>id</th><th>state</th><th>conv</th>
     <th>in msgs</th><th>out msgs</th><th>errors</th><th>ops</th></tr>`+
    rows.map(r=>`<tr><td>${r.id}</td><td>${pill(r.state)}</td>
     <td>${r.convergence}</td><td>${r.metrics.input_messages}</td>
     <td>${r.metrics.output_messages}</td>
     <td>${r.metrics.processing_errors+r.metrics.output_errors}</td>
     <td><button class="op" onclick="op('${r.id}','start')">start</button>
         <button class="op" onclick="op('${r.id}','stop')">stop</button>
         <button class="op" onclick="op('${r.id}','restart')">restart</button>
     </td></tr>`).join("")+"</table>";
 }else if(tab==="components"){
   const cs=await get("/api/v1/components");
   m.innerHTML=`<table><tr><th>kind</th><th>name</th><th>description</th>
    </tr>`+cs.map(c=>`<tr><td>${c.kind}</td><td>${c.name}</td>
    <td>${c.description||""}</td></tr>`).join("")+"</table>";
 }else if(tab==="events"){
   m.innerHTML=`<div id="log"></div>`;startSse();
 }else if(tab==="nodes"){
   const ns=await get("/nodes");
   m.innerHTML=`<table><tr><th>node</th><th>online</th><th>lease</th>
    </tr>`+ns.map(n=>`<tr><td>${n.node_id}</td><td>${n.online}</td>
    <td>${new Date(n.lease_expires*1000).toLocaleTimeString()}</td></tr>`
    ).join("")+"</table>";
 }else if(tab==="rollouts"){
   const rs=await get("/rollouts");
   m.innerHTML=`<pre>${JSON.stringify(rs,null,2)}</pre>`;
 }}catch(e){m.innerHTML=`<pre>error: ${e}</pre>`;}
}
window.op=async(id,o)=>{await post(`/api/v1/streams/${id}/${o}`);render();};
let es=null,lastId=0;
function startSse(){
 if(es)es.close();
 // EventSource can't set headers; token via query for the console only
 es=new EventSource(`/api/v1/events/stream`);
 es.onmessage=(ev)=>{lastId=ev.lastEventId||lastId;
  const d=document.createElement("div");d.textContent=ev.data;
  $("#log")?.prepend(d);};
}
nav();render();setInterval(()=>{if(tab!=="events")render();},3000);
</script></body></html>
"""


def mount_console(app) -> None:
    """Serve the console at / on a FastAPI app (node or hub)."""
    from fastapi.responses import HTMLResponse

    @app.get("/", response_class=HTMLResponse, include_in_schema=False)
    async def console():
        return CONSOLE_HTML
