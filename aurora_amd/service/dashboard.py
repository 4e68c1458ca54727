"""Single-file operations dashboard served at /ui.

The reference ships a Next.js client that talks to the
same REST/SSE surface; per the survey the API contract is what must be
preserved, and this page is the dependency-free equivalent for operating
the MI355X stack: live incident fleet (SSE-fed), per-incident drill-down
(thoughts / findings / citations / suggestions / timeline / waterfall),
SRE metrics header, and engine counters.  Vanilla JS + fetch streaming
(EventSource cannot send the X-User-Id/X-Org-Id auth headers).
"""

DASHBOARD_HTML = """<!doctype html>
<html><head><meta charset="utf-8"><title>aurora-mi355x</title><style>
body{font:14px/1.45 -apple-system,Segoe UI,sans-serif;margin:0;background:#0d1117;color:#c9d1d9}
header{display:flex;gap:24px;align-items:baseline;padding:12px 20px;background:#161b22;border-bottom:1px solid #30363d}
h1{font-size:16px;margin:0;color:#58a6ff}
.kpi{color:#8b949e}.kpi b{color:#c9d1d9}
main{display:grid;grid-template-columns:minmax(380px,1fr) 2fr;gap:0;height:calc(100vh - 49px)}
section{overflow:auto;padding:14px 18px}
#list{border-right:1px solid #30363d}
table{width:100%;border-collapse:collapse}
td,th{padding:5px 8px;text-align:left;border-bottom:1px solid #21262d;font-size:13px}
tr:hover{background:#161b22;cursor:pointer}
.sev-critical{color:#f85149}.sev-high{color:#ffa657}.sev-medium{color:#d29922}.sev-low{color:#8b949e}
.st-running{color:#58a6ff}.st-completed{color:#3fb950}.st-failed{color:#f85149}.st-pending{color:#8b949e}
pre{white-space:pre-wrap;background:#161b22;border:1px solid #30363d;border-radius:6px;padding:10px;font-size:12px}
h3{color:#58a6ff;border-bottom:1px solid #30363d;padding-bottom:4px}
input{background:#0d1117;color:#c9d1d9;border:1px solid #30363d;border-radius:4px;padding:3px 6px}
.pill{display:inline-block;background:#21262d;border-radius:10px;padding:1px 8px;margin:1px;font-size:12px}
</style></head><body>
<header><h1>aurora-mi355x</h1>
<span class="kpi">user <input id="u" size="8" value="admin"> org <input id="o" size="8" value="default"></span>
<span class="kpi" id="kpis">loading…</span>
<span class="kpi" id="eng"></span></header>
<main><section id="list"><table id="tbl"><thead><tr>
<th>incident</th><th>sev</th><th>status</th><th>source</th><th>title</th></tr></thead>
<tbody></tbody></table></section>
<section id="detail"><p class="kpi">Select an incident.</p></section></main>
<script>
const $=q=>document.querySelector(q);
const H=()=>({'X-User-Id':$('#u').value,'X-Org-Id':$('#o').value});
const api=async p=>{const r=await fetch(p,{headers:H()});if(!r.ok)throw new Error(p+': '+r.status);return r.json()};
const esc=s=>String(s??'').replace(/[&<>]/g,c=>({'&':'&amp;','<':'&lt;','>':'&gt;'}[c]));
async function kpis(){try{const m=await api('/metrics');
$('#kpis').innerHTML=Object.entries(m).filter(([k,v])=>typeof v!=='object')
.slice(0,8).map(([k,v])=>`${esc(k)} <b>${esc(typeof v==='number'?v.toFixed?v.toFixed(2):v:v)}</b>`).join(' · ');}catch(e){}}
async function list(){try{const xs=await api('/incidents');const tb=$('#tbl tbody');tb.innerHTML='';
for(const i of xs){const tr=document.createElement('tr');
tr.innerHTML=`<td>${esc((i.id||'').slice(0,10))}</td><td class="sev-${esc(i.severity)}">${esc(i.severity)}</td>`+
`<td class="st-${esc(i.aurora_status||i.status)}">${esc(i.aurora_status||i.status)}</td>`+
`<td>${esc(i.source)}</td><td>${esc((i.title||'').slice(0,60))}</td>`;
tr.onclick=()=>detail(i.id);tb.appendChild(tr);}}catch(e){}}
async function detail(id){const d=$('#detail');d.innerHTML='<p class="kpi">loading…</p>';
const out=[];const get=async(p,t)=>{try{const x=await api(p);if(x&&(x.length===undefined||x.length))
out.push(`<h3>${t}</h3><pre>${esc(JSON.stringify(x,null,1).slice(0,6000))}</pre>`);}catch(e){}};
try{const inc=await api('/incidents/'+id);
out.push(`<h3>${esc(inc.title||id)}</h3><pre>${esc(JSON.stringify(inc,null,1).slice(0,4000))}</pre>`);}catch(e){}
await get('/incidents/'+id+'/findings','Sub-agent findings');
await get('/incidents/'+id+'/thoughts','Thoughts');
await get('/incidents/'+id+'/citations','Citations');
await get('/incidents/'+id+'/suggestions','Suggestions');
await get('/incidents/'+id+'/timeline','Timeline');
await get('/incidents/'+id+'/hypotheses','Hypothesis ledger');
await get('/monitor/waterfall/'+id,'Tool waterfall');
try{const pm=await api('/postmortems/'+id);
if(pm&&pm.content)out.push(`<h3>Postmortem</h3><pre>${esc(pm.content.slice(0,6000))}</pre>`);}catch(e){}
d.innerHTML=out.join('')||'<p class="kpi">no data</p>';}
async function engine(){try{const st=await api('/engine/status');
$('#eng').innerHTML=`engine <b>${st.healthy?'healthy':'STUCK'}</b> · ${esc(st.model)} on ${esc(st.device)}`+
` · kv ${st.kv_pages_total-st.kv_pages_free}/${st.kv_pages_total} pages · ${st.running} running`+
` · ${(st.counters.generated_tokens||0).toLocaleString()} tokens`;}catch(e){$('#eng').textContent='';}}
async function sse(){try{const r=await fetch('/incidents/stream',{headers:H()});
const rd=r.body.getReader();const dec=new TextDecoder();
for(;;){const {done,value}=await rd.read();if(done)break;
if(dec.decode(value).includes('data:')){list();kpis();}}}catch(e){setTimeout(sse,3000);}}
list();kpis();engine();sse();setInterval(()=>{list();kpis();engine();},10000);
</script></body></html>"""
