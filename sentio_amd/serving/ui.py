"""Built-in chat UI (reference src/ui/streamlit_app.py capability: chat box,
document upload, backend health probe).  The reference ran a separate
Streamlit process talking HTTP to the API; here the UI is a single static
page served by the engine itself at /ui — no extra process, same
/chat + /embed + /health wire calls from the browser."""

UI_HTML = """<!doctype html>
<html lang="en">
<head>
<meta charset="utf-8"><title>sentio-amd</title>
<style>
 body{font-family:system-ui,sans-serif;max-width:780px;margin:24px auto;padding:0 12px;background:#111;color:#eee}
 h1{font-size:1.2rem} #log{border:1px solid #333;border-radius:8px;padding:12px;min-height:240px}
 .q{color:#8cf;margin:8px 0 2px} .a{white-space:pre-wrap;margin:0 0 10px}
 .src{color:#888;font-size:.8rem} textarea,input[type=text]{width:100%;background:#1a1a1a;color:#eee;border:1px solid #444;border-radius:6px;padding:8px;box-sizing:border-box}
 button{background:#2b6;border:0;border-radius:6px;padding:8px 16px;color:#fff;margin-top:6px;cursor:pointer}
 #status{float:right;font-size:.8rem;color:#888}
 section{margin-top:18px}
</style>
</head>
<body>
<h1>sentio-amd <span id="status">checking…</span></h1>
<div id="log"></div>
<section>
 <input type="text" id="q" placeholder="Ask a question…" onkeydown="if(event.key==='Enter')ask()">
 <button onclick="ask()">Send</button>
</section>
<section>
 <details><summary>Upload a document</summary>
  <textarea id="doc" rows="5" placeholder="Paste document text…"></textarea>
  <button onclick="upload()">Ingest</button> <span id="upmsg"></span>
 </details>
</section>
<script>
const log = document.getElementById('log');
async function health(){
  try{const r=await fetch('/health');const j=await r.json();
      document.getElementById('status').textContent=j.status+' · '+(j.services?j.services.device||'':'');}
  catch(e){document.getElementById('status').textContent='offline';}
}
health(); setInterval(health, 15000);
async function ask(){
  const q=document.getElementById('q').value.trim(); if(!q)return;
  document.getElementById('q').value='';
  log.insertAdjacentHTML('beforeend','<p class="q">'+esc(q)+'</p><p class="a">…</p>');
  const a=log.lastElementChild;
  try{
    const r=await fetch('/chat',{method:'POST',headers:{'Content-Type':'application/json'},
      body:JSON.stringify({question:q})});
    const j=await r.json();
    a.textContent=j.answer||JSON.stringify(j);
    if(j.sources&&j.sources.length)
      a.insertAdjacentHTML('afterend','<p class="src">sources: '+
        j.sources.map(s=>esc(s.source||s.text&&s.text.slice(0,40)||'?')).join(' · ')+'</p>');
  }catch(e){a.textContent='error: '+e;}
}
async function upload(){
  const t=document.getElementById('doc').value.trim(); if(!t)return;
  const m=document.getElementById('upmsg'); m.textContent='…';
  try{
    const r=await fetch('/embed',{method:'POST',headers:{'Content-Type':'application/json'},
      body:JSON.stringify({text:t,metadata:{source:'ui-upload'}})});
    const j=await r.json(); m.textContent='ingested '+(j.chunks??'?')+' chunks';
    document.getElementById('doc').value='';
  }catch(e){m.textContent='error: '+e;}
}
function esc(s){const d=document.createElement('div');d.textContent=s;return d.innerHTML;}
</script>
</body></html>
"""
