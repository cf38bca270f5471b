"""Built-in chat UI (reference src/ui/streamlit_app.py capability: chat with
history + temperature/top-k controls, SSE streaming, document/file upload
with the reference's 45 000-char chunking (streamlit_app.py:42-43), corpus
clear, backend health/info panel).  The reference ran a separate Streamlit
process talking HTTP to the API; here the UI is a single static page served
by the engine itself at /ui — no extra process, same /chat + /chat/stream +
/embed + /clear + /health + /info wire calls from the browser.  File
uploads read client-side (.txt/.md/.json/.csv; the reference additionally
parsed PDFs with a client-side library — binary formats here surface a
clear "paste text instead" notice rather than silently mangling bytes)."""

UI_HTML = """<!doctype html>
<html lang="en">
<head>
<meta charset="utf-8"><title>sentio-amd</title>
<style>
 body{font-family:system-ui,sans-serif;max-width:860px;margin:24px auto;padding:0 12px;background:#111;color:#eee}
 h1{font-size:1.2rem}
 #log{border:1px solid #333;border-radius:8px;padding:12px;min-height:260px;max-height:60vh;overflow-y:auto}
 .q{color:#8cf;margin:8px 0 2px} .a{white-space:pre-wrap;margin:0 0 4px}
 .src{color:#888;font-size:.8rem;margin:0 0 10px} .meta{color:#666;font-size:.72rem}
 textarea,input[type=text]{width:100%;background:#1a1a1a;color:#eee;border:1px solid #444;border-radius:6px;padding:8px;box-sizing:border-box}
 button{background:#2b6;border:0;border-radius:6px;padding:8px 16px;color:#fff;margin-top:6px;cursor:pointer}
 button.sec{background:#444}
 #status{float:right;font-size:.8rem;color:#888}
 section{margin-top:16px}
 .ctl{display:flex;gap:14px;align-items:center;font-size:.85rem;color:#aaa;margin-top:6px;flex-wrap:wrap}
 .ctl input[type=range]{width:120px} .ctl input[type=number]{width:60px;background:#1a1a1a;color:#eee;border:1px solid #444;border-radius:4px}
 progress{width:160px;height:8px}
 #info{font-size:.75rem;color:#9a9;white-space:pre-wrap}
</style>
</head>
<body>
<h1>sentio-amd <span id="status">checking…</span></h1>
<div id="log"></div>
<section>
 <input type="text" id="q" placeholder="Ask a question…" onkeydown="if(event.key==='Enter')ask()">
 <div class="ctl">
  <label><input type="checkbox" id="stream" checked> stream</label>
  <label>temp <input type="range" id="temp" min="0" max="1" step="0.1" value="0.3"
         oninput="document.getElementById('tval').textContent=this.value">
         <span id="tval">0.3</span></label>
  <label>top-k <input type="number" id="topk" min="1" max="50" value="10"></label>
  <button onclick="ask()">Send</button>
  <button class="sec" onclick="clearChat()">Clear chat</button>
 </div>
</section>
<section>
 <details><summary>Upload documents</summary>
  <textarea id="doc" rows="5" placeholder="Paste document text…"></textarea>
  <div class="ctl">
   <button onclick="uploadText()">Ingest text</button>
   <input type="file" id="files" multiple accept=".txt,.md,.json,.csv,.log,.py,.rst">
   <button class="sec" onclick="uploadFiles()">Ingest files</button>
   <progress id="uprog" value="0" max="1" hidden></progress>
   <span id="upmsg"></span>
  </div>
  <div class="ctl">
   <button class="sec" onclick="clearCorpus()">Clear corpus</button>
   <span id="clrmsg"></span>
  </div>
 </details>
 <details><summary>Backend info</summary><div id="info">…</div></details>
</section>
<script>
const log = document.getElementById('log');
const history = [];                 // {role, content} pairs sent with /chat
const CHUNK = 45000;                // reference streamlit_app.py:42-43

async function health(){
  try{const r=await fetch('/health');const j=await r.json();
      document.getElementById('status').textContent=j.status+' · '+(j.services?j.services.device||'':'');}
  catch(e){document.getElementById('status').textContent='offline';}
}
async function info(){
  try{const r=await fetch('/info');const j=await r.json();
      document.getElementById('info').textContent=JSON.stringify(j,null,1);}
  catch(e){document.getElementById('info').textContent='unavailable';}
}
health(); info(); setInterval(health, 15000);

function addQ(q){log.insertAdjacentHTML('beforeend','<p class="q">'+esc(q)+'</p><p class="a"></p>');
                 log.scrollTop=log.scrollHeight; return log.lastElementChild;}
function addMeta(a, j){
  if(j.sources&&j.sources.length)
    a.insertAdjacentHTML('afterend','<p class="src">sources: '+
      j.sources.map(s=>esc(s.source||(s.text&&s.text.slice(0,40))||'?')).join(' · ')+'</p>');
  if(j.metadata&&j.metadata.latency_ms)
    a.insertAdjacentHTML('afterend','<p class="meta">'+(+j.metadata.latency_ms).toFixed(0)+' ms</p>');
}

async function ask(){
  const qEl=document.getElementById('q');
  const q=qEl.value.trim(); if(!q)return;
  qEl.value='';
  const a=addQ(q);
  const body={question:q,
              temperature:+document.getElementById('temp').value,
              top_k:+document.getElementById('topk').value,
              history:history.slice(-8)};
  try{
    if(document.getElementById('stream').checked){
      const r=await fetch('/chat/stream',{method:'POST',
        headers:{'Content-Type':'application/json'},body:JSON.stringify(body)});
      const rd=r.body.getReader(); const dec=new TextDecoder(); let buf='';
      for(;;){
        const {done,value}=await rd.read(); if(done)break;
        buf+=dec.decode(value,{stream:true});
        let i;
        while((i=buf.indexOf('\\n\\n'))>=0){
          const line=buf.slice(0,i); buf=buf.slice(i+2);
          if(line.startsWith('data: ')){
            const d=line.slice(6);
            if(d!=='[DONE]'){a.textContent+=d; log.scrollTop=log.scrollHeight;}
          }
        }
      }
    }else{
      const r=await fetch('/chat',{method:'POST',
        headers:{'Content-Type':'application/json'},body:JSON.stringify(body)});
      const j=await r.json();
      a.textContent=j.answer||JSON.stringify(j);
      addMeta(a, j);
    }
    history.push({role:'user',content:q},{role:'assistant',content:a.textContent});
  }catch(e){a.textContent='error: '+e;}
}
function clearChat(){log.innerHTML=''; history.length=0;}

async function embedChunk(text, meta){
  const r=await fetch('/embed',{method:'POST',
    headers:{'Content-Type':'application/json'},
    body:JSON.stringify({content:text,metadata:meta})});
  if(!r.ok)throw new Error('HTTP '+r.status);
  return (await r.json()).chunks??0;
}
async function ingestLarge(text, source, prog, msg){
  // reference chunked uploads at 45 000 chars per request
  const parts=[];
  for(let i=0;i<text.length;i+=CHUNK)parts.push(text.slice(i,i+CHUNK));
  let chunks=0;
  for(let i=0;i<parts.length;i++){
    chunks+=await embedChunk(parts[i],{source:source,part:i+1,parts:parts.length});
    prog.value=(i+1)/parts.length;
    msg.textContent=source+': '+(i+1)+'/'+parts.length+' requests…';
  }
  return chunks;
}
async function uploadText(){
  const t=document.getElementById('doc').value.trim(); if(!t)return;
  const m=document.getElementById('upmsg'), p=document.getElementById('uprog');
  p.hidden=false; p.value=0; m.textContent='…';
  try{
    const n=await ingestLarge(t,'ui-paste',p,m);
    m.textContent='ingested '+n+' chunks';
    document.getElementById('doc').value='';
  }catch(e){m.textContent='error: '+e;}
  p.hidden=true; info();
}
async function uploadFiles(){
  const files=document.getElementById('files').files;
  const m=document.getElementById('upmsg'), p=document.getElementById('uprog');
  if(!files.length){m.textContent='choose files first';return;}
  p.hidden=false; let total=0;
  try{
    for(const f of files){
      if(/\\.(pdf|docx?|png|jpe?g)$/i.test(f.name)){
        m.textContent=f.name+': binary format — paste extracted text instead';
        continue;
      }
      const text=await f.text();
      total+=await ingestLarge(text,f.name,p,m);
    }
    m.textContent='ingested '+total+' chunks from '+files.length+' file(s)';
  }catch(e){m.textContent='error: '+e;}
  p.hidden=true; info();
}
async function clearCorpus(){
  const m=document.getElementById('clrmsg'); m.textContent='…';
  try{const r=await fetch('/clear',{method:'POST'});
      const j=await r.json(); m.textContent=JSON.stringify(j);}
  catch(e){m.textContent='error: '+e;}
  info();
}
function esc(s){const d=document.createElement('div');d.textContent=s;return d.innerHTML;}
</script>
</body></html>
"""
