"""Minimal built-in web UI (the reference ships a 237k-LoC React app;
this is a functional single-page chat + admin view served at / so the
stack is usable from a browser without a separate frontend build)."""

INDEX_HTML = """<!doctype html>
<html><head><meta charset="utf-8"><title>helix_amd</title>
<style>
 body{font-family:system-ui,sans-serif;margin:0;display:flex;height:100vh}
 #side{width:260px;background:#111;color:#eee;padding:12px;overflow-y:auto}
 #side h1{font-size:16px} #side div.s{padding:6px;cursor:pointer;border-radius:6px;font-size:13px}
 #side div.s:hover{background:#333}
 #main{flex:1;display:flex;flex-direction:column}
 #log{flex:1;overflow-y:auto;padding:16px;background:#f7f7f8}
 .msg{max-width:760px;margin:8px auto;padding:10px 14px;border-radius:10px;white-space:pre-wrap}
 .user{background:#d8e8ff} .assistant{background:#fff;border:1px solid #ddd}
 #bar{display:flex;padding:12px;gap:8px;border-top:1px solid #ddd}
 #inp{flex:1;padding:10px;font-size:15px}
 input,select,button{font-size:14px}
 #cfg{padding:8px 12px;display:flex;gap:8px;background:#eee;align-items:center}
</style></head><body>
<div id="side"><h1>helix_amd</h1>
 <button onclick="newSession()">+ new session</button>
 <div id="sessions"></div>
</div>
<div id="main">
 <div id="cfg">
  key <input id="key" size="14" value="admin-key">
  model <select id="model"></select>
  <span id="status"></span>
 </div>
 <div id="log"></div>
 <div id="bar">
  <input id="inp" placeholder="Say something..." onkeydown="if(event.key==='Enter')send()">
  <button onclick="send()">send</button>
 </div>
</div>
<script>
let sessionId = null;
const H = () => ({'Authorization':'Bearer '+document.getElementById('key').value,
                  'Content-Type':'application/json'});
async function loadModels(){
  const r = await fetch('/v1/models',{headers:H()});
  if(!r.ok) return;
  const sel = document.getElementById('model'); sel.innerHTML='';
  for(const m of (await r.json()).data){
    const o=document.createElement('option');o.value=o.textContent=m.id;sel.appendChild(o);}
}
async function loadSessions(){
  const r = await fetch('/api/v1/sessions',{headers:H()});
  if(!r.ok) return;
  const el = document.getElementById('sessions'); el.innerHTML='';
  for(const s of await r.json()){
    const d=document.createElement('div');d.className='s';d.textContent=s.name;
    d.onclick=()=>openSession(s.id);el.appendChild(d);}
}
function add(role, text){
  const d=document.createElement('div');d.className='msg '+role;d.textContent=text;
  document.getElementById('log').appendChild(d);
  d.scrollIntoView();return d;}
function newSession(){sessionId=null;document.getElementById('log').innerHTML='';}
async function openSession(id){
  sessionId=id;document.getElementById('log').innerHTML='';
  const r=await fetch('/api/v1/sessions/'+id,{headers:H()});
  const s=await r.json();
  for(const it of s.interactions||[]){
    add('user',it.prompt_message);
    if(it.response_message)add('assistant',it.response_message);}
}
async function send(){
  const inp=document.getElementById('inp');const text=inp.value.trim();
  if(!text)return; inp.value='';
  add('user',text);
  const out=add('assistant','');
  const body={messages:[{role:'user',content:text}],
              model:document.getElementById('model').value};
  if(sessionId)body.session_id=sessionId;
  const r=await fetch('/api/v1/sessions/chat',{method:'POST',headers:H(),
                      body:JSON.stringify(body)});
  const rd=r.body.getReader();const dec=new TextDecoder();let buf='';
  while(true){
    const {done,value}=await rd.read();if(done)break;
    buf+=dec.decode(value,{stream:true});
    let i;while((i=buf.indexOf('\\n\\n'))>=0){
      const line=buf.slice(0,i);buf=buf.slice(i+2);
      if(!line.startsWith('data: '))continue;
      const p=line.slice(6);if(p==='[DONE]')continue;
      try{const c=JSON.parse(p);
        if(c.type==='session'){sessionId=c.session_id;loadSessions();}
        else if(c.choices){out.textContent+=c.choices[0].delta?.content||'';}
        else if(c.error){out.textContent+='[error] '+c.error.message;}
      }catch(e){}
      out.scrollIntoView();
    }}
}
loadModels();loadSessions();
</script></body></html>"""
