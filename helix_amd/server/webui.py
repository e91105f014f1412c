"""Built-in web UI (the reference ships a 237k-LoC React frontend; this
is a dependency-free single-page app served at / covering the main
panes: chat sessions with SSE streaming, apps/helix.yaml, knowledge,
runner dashboard, and usage — so the stack is usable from a browser
without a separate frontend build)."""

INDEX_HTML = """<!doctype html>
<html><head><meta charset="utf-8"><title>helix_amd</title>
<style>
 body{font-family:system-ui,sans-serif;margin:0;display:flex;height:100vh}
 #side{width:260px;background:#111;color:#eee;padding:12px;overflow-y:auto;
       display:flex;flex-direction:column}
 #side h1{font-size:16px;margin:4px 0 10px}
 #tabs{display:flex;flex-wrap:wrap;gap:4px;margin-bottom:10px}
 #tabs button{background:#333;color:#eee;border:0;padding:6px 10px;
   border-radius:6px;cursor:pointer;font-size:12px}
 #tabs button.on{background:#4a7}
 #side div.s{padding:6px;cursor:pointer;border-radius:6px;font-size:13px}
 #side div.s:hover{background:#333}
 #main{flex:1;display:flex;flex-direction:column}
 .pane{flex:1;display:none;flex-direction:column;overflow:hidden}
 .pane.on{display:flex}
 #log{flex:1;overflow-y:auto;padding:16px;background:#f7f7f8}
 .msg{max-width:760px;margin:8px auto;padding:10px 14px;border-radius:10px;
   white-space:pre-wrap}
 .user{background:#d8e8ff} .assistant{background:#fff;border:1px solid #ddd}
 #bar{display:flex;padding:12px;gap:8px;border-top:1px solid #ddd}
 #inp{flex:1;padding:10px;font-size:15px}
 input,select,button,textarea{font-size:14px}
 #cfg{padding:8px 12px;display:flex;gap:8px;background:#eee;
   align-items:center}
 .lst{flex:1;overflow-y:auto;padding:16px}
 table{border-collapse:collapse;width:100%;font-size:13px}
 td,th{border:1px solid #ddd;padding:6px 8px;text-align:left}
 textarea{width:100%;height:180px;font-family:monospace}
 .card{background:#fff;border:1px solid #ddd;border-radius:8px;
   padding:12px;margin-bottom:12px}
 .pill{display:inline-block;padding:2px 8px;border-radius:10px;
   font-size:11px;background:#dfd}
 .pill.err{background:#fdd}
</style></head><body>
<div id="side"><h1>helix_amd</h1>
 <div id="tabs">
  <button data-p="chat" class="on">Chat</button>
  <button data-p="apps">Apps</button>
  <button data-p="knw">Knowledge</button>
  <button data-p="run">Runners</button>
  <button data-p="prj">Projects</button>
  <button data-p="org">Orgs</button>
  <button data-p="sbx">Sandboxes</button>
  <button data-p="img">Images</button>
  <button data-p="use">Usage</button>
 </div>
 <button onclick="newSession()">+ new session</button>
 <div id="sessions"></div>
</div>
<div id="main">
 <div id="cfg">
  key <input id="key" size="14" value="admin-key">
  model <select id="model"></select>
  <input id="srch" size="18" placeholder="search..."
   onkeydown="if(event.key==='Enter')doSearch()">
  <span id="srchout" style="font-size:12px"></span>
  <span id="status"></span>
 </div>
 <div class="pane on" id="p-chat">
  <div id="log"></div>
  <div id="bar">
   <input id="inp" placeholder="Say something..."
    onkeydown="if(event.key==='Enter')send()">
   <button onclick="send()">send</button>
  </div>
 </div>
 <div class="pane" id="p-apps"><div class="lst">
  <div class="card"><b>Apply helix.yaml / app config (JSON or CRD)</b>
   <textarea id="appyaml">{"name":"my-agent","assistants":[{"name":"default",
 "system_prompt":"You are helpful."}]}</textarea>
   <button onclick="applyApp()">apply</button> <span id="appmsg"></span>
  </div>
  <div id="applist"></div>
 </div></div>
 <div class="pane" id="p-knw"><div class="lst">
  <div class="card"><b>New knowledge source (text)</b><br>
   name <input id="kname" value="notes">
   <textarea id="ktext">Paste source text here.</textarea>
   <button onclick="addKnowledge()">index</button>
  </div>
  <div id="knwlist"></div>
 </div></div>
 <div class="pane" id="p-run"><div class="lst" id="runlist"></div></div>
 <div class="pane" id="p-prj"><div class="lst">
  <div class="card"><b>Projects</b>
   <input id="pname" placeholder="new project name">
   <button onclick="newProject()">create</button>
   <select id="psel" onchange="loadBoard()"></select>
   <input id="tname" placeholder="new task title">
   <button onclick="newTask()">add task</button>
  </div>
  <div id="board" style="display:flex;gap:8px;align-items:flex-start">
  </div>
 </div></div>
 <div class="pane" id="p-org"><div class="lst">
  <div class="card"><b>Organizations</b>
   <input id="oname" placeholder="new org name">
   <button onclick="newOrg()">create</button>
  </div>
  <div id="orglist"></div>
 </div></div>
 <div class="pane" id="p-sbx"><div class="lst">
  <div class="card"><b>Sandboxes</b>
   <button onclick="newSandbox()">new sandbox</button>
   <select id="ssel"></select>
   <input id="scmd" size="40" placeholder="command"
    onkeydown="if(event.key==='Enter')runCmd()">
   <button onclick="runCmd()">run</button>
   <pre id="sout" style="background:#111;color:#9f9;padding:8px;
     min-height:80px;white-space:pre-wrap"></pre>
  </div>
 </div></div>
 <div class="pane" id="p-img"><div class="lst">
  <div class="card"><b>Image generation</b><br>
   <input id="iprompt" size="40" placeholder="prompt"
    onkeydown="if(event.key==='Enter')genImage()">
   model <input id="imodel" size="10" value="flux-lite">
   n <input id="inum" size="2" value="1">
   <button onclick="genImage()">generate</button>
   <span id="imsg"></span>
  </div>
  <div id="imgout" style="display:flex;gap:8px;flex-wrap:wrap"></div>
 </div></div>
 <div class="pane" id="p-use"><div class="lst" id="uselist"></div></div>
</div>
<script>
let sessionId = null;
const $ = id => document.getElementById(id);
const esc = x => String(x==null?'':x).replace(/[&<>"']/g,
  ch => ({'&':'&amp;','<':'&lt;','>':'&gt;','"':'&quot;',
          "'":'&#39;'}[ch]));
const H = () => ({'Authorization':'Bearer '+$('key').value,
                  'Content-Type':'application/json'});
for(const b of document.querySelectorAll('#tabs button')){
  b.onclick = () => {
    document.querySelectorAll('#tabs button').forEach(x=>x.classList
      .remove('on'));
    document.querySelectorAll('.pane').forEach(x=>x.classList.remove('on'));
    b.classList.add('on');
    $('p-'+b.dataset.p).classList.add('on');
    ({apps:loadApps, knw:loadKnowledge, run:loadRunners,
      use:loadUsage, prj:loadProjects, org:loadOrgs,
      sbx:loadSandboxes})[b.dataset.p]?.();
  };
}
async function loadModels(){
  const r = await fetch('/v1/models',{headers:H()});
  if(!r.ok) return;
  const sel = $('model'); sel.innerHTML='';
  for(const m of (await r.json()).data){
    const o=document.createElement('option');
    o.value=o.textContent=m.id;sel.appendChild(o);}
}
async function loadSessions(){
  const r = await fetch('/api/v1/sessions',{headers:H()});
  if(!r.ok) return;
  const el = $('sessions'); el.innerHTML='';
  for(const s of await r.json()){
    const d=document.createElement('div');d.className='s';
    d.textContent=s.name;
    d.onclick=()=>openSession(s.id);el.appendChild(d);}
}
function add(role, text){
  const d=document.createElement('div');d.className='msg '+role;
  d.textContent=text;
  $('log').appendChild(d);
  d.scrollIntoView();return d;}
function newSession(){sessionId=null;$('log').innerHTML='';}
async function openSession(id){
  sessionId=id;$('log').innerHTML='';
  const r=await fetch('/api/v1/sessions/'+id,{headers:H()});
  const s=await r.json();
  for(const it of s.interactions||[]){
    add('user',it.prompt_message);
    if(it.response_message)add('assistant',it.response_message);}
}
async function send(){
  const inp=$('inp');const text=inp.value.trim();
  if(!text)return; inp.value='';
  add('user',text);
  const out=add('assistant','');
  const body={messages:[{role:'user',content:text}],
              model:$('model').value};
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
// ---- apps pane ----
async function applyApp(){
  let cfg;
  try{cfg=JSON.parse($('appyaml').value);}catch(e){
    $('appmsg').textContent='bad JSON: '+e;return;}
  const r=await fetch('/api/v1/apps',{method:'POST',headers:H(),
    body:JSON.stringify({config:cfg})});
  $('appmsg').textContent=r.ok?'applied':'error '+r.status;
  loadApps();
}
async function loadApps(){
  const r=await fetch('/api/v1/apps',{headers:H()});if(!r.ok)return;
  const el=$('applist');el.innerHTML='';
  for(const a of await r.json()){
    const d=document.createElement('div');d.className='card';
    const cfg=a.config?.helix||a.config||{};
    d.innerHTML='<b>'+esc(cfg.name||a.id)+'</b> <span class="pill">'+
      (cfg.assistants?.length||0)+' assistants</span><br><small>'+a.id+
      '</small>';
    el.appendChild(d);}
}
// ---- knowledge pane ----
async function addKnowledge(){
  await fetch('/api/v1/knowledge',{method:'POST',headers:H(),
    body:JSON.stringify({name:$('kname').value,
      source:{text:$('ktext').value}})});
  loadKnowledge();
}
async function refreshKnw(kid){
  await fetch('/api/v1/knowledge/'+kid+'/refresh',{method:'POST',
    headers:H()});loadKnowledge();
}
async function loadKnowledge(){
  const r=await fetch('/api/v1/knowledge',{headers:H()});if(!r.ok)return;
  const el=$('knwlist');el.innerHTML='';
  for(const k of await r.json()){
    const d=document.createElement('div');d.className='card';
    const prog=(k.state==='indexing'&&k.progress_percent!=null)
      ?(' '+k.progress_percent+'%'):'';
    d.innerHTML='<b>'+esc(k.name)+'</b> <span class="pill'+
      (k.state==='error'?' err':'')+'">'+k.state+prog+'</span>'+
      (k.version?' v'+k.version:'')+
      ' <button onclick="refreshKnw(\''+k.id+'\')">refresh</button>';
    el.appendChild(d);}
}
// ---- runners pane ----
async function loadRunners(){
  const r=await fetch('/api/v1/admin/runners',{headers:H()});
  const el=$('runlist');
  if(!r.ok){el.textContent='admin key required';return;}
  const rs=await r.json();el.innerHTML='';
  if(!rs.length){el.textContent='no runners connected';return;}
  for(const x of rs){
    const d=document.createElement('div');d.className='card';
    const models=(x.models||[]).map(m=>m.model_id||m).join(', ');
    d.innerHTML='<b>'+esc(x.id)+'</b> <span class="pill">'+
      (x.status||'ready')+'</span><br>GPU: '+(x.gpu||'?')+
      '<br>models: '+models;
    el.appendChild(d);}
}
// ---- usage pane ----
async function loadUsage(){
  const r=await fetch('/api/v1/usage',{headers:H()});
  const el=$('uselist');
  if(!r.ok){el.textContent='no usage yet';return;}
  const u=await r.json();
  el.innerHTML='<div class="card"><pre>'+
    JSON.stringify(u,null,2).slice(0,4000)+'</pre></div>';
}
// ---- projects / kanban pane (reference frontend kanban board) ----
const STATES=['backlog','planning','spec_review','in_progress','pr',
              'merged','failed'];
async function loadProjects(){
  const r=await fetch('/api/v1/projects',{headers:H()});if(!r.ok)return;
  const sel=$('psel');const cur=sel.value;sel.innerHTML='';
  for(const p of await r.json()){
    const o=document.createElement('option');
    o.value=p.id;o.textContent=p.name||p.id;sel.appendChild(o);}
  if(cur)sel.value=cur;
  loadBoard();
}
async function newProject(){
  await fetch('/api/v1/projects',{method:'POST',headers:H(),
    body:JSON.stringify({name:$('pname').value})});
  $('pname').value='';loadProjects();
}
async function newTask(){
  const pid=$('psel').value;if(!pid)return;
  await fetch('/api/v1/projects/'+pid+'/tasks',{method:'POST',
    headers:H(),body:JSON.stringify({title:$('tname').value})});
  $('tname').value='';loadBoard();
}
async function moveTask(tid,st){
  await fetch('/api/v1/spec-tasks/'+tid+'/transition',{method:'POST',
    headers:H(),body:JSON.stringify({state:st})});
  loadBoard();
}
async function planTask(tid){
  await fetch('/api/v1/spec-tasks/'+tid+'/plan',{method:'POST',
    headers:H()});loadBoard();
}
async function implementTask(tid){
  await fetch('/api/v1/spec-tasks/'+tid+'/implement',{method:'POST',
    headers:H()});loadBoard();
}
async function mergeTask(tid){
  await fetch('/api/v1/spec-tasks/'+tid+'/merge',{method:'POST',
    headers:H()});loadBoard();
}
async function loadBoard(){
  const pid=$('psel').value;const el=$('board');el.innerHTML='';
  if(!pid)return;
  const r=await fetch('/api/v1/projects/'+pid+'/tasks',{headers:H()});
  if(!r.ok)return;
  const tasks=await r.json();
  for(const st of STATES){
    const col=document.createElement('div');
    col.style.cssText='flex:1;background:#eee;border-radius:8px;'+
      'padding:6px;min-height:120px';
    col.innerHTML='<b style="font-size:12px">'+st+'</b>';
    for(const t of tasks.filter(x=>x.state===st)){
      const c=document.createElement('div');c.className='card';
      c.style.padding='6px';c.style.fontSize='12px';
      let btns='';
      if(st==='backlog')btns='<button onclick="planTask(\''+t.id+
        '\')">plan</button>';
      if(st==='spec_review')btns='<button onclick="implementTask(\''+
        t.id+'\')">implement</button>';
      if(st==='pr')btns='<button onclick="mergeTask(\''+t.id+
        '\')">merge</button>';
      c.innerHTML='<b>'+esc(t.title)+'</b><br>'+btns+
        ' <select onchange="moveTask(\''+t.id+
        '\',this.value)"><option>move...</option>'+
        STATES.map(x=>'<option>'+x+'</option>').join('')+'</select>';
      col.appendChild(c);}
    el.appendChild(col);}
}
// ---- orgs pane (reference org admin) ----
async function newOrg(){
  await fetch('/api/v1/organizations',{method:'POST',headers:H(),
    body:JSON.stringify({name:$('oname').value})});
  $('oname').value='';loadOrgs();
}
async function loadOrgs(){
  const r=await fetch('/api/v1/organizations',{headers:H()});
  if(!r.ok)return;
  const el=$('orglist');el.innerHTML='';
  for(const o of await r.json()){
    const d=document.createElement('div');d.className='card';
    d.innerHTML='<b>'+esc(o.name||o.id)+'</b> <small>'+o.id+'</small>'+
      '<div id="teams-'+o.id+'"></div>'+
      '<input id="tn-'+o.id+'" placeholder="team name" size="12">'+
      '<button onclick="newTeam(\''+o.id+'\')">add team</button> '+
      '<input id="mu-'+o.id+'" placeholder="user id" size="16">'+
      '<button onclick="addMember(\''+o.id+'\')">add member</button>';
    el.appendChild(d);
    fetch('/api/v1/organizations/'+o.id+'/teams',{headers:H()})
      .then(r=>r.ok?r.json():[]).then(ts=>{
        $('teams-'+o.id).textContent='teams: '+
          (ts.map(t=>t.name).join(', ')||'none');});
  }
}
async function newTeam(oid){
  await fetch('/api/v1/organizations/'+oid+'/teams',{method:'POST',
    headers:H(),body:JSON.stringify({name:$('tn-'+oid).value})});
  loadOrgs();
}
async function addMember(oid){
  await fetch('/api/v1/organizations/'+oid+'/members',{method:'POST',
    headers:H(),body:JSON.stringify({user_id:$('mu-'+oid).value})});
  loadOrgs();
}
// ---- sandboxes pane (reference dev-container exec) ----
async function loadSandboxes(){
  const r=await fetch('/api/v1/sandboxes',{headers:H()});if(!r.ok)return;
  const sel=$('ssel');const cur=sel.value;sel.innerHTML='';
  for(const s of await r.json()){
    const o=document.createElement('option');
    o.value=s.id;o.textContent=s.name||s.id;sel.appendChild(o);}
  if(cur)sel.value=cur;
}
async function newSandbox(){
  await fetch('/api/v1/sandboxes',{method:'POST',headers:H(),
    body:JSON.stringify({name:'ui-'+Date.now()})});
  loadSandboxes();
}
async function runCmd(){
  const sid=$('ssel').value;if(!sid)return;
  $('sout').textContent+='$ '+$('scmd').value+'\n';
  const r=await fetch('/api/v1/sandboxes/'+sid+'/exec',{method:'POST',
    headers:H(),body:JSON.stringify({command:$('scmd').value})});
  if(r.ok){const o=await r.json();
    $('sout').textContent+=o.stdout+(o.stderr||'')+
      (o.exit_code?('[exit '+o.exit_code+']\n'):'');}
  else $('sout').textContent+='[error '+r.status+']\n';
  $('scmd').value='';
}
async function genImage(){
  $('imsg').textContent='generating...';
  const r = await fetch('/v1/images/generations',{method:'POST',
    headers:H(),body:JSON.stringify({prompt:$('iprompt').value,
      model:$('imodel').value,n:parseInt($('inum').value)||1})});
  if(!r.ok){$('imsg').textContent='error '+r.status;return;}
  $('imsg').textContent='';
  const out = $('imgout'); out.innerHTML='';
  for(const d of (await r.json()).data){
    const im=document.createElement('img');
    im.src='data:image/png;base64,'+d.b64_json;
    im.style.cssText='max-width:256px;border:1px solid #ddd;'+
      'border-radius:8px';
    out.appendChild(im);}
}
async function doSearch(){
  const q=$('srch').value.trim();if(!q)return;
  const r=await fetch('/api/v1/search?q='+encodeURIComponent(q),
    {headers:H()});
  if(!r.ok){$('srchout').textContent='error';return;}
  const o=await r.json();
  const n=o.sessions.length+o.apps.length+o.knowledge.length+
    o.tasks.length;
  $('srchout').textContent=n+' hits: '+
    o.sessions.slice(0,3).map(s=>s.name).join(', ');
  if(o.sessions.length)openSession(o.sessions[0].id);
}
loadModels();loadSessions();
</script></body></html>"""
