"""Built-in dashboard (reference ships a React SPA, src/ui/, served statically
by the API server; that code is reference material and is not copied). This is
a self-contained single-file dashboard over the same API + WS: rooms, status,
workers, goals, decisions, activity, clerk chat, live event stream."""

DASHBOARD_HTML = r"""<!doctype html>
<html lang="en"><head><meta charset="utf-8">
<title>room_amd</title>
<style>
 :root { color-scheme: dark; }
 body { font-family: ui-monospace, Menlo, monospace; background:#0d1117;
        color:#d0d7de; margin:0; }
 header { padding:10px 16px; background:#161b22; display:flex; gap:16px;
          align-items:center; border-bottom:1px solid #30363d; }
 h1 { font-size:15px; margin:0; color:#7ee787; }
 main { display:grid; grid-template-columns: 260px 1fr 330px; gap:10px;
        padding:10px; height:calc(100vh - 58px); box-sizing:border-box; }
 section { background:#161b22; border:1px solid #30363d; border-radius:6px;
           padding:10px; overflow:auto; }
 h2 { font-size:12px; text-transform:uppercase; color:#8b949e; margin:0 0 8px; }
 .room { padding:6px 8px; border-radius:4px; cursor:pointer; }
 .room:hover, .room.sel { background:#21262d; }
 .tag { font-size:10px; padding:1px 6px; border-radius:8px; background:#21262d;
        color:#7ee787; margin-left:6px; }
 .tag.paused { color:#d29922; } .tag.stopped { color:#f85149; }
 .ev { font-size:11px; padding:3px 0; border-bottom:1px dotted #21262d; }
 .bar { background:#21262d; height:6px; border-radius:3px; margin-top:2px; }
 .bar>div { background:#58a6ff; height:6px; border-radius:3px; }
 button, input { background:#21262d; color:#d0d7de; border:1px solid #30363d;
        border-radius:4px; padding:5px 8px; font:inherit; }
 button:hover { background:#30363d; cursor:pointer; }
 #chatlog { height:140px; overflow:auto; font-size:11px; }
 table { width:100%; font-size:11px; border-collapse:collapse; }
 td, th { text-align:left; padding:3px 4px; border-bottom:1px solid #21262d; }
 .muted { color:#8b949e; }
</style></head><body>
<header>
 <h1>room_amd</h1><span class="muted" id="status">connecting…</span>
 <span style="flex:1"></span>
 <input id="newroom" placeholder="new room name">
 <button onclick="createRoom()">create room</button>
</header>
<main>
 <section>
  <h2>Rooms</h2><div id="rooms"></div>
  <h2 style="margin-top:14px">Clerk</h2>
  <div id="chatlog"></div>
  <div style="display:flex;gap:4px;margin-top:6px">
   <input id="chatin" style="flex:1" placeholder="ask the clerk…"
          onkeydown="if(event.key==='Enter')clerkSend()">
   <button onclick="clerkSend()">send</button>
  </div>
 </section>
 <section>
  <h2 id="roomtitle">Room</h2>
  <div id="roomctl" style="margin-bottom:8px"></div>
  <nav id="tabs" style="margin-bottom:8px"></nav>
  <div id="tabbody"></div>
 </section>
 <section><h2>Live activity</h2><div id="events"></div></section>
</main>
<script>
let token = localStorage.getItem('roomamd_token');
let sel = null;
const $$ = id => document.getElementById(id);
async function api(path, opts) {
  const r = await fetch('/api' + path, Object.assign({
    headers: {'Authorization': 'Bearer ' + token,
              'Content-Type': 'application/json'}}, opts));
  if (r.status === 401) { await handshake(); return api(path, opts); }
  return r.json();
}
async function handshake() {
  const r = await fetch('/api/auth/handshake', {method: 'POST'});
  token = (await r.json()).token;
  localStorage.setItem('roomamd_token', token);
}
async function loadRooms() {
  const rooms = await api('/rooms');
  $$('rooms').innerHTML = rooms.map(r =>
    `<div class="room ${sel===r.id?'sel':''}" onclick="select(${r.id})">` +
    `${r.name}<span class="tag ${r.status}">${r.status}</span></div>`).join('');
  if (sel === null && rooms.length) select(rooms[0].id);
}
const TABS = ['overview','tasks','skills','memory','messages','wallet','credentials','settings','status','help'];
let tab = 'overview';
function tabbar() {
  $$('tabs').innerHTML = TABS.map(t =>
    `<button style="${t===tab?'background:#30363d':''}" ` +
    `onclick="setTab('${t}')">${t}</button>`).join(' ');
}
async function setTab(t) { tab = t; if (sel !== null) await select(sel); }
async function select(id) {
  sel = id; loadRooms(); tabbar();
  const st = await api(`/rooms/${id}/status`);
  $$('roomtitle').textContent = `${st.room.name} — ${st.room.goal || 'no goal'}`;
  $$('roomctl').innerHTML =
    `<button onclick="roomAct(${id},'start')">start</button> ` +
    `<button onclick="roomAct(${id},'pause')">pause</button> ` +
    `<button onclick="roomAct(${id},'restart')">restart</button>` +
    ` <span class="muted">cycles: ${st.token_usage.cycles}</span>`;
  const b = $$('tabbody');
  if (tab === 'overview') {
    const decs = await api(`/rooms/${id}/decisions`);
    b.innerHTML = '<h2>Goals</h2>' + (st.goals.map(g =>
      `<div class="ev">${g.description} <span class="muted">${g.status}</span>` +
      `<div class="bar"><div style="width:${Math.round((g.progress||0)*100)}%"></div></div></div>`
    ).join('') || '<span class="muted">none</span>') +
    '<h2 style="margin-top:10px">Workers</h2><table>' +
    '<tr><th>name</th><th>role</th><th>state</th></tr>' +
    st.workers.map(w => `<tr><td>${w.name}</td><td>${w.role||''}</td>` +
                        `<td>${w.agent_state}</td></tr>`).join('') + '</table>' +
    '<h2 style="margin-top:10px">Decisions</h2><table>' +
    '<tr><th>proposal</th><th>status</th></tr>' +
    decs.slice(0, 8).map(d => `<tr><td>${d.proposal}</td><td>${d.status}</td></tr>`).join('')
    + '</table>';
  } else if (tab === 'tasks') {
    const ts = await api(`/tasks?room_id=${id}`);
    b.innerHTML = '<table><tr><th>name</th><th>trigger</th><th>status</th>' +
      '<th>runs</th><th></th></tr>' + ts.map(t =>
      `<tr><td>${t.name}</td><td>${t.cron_expression||t.trigger_type}</td>` +
      `<td>${t.status}</td><td>${t.run_count}</td>` +
      `<td><button onclick="api('/tasks/${t.id}/run',{method:'POST'})">run</button></td></tr>`
      ).join('') + '</table>';
  } else if (tab === 'skills') {
    const sk = await api(`/rooms/${id}/skills`);
    b.innerHTML = '<table><tr><th>name</th><th>v</th><th>auto</th><th>content</th></tr>' +
      sk.map(x => `<tr><td>${x.name}</td><td>${x.version}</td>` +
      `<td>${x.auto_activate?'on':''}</td>` +
      `<td class="muted">${(x.content||'').slice(0,120)}</td></tr>`).join('') + '</table>';
  } else if (tab === 'memory') {
    const ents = await api(`/memory/entities?room_id=${id}&limit=50`);
    b.innerHTML = '<div style="display:flex;gap:4px;margin-bottom:6px">' +
      '<input id="memq" style="flex:1" placeholder="hybrid search…" ' +
      'onkeydown="if(event.key===\'Enter\')memSearch()">' +
      '<button onclick="memSearch()">search</button></div><div id="memres"></div>' +
      '<table><tr><th>id</th><th>name</th><th>type</th><th>created</th></tr>' +
      ents.map(e => `<tr><td>${e.id}</td><td>${e.name}</td><td>${e.type}</td>` +
      `<td class="muted">${e.created_at}</td></tr>`).join('') + '</table>';
  } else if (tab === 'messages') {
    const ms = await api(`/rooms/${id}/messages`);
    const es = await api(`/rooms/${id}/escalations`);
    b.innerHTML = '<h2>Escalations</h2>' + (es.map(e =>
      `<div class="ev">${e.question} <span class="muted">${e.status}</span></div>`
      ).join('') || '<span class="muted">none</span>') +
      '<h2 style="margin-top:10px">Inter-room messages</h2>' + (ms.map(m =>
      `<div class="ev"><b>${m.subject}</b> <span class="muted">${m.direction} ` +
      `${m.status}</span><br>${(m.body||'').slice(0,200)}</div>`).join('')
      || '<span class="muted">none</span>');
  } else if (tab === 'wallet') {
    const w = await api(`/rooms/${id}/wallet`);
    const tx = await api(`/rooms/${id}/wallet/transactions`);
    b.innerHTML = `<div class="ev">address: <b>${w.address||'—'}</b></div>` +
      '<h2 style="margin-top:10px">Transactions</h2><table>' +
      '<tr><th>type</th><th>amount</th><th>counterparty</th><th>status</th></tr>' +
      tx.map(t => `<tr><td>${t.type}</td><td>${t.amount}</td>` +
      `<td class="muted">${(t.counterparty||'—').slice(0,14)}</td>` +
      `<td>${t.status}</td></tr>`).join('') + '</table>';
  } else if (tab === 'credentials') {
    const cs = await api(`/rooms/${id}/credentials`);
    b.innerHTML = '<div style="display:flex;gap:4px;margin-bottom:6px">' +
      '<input id="credname" placeholder="name">' +
      '<input id="credval" placeholder="value" type="password" style="flex:1">' +
      '<button onclick="credAdd()">add</button></div>' +
      '<table><tr><th>name</th><th>type</th><th>by</th><th>created</th></tr>' +
      cs.map(x => `<tr><td>${x.name}</td><td>${x.type}</td>` +
      `<td>${x.provided_by}</td><td class="muted">${x.created_at}</td></tr>`
      ).join('') + '</table>' +
      '<div class="muted" style="margin-top:6px">values are AES-256-GCM ' +
      'encrypted at rest; agents read them with room_get_credential</div>';
  } else if (tab === 'status') {
    const st2 = await api('/status');
    const prof = await api('/status/http-profile');
    const lm = await api('/local-model/status');
    b.innerHTML = '<h2>Server</h2><pre class="muted" style="font-size:11px">' +
      JSON.stringify(st2, null, 1) + '</pre>' +
      '<h2>Engine</h2><pre class="muted" style="font-size:11px">' +
      JSON.stringify(lm, null, 1) + '</pre>' +
      '<h2>HTTP profile</h2><pre class="muted" style="font-size:11px">' +
      JSON.stringify(prof, null, 1).slice(0, 3000) + '</pre>';
  } else if (tab === 'help') {
    b.innerHTML = `
      <h2>room_amd — MI355X-native agent swarm</h2>
      <div class="ev">Each room runs a <b>queen</b> (control plane: goals,
      delegation, quorum announcements) and <b>workers</b> (executors).
      Agents cycle observe → prompt → decode on the in-process CDNA4
      engine → tools → persist. Decisions use announce-and-object with
      keeper override.</div>
      <div class="ev"><b>Start here:</b> create a room, give it a goal,
      press start. Watch live activity on the right; answer escalations in
      the messages tab; fund the wallet from the wallet tab.</div>
      <div class="ev"><b>Interfaces:</b> this dashboard, the REST API
      (docs/API.md), the MCP stdio server (<code>room-amd mcp</code>, 95
      tools) and webhooks (tasks + queen wake).</div>
      <div class="ev"><b>Keeper contact:</b> verify an email under
      contacts so escalations reach you; or read the outbox at
      ~/.roomamd/outbox.jsonl.</div>`;
  } else if (tab === 'settings') {
    const room = await api(`/rooms/${id}`);
    const vh = await api(`/rooms/${id}/voter-health`);
    b.innerHTML = '<h2>Room config</h2><pre class="muted" style="font-size:11px">' +
      JSON.stringify(room.config || {}, null, 1) + '</pre>' +
      '<h2>Voter health</h2><table><tr><th>worker</th><th>cast</th>' +
      '<th>missed</th><th>healthy</th></tr>' +
      vh.map(v => `<tr><td>${v.worker_name}</td><td>${v.votes_cast}</td>` +
      `<td>${v.votes_missed}</td><td>${v.is_healthy?'yes':'NO'}</td></tr>`
      ).join('') + '</table>';
  }
}
async function credAdd() {
  const name = $$('credname').value.trim(), value = $$('credval').value;
  if (!name || !value) return;
  await api(`/rooms/${sel}/credentials`, {method:'POST',
            body: JSON.stringify({name, value})});
  setTab('credentials');
}
async function memSearch() {
  const qy = $$('memq').value.trim();
  if (!qy) return;
  const hits = await api(`/memory/search?query=${encodeURIComponent(qy)}&room_id=${sel}`);
  $$('memres').innerHTML = hits.map(h =>
    `<div class="ev"><b>${h.name}</b> <span class="muted">${(h.score||0).toFixed(3)}</span></div>`
    ).join('') || '<span class="muted">no hits</span>';
}
async function roomAct(id, act) { await api(`/rooms/${id}/${act}`, {method:'POST'}); select(id); }
async function createRoom() {
  const name = $$('newroom').value.trim();
  if (!name) return;
  await api('/rooms', {method:'POST', body: JSON.stringify({name})});
  $$('newroom').value = ''; loadRooms();
}
async function clerkSend() {
  const content = $$('chatin').value.trim();
  if (!content) return;
  $$('chatin').value = '';
  chatLine('you', content);
  const r = await api('/clerk/chat', {method:'POST',
                                      body: JSON.stringify({content})});
  chatLine('clerk', r.reply || '(no reply)');
}
function chatLine(who, text) {
  $$('chatlog').innerHTML += `<div class="ev"><b>${who}:</b> ${text}</div>`;
  $$('chatlog').scrollTop = 1e9;
}
function eventLine(e) {
  $$('events').innerHTML =
    `<div class="ev"><span class="muted">${e.channel}</span> ${e.type}` +
    (e.data && e.data.content ? ' — ' + e.data.content : '') + '</div>' +
    $$('events').innerHTML.slice(0, 40000);
}
async function connectWs() {
  const ws = new WebSocket(
    `${location.protocol==='https:'?'wss':'ws'}://${location.host}/ws?token=${token}`);
  ws.onopen = () => {
    $$('status').textContent = 'live';
    ws.send(JSON.stringify({type:'subscribe', channel:'*'}));
    ['rooms','runs','clerk'].forEach(c =>
      ws.send(JSON.stringify({type:'subscribe', channel:c})));
  };
  ws.onmessage = m => {
    const e = JSON.parse(m.data);
    if (e.type === 'ping') return;
    eventLine(e);
    if (['room_created','cycle_finished'].includes(e.type)) loadRooms();
  };
  ws.onclose = () => { $$('status').textContent = 'reconnecting…';
                       setTimeout(connectWs, 2000); };
}
(async () => {
  if (!token) await handshake();
  await loadRooms();
  connectWs();
  setInterval(() => { if (sel !== null) select(sel); }, 10000);
})();
</script></body></html>
"""
