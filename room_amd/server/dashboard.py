"""Built-in dashboard (reference ships a 13k-LoC React SPA, src/ui/, served
statically by the API server; that code is reference material and is not
copied). This is a self-contained single-file dashboard over the same API +
WS with tab parity to the SPA: rooms sidebar + clerk chat, and per-room
overview / goals / votes / workers / tasks / skills / memory / messages /
wallet / credentials / room-settings tabs plus global settings, status and
help — each with its write actions (create/vote/object/keeper-vote, answer
escalations, add workers/tasks/skills/memories/credentials, edit settings,
check updates), not just read views."""

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
 .badge { font-size:10px; background:#d29922; color:#000; border-radius:8px;
        padding:0 5px; margin-left:4px; }
 .ev { font-size:11px; padding:3px 0; border-bottom:1px dotted #21262d; }
 .bar { background:#21262d; height:6px; border-radius:3px; margin-top:2px; }
 .bar>div { background:#58a6ff; height:6px; border-radius:3px; }
 button, input, select { background:#21262d; color:#d0d7de;
        border:1px solid #30363d; border-radius:4px; padding:5px 8px;
        font:inherit; }
 button:hover { background:#30363d; cursor:pointer; }
 button.mini { font-size:10px; padding:2px 6px; }
 #chatlog { height:140px; overflow:auto; font-size:11px; }
 table { width:100%; font-size:11px; border-collapse:collapse; }
 td, th { text-align:left; padding:3px 4px; border-bottom:1px solid #21262d; }
 .muted { color:#8b949e; }
 .row { display:flex; gap:4px; margin-bottom:6px; flex-wrap:wrap; }
 .row input { flex:1; min-width:60px; }
</style></head><body>
<header>
 <h1>room_amd</h1><span class="muted" id="status">connecting…</span>
 <span style="flex:1"></span>
 <input id="newroom" placeholder="new room name">
 <button onclick="createRoom()">create room</button>
 <button onclick="createFromTemplate()">from template…</button>
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
const esc = s => String(s ?? '').replace(/[&<>"]/g,
  c => ({'&':'&amp;','<':'&lt;','>':'&gt;','"':'&quot;'}[c]));
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
  const parts = await Promise.all(rooms.map(async r => {
    let badge = '';
    try {
      const b = await api(`/rooms/${r.id}/badges`);
      const n = b.pending_escalations + b.active_votes;
      if (n) badge = `<span class="badge">${n}</span>`;
    } catch (e) {}
    return `<div class="room ${sel===r.id?'sel':''}" onclick="select(${r.id})">` +
      `${esc(r.name)}<span class="tag ${r.status}">${r.status}</span>${badge}</div>`;
  }));
  $$('rooms').innerHTML = parts.join('');
  if (sel === null && rooms.length) select(rooms[0].id);
}
const TABS = ['overview','goals','votes','workers','tasks','skills','memory',
              'messages','wallet','credentials','room-settings','settings',
              'status','help'];
let tab = 'overview';
function tabbar() {
  $$('tabs').innerHTML = TABS.map(t =>
    `<button class="mini" style="${t===tab?'background:#30363d':''}" ` +
    `onclick="setTab('${t}')">${t}</button>`).join(' ');
}
async function setTab(t) { tab = t; if (sel !== null) await select(sel); }

const RENDER = {
 async overview(id, st) {
  const acts = await api(`/rooms/${id}/activity`);
  return '<h2>Goals</h2>' + (st.goals.map(g =>
    `<div class="ev">${esc(g.description)} <span class="muted">${g.status}</span>` +
    `<div class="bar"><div style="width:${Math.round((g.progress||0)*100)}%"></div></div></div>`
  ).join('') || '<span class="muted">none</span>') +
  '<h2 style="margin-top:10px">Workers</h2><table>' +
  '<tr><th>name</th><th>role</th><th>state</th></tr>' +
  st.workers.map(w => `<tr><td>${esc(w.name)}</td><td>${esc(w.role||'')}</td>` +
                      `<td>${w.agent_state}</td></tr>`).join('') + '</table>' +
  '<h2 style="margin-top:10px">Recent activity</h2>' +
  acts.slice(0, 12).map(a =>
    `<div class="ev"><span class="muted">${a.event_type}</span> ${esc(a.description)}</div>`
  ).join('');
 },
 async goals(id, st) {
  return `<div class="row"><input id="goaldesc" placeholder="new goal…">` +
    `<button onclick="goalAdd()">add goal</button></div>` +
    '<table><tr><th>goal</th><th>status</th><th>progress</th><th></th></tr>' +
    st.goals.map(g => `<tr><td>${esc(g.description)}</td><td>${g.status}</td>` +
      `<td><div class="bar"><div style="width:${Math.round((g.progress||0)*100)}%"></div></div></td>` +
      `<td><button class="mini" onclick="goalDone(${g.id})">complete</button> ` +
      `<button class="mini" onclick="goalDrop(${g.id})">abandon</button></td></tr>`
    ).join('') + '</table>';
 },
 async votes(id) {
  const decs = await api(`/rooms/${id}/decisions`);
  return `<div class="row"><input id="propin" placeholder="announce a decision…">` +
    `<select id="proptype"><option>low_impact</option><option>strategy</option>` +
    `<option>resource</option><option>high_impact</option></select>` +
    `<button onclick="propose()">announce</button></div>` +
    '<table><tr><th>proposal</th><th>type</th><th>status</th><th>result</th><th></th></tr>' +
    decs.map(d => `<tr><td>${esc(d.proposal)}</td><td>${d.decision_type}</td>` +
      `<td>${d.status}</td><td class="muted">${esc(d.result||'')}</td><td>` +
      (['announced','voting'].includes(d.status)
        ? `<button class="mini" onclick="keeperVote(${d.id},'yes')">keeper ✓</button> ` +
          `<button class="mini" onclick="keeperVote(${d.id},'no')">keeper ✗</button>`
        : '') + '</td></tr>').join('') + '</table>';
 },
 async workers(id, st) {
  return `<div class="row"><input id="wname" placeholder="name">` +
    `<select id="wrole"><option>executor</option><option>researcher</option>` +
    `<option>analyst</option><option>writer</option><option>guardian</option></select>` +
    `<button onclick="workerAdd()">add worker</button></div>` +
    '<table><tr><th>name</th><th>role</th><th>state</th><th>model</th><th></th></tr>' +
    st.workers.map(w => `<tr><td>${esc(w.name)}</td><td>${esc(w.role||'')}</td>` +
      `<td>${w.agent_state}</td><td class="muted">${esc(w.model||'room default')}</td>` +
      `<td><button class="mini" onclick="api('/workers/${w.id}/start',{method:'POST'}).then(()=>select(sel))">start</button> ` +
      `<button class="mini" onclick="api('/workers/${w.id}/stop',{method:'POST'}).then(()=>select(sel))">stop</button></td></tr>`
    ).join('') + '</table>' +
    `<div class="row" style="margin-top:8px">` +
    `<button class="mini" onclick="api('/rooms/${id}/prompts/export',{method:'POST'}).then(r=>alert(JSON.stringify(r)))">export prompts</button>` +
    `<button class="mini" onclick="api('/rooms/${id}/prompts/import',{method:'POST',body:'{}'}).then(r=>alert(JSON.stringify(r)))">import prompts</button></div>`;
 },
 async tasks(id) {
  const ts = await api(`/tasks?room_id=${id}`);
  return `<div class="row"><input id="tname" placeholder="task name">` +
    `<input id="tcron" placeholder="cron (blank = once)" style="max-width:130px">` +
    `<input id="tprompt" placeholder="prompt"><button onclick="taskAdd()">add</button></div>` +
    '<table><tr><th>name</th><th>trigger</th><th>status</th><th>runs</th><th></th></tr>' +
    ts.map(t => `<tr><td>${esc(t.name)}</td><td>${esc(t.cron_expression||t.trigger_type)}</td>` +
    `<td>${t.status}</td><td>${t.run_count}</td>` +
    `<td><button class="mini" onclick="api('/tasks/${t.id}/run',{method:'POST'})">run</button> ` +
    `<button class="mini" onclick="api('/tasks/${t.id}/${t.status==='paused'?'resume':'pause'}',{method:'POST'}).then(()=>select(sel))">` +
    `${t.status==='paused'?'resume':'pause'}</button></td></tr>`).join('') + '</table>';
 },
 async skills(id) {
  const sk = await api(`/rooms/${id}/skills`);
  return `<div class="row"><input id="skname" placeholder="skill name">` +
    `<input id="skctx" placeholder="activation context" style="max-width:150px">` +
    `<input id="skcontent" placeholder="content"><button onclick="skillAdd()">add</button></div>` +
    '<table><tr><th>name</th><th>v</th><th>auto</th><th>content</th></tr>' +
    sk.map(x => `<tr><td>${esc(x.name)}</td><td>${x.version}</td>` +
    `<td>${x.auto_activate?'on':''}</td>` +
    `<td class="muted">${esc((x.content||'').slice(0,120))}</td></tr>`).join('') + '</table>';
 },
 async memory(id) {
  const ents = await api(`/memory/entities?room_id=${id}&limit=50`);
  return '<div class="row">' +
    '<input id="memq" placeholder="hybrid search…" ' +
    'onkeydown="if(event.key===\'Enter\')memSearch()">' +
    '<button onclick="memSearch()">search</button></div><div id="memres"></div>' +
    `<div class="row"><input id="memname" placeholder="entity name" style="max-width:140px">` +
    `<input id="memobs" placeholder="observation"><button onclick="memAdd()">remember</button></div>` +
    '<table><tr><th>id</th><th>name</th><th>type</th><th>created</th></tr>' +
    ents.map(e => `<tr><td>${e.id}</td><td>${esc(e.name)}</td><td>${e.type}</td>` +
    `<td class="muted">${e.created_at}</td></tr>`).join('') + '</table>';
 },
 async messages(id) {
  const ms = await api(`/rooms/${id}/messages`);
  const es = await api(`/rooms/${id}/escalations`);
  return '<h2>Escalations</h2>' + (es.map(e =>
    `<div class="ev">${esc(e.question)} <span class="muted">${e.status}</span>` +
    (e.status === 'pending'
      ? `<div class="row" style="margin-top:3px"><input id="ans${e.id}" placeholder="answer…">` +
        `<button class="mini" onclick="answerEsc(${e.id})">answer</button></div>`
      : e.answer ? `<div class="muted">→ ${esc(e.answer)}</div>` : '') + '</div>'
    ).join('') || '<span class="muted">none</span>') +
    '<h2 style="margin-top:10px">Inter-room messages</h2>' + (ms.map(m =>
    `<div class="ev"><b>${esc(m.subject)}</b> <span class="muted">${m.direction} ` +
    `${m.status}</span><br>${esc((m.body||'').slice(0,200))}</div>`).join('')
    || '<span class="muted">none</span>');
 },
 async wallet(id) {
  const w = await api(`/rooms/${id}/wallet`);
  const tx = await api(`/rooms/${id}/wallet/transactions`);
  const sum = await api(`/rooms/${id}/wallet/summary`);
  return `<div class="ev">address: <b>${w.address||'—'}</b></div>` +
    `<div class="ev">income ${sum.totalIncome} · expenses ${sum.totalExpenses}` +
    ` · net ${sum.netProfit} · ${sum.transactionCount} txs</div>` +
    '<h2 style="margin-top:10px">Transactions</h2><table>' +
    '<tr><th>type</th><th>amount</th><th>counterparty</th><th>status</th></tr>' +
    tx.map(t => `<tr><td>${t.type}</td><td>${t.amount}</td>` +
    `<td class="muted">${esc((t.counterparty||'—').slice(0,14))}</td>` +
    `<td>${t.status}</td></tr>`).join('') + '</table>';
 },
 async credentials(id) {
  const cs = await api(`/rooms/${id}/credentials`);
  return '<div class="row">' +
    '<input id="credname" placeholder="name" style="max-width:140px">' +
    '<input id="credval" placeholder="value" type="password">' +
    '<button onclick="credAdd()">add</button></div>' +
    '<table><tr><th>name</th><th>type</th><th>by</th><th>created</th></tr>' +
    cs.map(x => `<tr><td>${esc(x.name)}</td><td>${x.type}</td>` +
    `<td>${x.provided_by}</td><td class="muted">${x.created_at}</td></tr>`
    ).join('') + '</table>' +
    '<div class="muted" style="margin-top:6px">values are AES-256-GCM ' +
    'encrypted at rest; agents read them with room_get_credential</div>';
 },
 async 'room-settings'(id) {
  const room = await api(`/rooms/${id}`);
  const vh = await api(`/rooms/${id}/voter-health`);
  return `<div class="row"><input id="roomgoal" value="${esc(room.goal||'')}" ` +
    `placeholder="objective"><button onclick="roomGoalSave()">save goal</button></div>` +
    `<div class="row"><span class="muted">queen gap (ms)</span>` +
    `<input id="qgap" value="${room.queen_cycle_gap_ms||''}" style="max-width:100px">` +
    `<span class="muted">max turns</span>` +
    `<input id="qturns" value="${room.queen_max_turns||''}" style="max-width:60px">` +
    `<button onclick="roomPaceSave()">save pacing</button></div>` +
    '<h2>Config</h2><pre class="muted" style="font-size:11px">' +
    esc(JSON.stringify(room.config || {}, null, 1)) + '</pre>' +
    '<h2>Voter health</h2><table><tr><th>worker</th><th>cast</th>' +
    '<th>missed</th><th>healthy</th></tr>' +
    vh.map(v => `<tr><td>${esc(v.worker_name)}</td><td>${v.votes_cast}</td>` +
    `<td>${v.votes_missed}</td><td>${v.is_healthy?'yes':'NO'}</td></tr>`
    ).join('') + '</table>' +
    `<div class="row" style="margin-top:8px"><button onclick="roomDelete(${id})" ` +
    `style="color:#f85149">delete room</button></div>`;
 },
 async settings(id) {
  const s = await api('/settings');
  const entries = Array.isArray(s) ? s : Object.entries(s).map(([k,v])=>({key:k,value:v}));
  return '<h2>Global settings</h2>' +
    `<div class="row"><input id="setk" placeholder="key" style="max-width:180px">` +
    `<input id="setv" placeholder="value"><button onclick="settingSave()">set</button></div>` +
    '<table><tr><th>key</th><th>value</th></tr>' +
    entries.map(e => `<tr><td>${esc(e.key)}</td><td class="muted">${esc(String(e.value).slice(0,80))}</td></tr>`).join('') +
    '</table><h2 style="margin-top:10px">Clerk</h2>' +
    `<div class="row"><select id="cmode"><option value="auto">commentary auto</option>` +
    `<option value="light">commentary light</option></select>` +
    `<button onclick="clerkModeSave()">apply</button></div>` +
    '<h2 style="margin-top:10px">Updates</h2>' +
    `<div class="row"><button onclick="checkUpdate()">check for updates</button>` +
    `<span class="muted" id="updres"></span></div>`;
 },
 async status(id) {
  const st2 = await api('/status');
  const lm = await api('/local-model/status');
  const prof = await api('/status/http-profile');
  return '<h2>Server</h2><pre class="muted" style="font-size:11px">' +
    esc(JSON.stringify(st2, null, 1)) + '</pre>' +
    '<h2>Engine</h2><pre class="muted" style="font-size:11px">' +
    esc(JSON.stringify(lm, null, 1)) + '</pre>' +
    '<h2>HTTP profile</h2><pre class="muted" style="font-size:11px">' +
    esc(JSON.stringify(prof, null, 1).slice(0, 3000)) + '</pre>';
 },
 async help(id) {
  return `
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
    (docs/API.md), the MCP stdio server (<code>room-amd mcp</code>, 95 native + 76 reference-name alias
    tools) and webhooks (tasks + queen wake).</div>
    <div class="ev"><b>Keeper contact:</b> verify an email under
    contacts so escalations reach you; or read the outbox at
    ~/.roomamd/outbox.jsonl.</div>`;
 },
};

async function select(id) {
  sel = id; loadRooms(); tabbar();
  const st = await api(`/rooms/${id}/status`);
  $$('roomtitle').textContent = `${st.room.name} — ${st.room.goal || 'no goal'}`;
  $$('roomctl').innerHTML =
    `<button onclick="roomAct(${id},'start')">start</button> ` +
    `<button onclick="roomAct(${id},'pause')">pause</button> ` +
    `<button onclick="roomAct(${id},'restart')">restart</button>` +
    ` <span class="muted">cycles: ${st.token_usage.cycles}</span>`;
  $$('tabbody').innerHTML = await (RENDER[tab] || RENDER.overview)(id, st);
}

// ---- actions
async function goalAdd() {
  const d = $$('goaldesc').value.trim(); if (!d) return;
  await api(`/rooms/${sel}/goals`, {method:'POST', body: JSON.stringify({description:d})});
  select(sel);
}
async function goalDone(id) {
  await api(`/goals/${id}`, {method:'PATCH', body: JSON.stringify({status:'completed'})});
  select(sel);
}
async function goalDrop(id) {
  await api(`/goals/${id}`, {method:'PATCH', body: JSON.stringify({status:'abandoned'})});
  select(sel);
}
async function propose() {
  const p = $$('propin').value.trim(); if (!p) return;
  await api(`/rooms/${sel}/decisions`, {method:'POST',
    body: JSON.stringify({proposal:p, decision_type: $$('proptype').value})});
  select(sel);
}
async function keeperVote(id, v) {
  await api(`/decisions/${id}/keeper-vote`, {method:'POST', body: JSON.stringify({vote:v})});
  select(sel);
}
async function workerAdd() {
  const n = $$('wname').value.trim(); if (!n) return;
  await api(`/rooms/${sel}/workers`, {method:'POST',
    body: JSON.stringify({name:n, role: $$('wrole').value,
                          system_prompt: `You are ${n}, a ${$$('wrole').value}.`})});
  select(sel);
}
async function taskAdd() {
  const n = $$('tname').value.trim(), p = $$('tprompt').value.trim();
  if (!n || !p) return;
  const cron = $$('tcron').value.trim();
  await api('/tasks', {method:'POST', body: JSON.stringify(
    {name:n, prompt:p, room_id: sel, cron_expression: cron || undefined})});
  select(sel);
}
async function skillAdd() {
  const n = $$('skname').value.trim(), c = $$('skcontent').value.trim();
  if (!n || !c) return;
  await api(`/rooms/${sel}/skills`, {method:'POST', body: JSON.stringify(
    {name:n, content:c, activation_context: $$('skctx').value.trim()})});
  select(sel);
}
async function memAdd() {
  const n = $$('memname').value.trim(), o = $$('memobs').value.trim();
  if (!n || !o) return;
  await api('/memory/entities', {method:'POST', body: JSON.stringify(
    {name:n, content:o, room_id: sel})});
  select(sel);
}
async function answerEsc(id) {
  const a = $$('ans'+id).value.trim(); if (!a) return;
  await api(`/escalations/${id}/resolve`, {method:'POST', body: JSON.stringify({answer:a})});
  select(sel);
}
async function credAdd() {
  const name = $$('credname').value.trim(), value = $$('credval').value;
  if (!name || !value) return;
  await api(`/rooms/${sel}/credentials`, {method:'POST',
            body: JSON.stringify({name, value})});
  select(sel);
}
async function roomGoalSave() {
  await api(`/rooms/${sel}`, {method:'PATCH',
    body: JSON.stringify({goal: $$('roomgoal').value})});
  select(sel);
}
async function roomPaceSave() {
  await api(`/rooms/${sel}`, {method:'PATCH', body: JSON.stringify(
    {queen_cycle_gap_ms: parseInt($$('qgap').value) || undefined,
     queen_max_turns: parseInt($$('qturns').value) || undefined})});
  select(sel);
}
async function roomDelete(id) {
  if (!confirm('Delete this room permanently?')) return;
  await api(`/rooms/${id}`, {method:'DELETE'});
  sel = null; loadRooms();
}
async function settingSave() {
  const k = $$('setk').value.trim(); if (!k) return;
  await api(`/settings/${encodeURIComponent(k)}`, {method:'PUT',
    body: JSON.stringify({value: $$('setv').value})});
  select(sel);
}
async function clerkModeSave() {
  await api('/clerk/settings', {method:'PUT',
    body: JSON.stringify({commentary: true, pace: $$('cmode').value})});
  await api('/settings/clerk_commentary_mode', {method:'PUT',
    body: JSON.stringify({value: $$('cmode').value})});
}
async function checkUpdate() {
  const u = await api('/status/check-update', {method:'POST'});
  $$('updres').textContent = u.updateAvailable
    ? `update available: ${u.latestVersion}` : `up to date (${u.state})`;
}
async function memSearch() {
  const qy = $$('memq').value.trim();
  if (!qy) return;
  const hits = await api(`/memory/search?q=${encodeURIComponent(qy)}&room_id=${sel}`);
  $$('memres').innerHTML = hits.map(h =>
    `<div class="ev"><b>${esc(h.name)}</b> <span class="muted">${(h.score||0).toFixed(3)}</span></div>`
    ).join('') || '<span class="muted">no hits</span>';
}
async function roomAct(id, act) { await api(`/rooms/${id}/${act}`, {method:'POST'}); select(id); }
async function createRoom() {
  const name = $$('newroom').value.trim();
  if (!name) return;
  await api('/rooms', {method:'POST', body: JSON.stringify({name})});
  $$('newroom').value = ''; loadRooms();
}
async function createFromTemplate() {
  const ts = await api('/templates');
  const names = ts.map(t => t.id || t.name).join(', ');
  const pick = prompt(`template? (${names})`);
  if (!pick) return;
  await api('/rooms/from-template', {method:'POST',
    body: JSON.stringify({template: pick, name: $$('newroom').value.trim() || pick})});
  loadRooms();
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
  $$('chatlog').innerHTML += `<div class="ev"><b>${who}:</b> ${esc(text)}</div>`;
  $$('chatlog').scrollTop = 1e9;
}
function eventLine(e) {
  $$('events').innerHTML =
    `<div class="ev"><span class="muted">${esc(e.channel)}</span> ${esc(e.type)}` +
    (e.data && e.data.content ? ' — ' + esc(e.data.content) : '') + '</div>' +
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
    setInterval(() => api('/clerk/presence', {method:'POST'}), 45000);
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
