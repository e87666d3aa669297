/* Admin: media servers, backup/restore, plugins, migration wizard UI,
   config editor, users (reference: app_music_servers.py, app_backup.py,
   plugin/blueprint.py, app_provider_migration.py, setup manager). */
"use strict";

const Admin = {
  render(el) {
    el.innerHTML = `
      <div class="grid">
        <section><h2>Media servers</h2>
          <table id="ad-servers"></table>
          <h3>Add / update</h3>
          <div class="row">
            <input id="sv-id" size="10" placeholder="server id">
            <select id="sv-type"><option>navidrome</option>
              <option>jellyfin</option><option>emby</option>
              <option>lyrion</option><option>plex</option>
              <option>synthetic</option></select></div>
          <div class="row">
            <input id="sv-url" size="24" placeholder="base url">
            <input id="sv-user" size="10" placeholder="username">
            <input id="sv-cred" size="12" placeholder="token/password"
              type="password">
            <button class="primary" onclick="Admin.saveServer()">Save
              </button></div></section>

        <section><h2>Backup / restore</h2>
          <div class="row">
            <a href="/api/backup"><button>Download backup</button></a>
            <input type="file" id="ad-restore-file">
            <button onclick="Admin.restore()">Restore</button></div>
          <p class="muted">Backups are portable logical snapshots (SQLite
            artifact) on both storage backends; restore refuses while
            tasks run.</p>
          <h3>Users</h3><ul class="list" id="ad-users"></ul>
          <h3>Catalogue browser</h3>
          <div class="row"><input id="br-q" size="14" placeholder="filter">
            <button onclick="Admin.browse(0)">Browse</button>
            <button onclick="Admin.browse(Admin.brOffset +
              Admin.brLimit)">Next page</button>
            <span id="br-info" class="muted"></span></div>
          <table id="ad-browse"></table></section>

        <section class="wide"><h2>Provider migration wizard</h2>
          <div class="steps"><span class="on">Target</span><span>Probe
            </span><span>Preview</span><span>Apply</span></div>
          <div class="row">
            <select id="mg-type"><option>synthetic</option>
              <option>navidrome</option><option>jellyfin</option>
              <option>emby</option><option>lyrion</option>
              <option>plex</option></select>
            <input id="mg-url" size="22" placeholder="target base url">
            <input id="mg-cred" size="12" placeholder="token/password"
              type="password">
            <input id="mg-src" size="10" value="default"
              title="source server id">
            <input id="mg-dst" size="10" value="migrated"
              title="target server id">
          </div>
          <div class="row">
            <button onclick="Admin.mgProbe()">1 · Probe</button>
            <button onclick="Admin.mgPreview()">2 · Preview matches
              </button>
            <button class="primary" id="mg-apply" disabled
              onclick="Admin.mgApply()">3 · Apply migration</button>
            <label style="margin:0"><input type="checkbox"
              id="mg-remove"> remove source mappings</label></div>
          <pre id="mg-out"></pre>
          <h3>Album-by-album review (session)</h3>
          <div class="row">
            <button onclick="Admin.msStart()">Open session</button>
            <button onclick="Admin.msDryRun()">Dry run</button>
            <button class="primary" id="ms-exec" disabled
              onclick="Admin.msExecute()">Finalize + execute</button>
            <span id="ms-info" class="muted"></span></div>
          <table id="ms-albums"></table></section>

        <section class="wide"><h2>Plugins</h2>
          <table id="ad-plugins"></table>
          <div class="row">
            <input id="pl-name2" size="12" placeholder="plugin name">
            <input type="file" id="pl-file">
            <button onclick="Admin.uploadPlugin()">Upload zip</button>
          </div></section>

        <section class="wide"><h2>Settings
          <span class="muted">(persisted to app_config; workers hydrate
          per job)</span></h2>
          <div class="row"><input id="cfg-filter" size="18"
            placeholder="filter settings...">
            <button onclick="Admin.loadConfig()">Reload</button></div>
          <table id="ad-config"></table></section>
      </div>`;
    this.refresh();
    this.loadConfig();
  },

  async refresh() {
    const servers = await AM.api("/api/servers").catch(() => []);
    document.getElementById("ad-servers").innerHTML =
      `<tr><th>id</th><th>type</th><th>url</th><th></th></tr>` +
      servers.map(s => `
        <tr><td>${AM.esc(s.server_id)}</td><td>${AM.esc(s.server_type)}
        </td><td class="muted">${AM.esc(s.base_url || "")}</td>
        <td><button onclick="Admin.delServer('${AM.esc(s.server_id)}')">
          delete</button></td></tr>`).join("");
    const users = await AM.api("/api/users").catch(() => []);
    document.getElementById("ad-users").innerHTML =
      users.map(u => `<li>${AM.esc(u.username)}
        <span class="tag">${AM.esc(u.role)}</span></li>`).join("");
    const plugins = await AM.api("/api/plugins").catch(() => []);
    document.getElementById("ad-plugins").innerHTML =
      `<tr><th>name</th><th>enabled</th><th></th></tr>` +
      (plugins.map(p => `
        <tr><td>${AM.esc(p.name)}</td><td>${p.enabled ? "yes" : "no"}</td>
        <td><button onclick="Admin.delPlugin('${AM.esc(p.name)}')">
          delete</button></td></tr>`).join("") ||
       `<tr><td colspan="3" class="muted">none installed</td></tr>`);
  },

  async saveServer() {
    try {
      await AM.api("/api/servers", { json: {
        server_id: document.getElementById("sv-id").value || "default",
        server_type: document.getElementById("sv-type").value,
        base_url: document.getElementById("sv-url").value,
        username: document.getElementById("sv-user").value,
        credential: document.getElementById("sv-cred").value } });
      this.refresh();
    } catch (e) { AM.status(e.message, "err"); }
  },

  async delServer(id) {
    await AM.api(`/api/servers/${encodeURIComponent(id)}`,
      { method: "DELETE" });
    this.refresh();
  },

  async restore() {
    const f = document.getElementById("ad-restore-file").files[0];
    if (!f) { AM.status("choose a backup file", "warn"); return; }
    const r = await fetch("/api/restore", { method: "POST", body: f });
    const body = await r.json();
    AM.status(r.ok ? "restored" : body.error, r.ok ? "ok" : "err");
  },

  mgConfig() {
    return { server_type: document.getElementById("mg-type").value,
             server_config: {
               base_url: document.getElementById("mg-url").value,
               credential: document.getElementById("mg-cred").value } };
  },

  async mgProbe() {
    const out = await AM.api("/api/migration/probe",
      { json: this.mgConfig() }).catch(e => ({ error: e.message }));
    document.getElementById("mg-out").textContent =
      JSON.stringify(out, null, 2);
  },

  async mgPreview() {
    const body = Object.assign(this.mgConfig(),
      { source_server_id: document.getElementById("mg-src").value });
    const out = await AM.api("/api/migration/preview", { json: body })
      .catch(e => ({ error: e.message }));
    document.getElementById("mg-out").textContent =
      JSON.stringify(out, null, 2);
    document.getElementById("mg-apply").disabled =
      !(out.match_ratio >= 0.5);
    if (out.match_ratio !== undefined && out.match_ratio < 0.5)
      AM.status("match ratio below 0.5 — apply locked", "warn");
  },

  async mgApply() {
    const body = Object.assign(this.mgConfig(), {
      source_server_id: document.getElementById("mg-src").value,
      target_server_id: document.getElementById("mg-dst").value,
      apply: true,
      remove_source: document.getElementById("mg-remove").checked });
    const out = await AM.api("/api/migration/start", { json: body });
    AM.status(`migration queued: ${out.task_id.slice(0, 8)}`, "ok");
    const poll = async () => {
      const st = await AM.api(`/api/migration/status/${out.task_id}`);
      document.getElementById("mg-out").textContent =
        JSON.stringify(st, null, 2);
      if (st.status === "PENDING" || st.status === "RUNNING")
        setTimeout(poll, 1500);
    };
    poll();
  },

  msId: 0,

  async msStart() {
    const out = await AM.api("/api/migration/session/start",
      { json: Object.assign(this.mgConfig(), {
          source_server_id: document.getElementById("mg-src").value }) })
      .catch(e => ({ error: e.message }));
    if (out.error) { AM.status(out.error, "err"); return; }
    this.msId = out.session_id;
    document.getElementById("ms-info").textContent =
      `session ${out.session_id}: ${out.target_tracks} target tracks`;
  },

  async msDryRun() {
    if (!this.msId) { AM.status("open a session first", "warn"); return; }
    const rep = await AM.api("/api/migration/dry-run",
      { json: { session_id: this.msId } });
    document.getElementById("ms-info").textContent =
      `matched ${rep.matched}/${rep.total} ` +
      `(${Math.round(rep.match_ratio * 100)}%)`;
    document.getElementById("ms-exec").disabled = !(rep.match_ratio >= 0.5);
    const albums = await AM.api(
      `/api/migration/matched-albums/${this.msId}`);
    document.getElementById("ms-albums").innerHTML =
      `<tr><th>album</th><th>matched</th><th>decision</th><th></th></tr>` +
      albums.map(a => `
        <tr><td>${AM.esc(a.album)}</td>
        <td>${a.matched}/${a.total}${a.complete ? "" : " ⚠"}</td>
        <td>${AM.esc(a.decision)}</td>
        <td><button onclick="Admin.msSkip('${AM.esc(a.album)}',
          ${a.decision === "skip"})">
          ${a.decision === "skip" ? "unskip" : "skip"}</button>
        </td></tr>`).join("");
  },

  async msSkip(album, undo) {
    await AM.api("/api/migration/skip-album",
      { json: { session_id: this.msId, album, undo } });
    this.msDryRun();
  },

  async msExecute() {
    const fin = await AM.api("/api/migration/finalize-dry-run",
      { json: { session_id: this.msId } }).catch(e => ({ error: e.message }));
    if (fin.error) { AM.status(fin.error, "err"); return; }
    const out = await AM.api("/api/migration/execute",
      { json: { session_id: this.msId,
                target_server_id: document.getElementById("mg-dst").value,
                remove_source:
                  document.getElementById("mg-remove").checked } });
    AM.status(out.applied ? `migrated ${out.written} mappings`
                          : out.reason, out.applied ? "ok" : "err");
  },

  async uploadPlugin() {
    const f = document.getElementById("pl-file").files[0];
    const name = document.getElementById("pl-name2").value;
    if (!f || !name) { AM.status("name + zip required", "warn"); return; }
    const r = await fetch(`/api/plugins?name=${encodeURIComponent(name)}`,
      { method: "POST", body: f });
    const body = await r.json();
    AM.status(r.ok ? "plugin loaded" : body.error, r.ok ? "ok" : "err");
    this.refresh();
  },

  async delPlugin(name) {
    await AM.api(`/api/plugins/${encodeURIComponent(name)}`,
      { method: "DELETE" });
    this.refresh();
  },

  brOffset: 0, brLimit: 0,

  async browse(offset) {
    const q = document.getElementById("br-q").value;
    const out = await AM.api(`/api/dashboard/browse?offset=${offset}` +
      (q ? `&q=${encodeURIComponent(q)}` : ""));
    this.brOffset = out.offset; this.brLimit = out.limit;
    document.getElementById("br-info").textContent =
      `${out.offset}-${out.offset + out.rows.length} of ${out.total}`;
    document.getElementById("ad-browse").innerHTML =
      `<tr><th>title</th><th>artist</th><th>album</th><th>bpm</th></tr>` +
      out.rows.map(r => `<tr><td>${AM.esc(r.title)}</td>
        <td>${AM.esc(r.author)}</td><td>${AM.esc(r.album)}</td>
        <td>${Math.round(r.tempo || 0)}</td></tr>`).join("");
  },

  async loadConfig() {
    const out = await AM.api("/api/config").catch(() => null);
    if (!out) return;
    const filter = (document.getElementById("cfg-filter").value || "")
      .toUpperCase();
    const rows = Object.entries(out.config)
      .filter(([k]) => !filter || k.includes(filter))
      .sort((a, b) => a[0].localeCompare(b[0]));
    document.getElementById("ad-config").innerHTML = rows.map(([k, v]) => `
      <tr><th>${AM.esc(k)}${out.overrides[k] !== undefined
        ? ' <span class="tag">db</span>' : ""}</th>
      <td><input data-key="${AM.esc(k)}" value="${AM.esc(v)}" size="22"
        onchange="Admin.setConfig(this)"></td></tr>`).join("");
    document.getElementById("cfg-filter").onkeyup = () => this.loadConfig();
  },

  async setConfig(input) {
    try {
      await AM.api("/api/config",
        { json: { [input.dataset.key]: input.value } });
      AM.status(`${input.dataset.key} saved`, "ok");
    } catch (e) { AM.status(e.message, "err"); }
  },
};
AM.register("admin", el => Admin.render(el));
