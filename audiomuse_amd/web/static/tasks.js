/* Dashboard: stats, task control, cron (reference: app_dashboard.py,
   app.py task-control routes, app_cron.py). */
"use strict";

const Tasks = {
  timer: null,

  render(el) {
    el.innerHTML = `
      <div class="grid">
        <section><h2>Library</h2><table id="dash-stats"></table></section>
        <section><h2>Start a run</h2>
          <div class="row">
            <button class="primary" onclick="Tasks.start('analysis')">
              Run analysis</button>
            <button onclick="Tasks.start('clustering')">Run clustering
              </button>
            <button onclick="Tasks.rebuild()">Rebuild indexes</button>
          </div>
          <p class="muted">Analysis scans the configured media servers,
            embeds every track on the GPU and rebuilds all nine indexes;
            clustering evolves playlists from the stored vectors.</p>
          <h3>Queue</h3><table id="queue-stats"></table></section>
        <section class="wide"><h2>Active tasks</h2>
          <table id="task-table"><thead><tr><th>task</th><th>type</th>
            <th>status</th><th>progress</th><th></th></tr></thead>
            <tbody></tbody></table></section>
        <section class="wide"><h2>Scheduled (cron)</h2>
          <div class="row">
            <input id="cron-name" placeholder="name" size="12">
            <input id="cron-sched" placeholder="0 3 * * *" size="10">
            <select id="cron-type"><option>rebuild_indexes</option>
              <option>analysis</option><option>clustering</option>
              <option>sonic_fingerprint</option><option>clean_orphans</option>
              <option>multiserver_sync</option>
              <option>chromaprint_backfill</option></select>
            <button onclick="Tasks.addCron()">Add</button></div>
          <table id="cron-table"></table></section>
      </div>`;
    this.refresh(el);
    clearInterval(this.timer);
    this.timer = setInterval(() => {
      if (document.getElementById("view-dashboard").classList.contains("on"))
        this.refresh(el);
    }, 3000);
  },

  async refresh() {
    try {
      const [stats, queue, tasks, cron] = await Promise.all([
        AM.api("/api/dashboard"), AM.api("/api/queue/stats"),
        AM.api("/api/active_tasks"), AM.api("/api/cron")]);
      document.getElementById("dash-stats").innerHTML =
        Object.entries(stats).map(([k, v]) =>
          `<tr><th>${AM.esc(k)}</th><td>${AM.esc(v)}</td></tr>`).join("");
      document.getElementById("queue-stats").innerHTML =
        Object.entries(queue).map(([k, v]) =>
          `<tr><th>${AM.esc(k)}</th><td>${AM.esc(v)}</td></tr>`).join("");
      document.querySelector("#task-table tbody").innerHTML =
        (tasks || []).map(t => `
          <tr><td class="muted">${AM.esc(t.task_id.slice(0, 8))}</td>
          <td>${AM.esc(t.task_type)}</td><td>${AM.esc(t.status)}</td>
          <td><div class="progress"><div style="width:${
            Math.round((t.progress || 0) * 100)}%"></div></div></td>
          <td><button onclick="Tasks.cancel('${AM.esc(t.task_id)}')">
            cancel</button></td></tr>`).join("") ||
        `<tr><td colspan="5" class="muted">queue idle</td></tr>`;
      document.getElementById("cron-table").innerHTML =
        (cron || []).map(c => `
          <tr><td>${AM.esc(c.name)}</td><td>${AM.esc(c.schedule)}</td>
          <td>${AM.esc(c.task_type)}</td>
          <td><button onclick="Tasks.delCron(${c.id})">delete</button>
          </td></tr>`).join("");
    } catch (e) { /* auth overlays handle themselves */ }
  },

  async start(kind) {
    try {
      const out = await AM.api(`/api/${kind}/start`, { json: {} });
      AM.status(`${kind} queued: ${out.task_id.slice(0, 8)}`, "ok");
      this.refresh();
    } catch (e) { AM.status(e.message, "err"); }
  },

  async rebuild() {
    try {
      const out = await AM.api("/api/index/rebuild", { json: {} });
      AM.status(`rebuild queued: ${(out.task_id || "").slice(0, 8)}`, "ok");
    } catch (e) { AM.status(e.message, "err"); }
  },

  async cancel(tid) {
    await AM.api(`/api/task/${tid}/cancel`, { method: "POST" });
    this.refresh();
  },

  async addCron() {
    try {
      await AM.api("/api/cron", { json: {
        name: document.getElementById("cron-name").value,
        schedule: document.getElementById("cron-sched").value || "0 3 * * *",
        task_type: document.getElementById("cron-type").value } });
      this.refresh();
    } catch (e) { AM.status(e.message, "err"); }
  },

  async delCron(id) {
    await AM.api(`/api/cron/${id}`, { method: "DELETE" });
    this.refresh();
  },
};
AM.register("dashboard", el => Tasks.render(el));
AM.views.dashboard.refresh = el => Tasks.refresh(el);
