/* Song alchemy: add/subtract mixing, anchors, radios, sonic fingerprint
   (reference: app_alchemy.py + app_sonic_fingerprint.py pages). */
"use strict";

const Alchemy = {
  add: [], subtract: [],

  render(el) {
    el.innerHTML = `
      <div class="grid">
        <section><h2>Mix ingredients</h2>
          <div class="row"><input id="al-q" size="24"
              placeholder="search tracks to add...">
            <button onclick="Alchemy.search()">Search</button></div>
          <ul class="list" id="al-results" style="max-height:200px"></ul>
          <h3>Adding (<span id="al-nadd">0</span>)</h3>
          <ul class="list" id="al-add" style="max-height:120px"></ul>
          <h3>Subtracting (<span id="al-nsub">0</span>)</h3>
          <ul class="list" id="al-sub" style="max-height:120px"></ul>
          <div class="row">
            <label style="margin:0">temperature</label>
            <input id="al-temp" value="0.15" size="4">
            <label style="margin:0">n</label>
            <input id="al-n" value="25" size="3">
            <button class="primary" onclick="Alchemy.run()">Transmute
              </button>
            <button onclick="Alchemy.clear()">Clear</button></div>
        </section>

        <section><h2>Result</h2>
          <ul class="list" id="al-out" style="max-height:380px"></ul>
          <div class="row">
            <input id="al-anchor" size="14" placeholder="anchor name">
            <button onclick="Alchemy.saveAnchor()">Save result as
              anchor</button></div></section>

        <section><h2>Anchors &amp; radios</h2>
          <ul class="list" id="al-anchors" style="max-height:140px"></ul>
          <div class="row">
            <input id="al-radio" size="14" placeholder="radio name">
            <button onclick="Alchemy.saveRadio()">Save current mix as
              radio</button></div>
          <ul class="list" id="al-radios" style="max-height:140px"></ul>
        </section>

        <section><h2>Sonic fingerprint</h2>
          <p class="muted">Taste vector from your listening history
            (30-day half-life recency weights) expanded to fresh
            recommendations.</p>
          <div class="row">
            <button onclick="Alchemy.fingerprint()">From current mix as
              history</button></div>
          <ul class="list" id="al-fp"></ul></section>
      </div>`;
    AM.bindTrackList(document.getElementById("al-out"),
      id => { location.hash = "#library"; Library.pick(id); });
    this.refreshSide();
  },

  async search() {
    const q = document.getElementById("al-q").value;
    const out = await AM.api(`/api/search_tracks?q=${encodeURIComponent(q)}`);
    const ul = document.getElementById("al-results");
    ul.innerHTML = out.map(t => `
      <li>${AM.esc(t.title)} <span class="by">${AM.esc(t.author)}</span>
        <button data-act="add" data-id="${AM.esc(t.item_id)}">+</button>
        <button data-act="sub" data-id="${AM.esc(t.item_id)}">−</button>
      </li>`).join("");
    ul.onclick = ev => {
      const b = ev.target.closest("button[data-act]");
      if (!b) return;
      (b.dataset.act === "add" ? this.add : this.subtract)
        .push(b.dataset.id);
      this.renderSets();
    };
  },

  renderSets() {
    document.getElementById("al-nadd").textContent = this.add.length;
    document.getElementById("al-nsub").textContent = this.subtract.length;
    document.getElementById("al-add").innerHTML =
      this.add.map(i => `<li>${AM.esc(i)}</li>`).join("");
    document.getElementById("al-sub").innerHTML =
      this.subtract.map(i => `<li>${AM.esc(i)}</li>`).join("");
  },

  clear() { this.add = []; this.subtract = []; this.renderSets(); },

  async run() {
    try {
      const out = await AM.api("/api/alchemy", { json: {
        add: this.add, subtract: this.subtract,
        n: +document.getElementById("al-n").value || 25,
        temperature: +document.getElementById("al-temp").value || 0 } });
      this.last = out;
      document.getElementById("al-out").innerHTML =
        out.map(t => AM.trackLi(t)).join("") ||
        `<li class="muted">empty mix</li>`;
    } catch (e) { AM.status(e.message, "err"); }
  },

  async saveAnchor() {
    const name = document.getElementById("al-anchor").value;
    const ids = (this.last || []).map(t => t.item_id);
    if (!name || !ids.length) {
      AM.status("need a name and a result", "warn"); return;
    }
    await AM.api("/api/alchemy/anchors",
      { json: { name, item_ids: ids } });
    this.refreshSide();
  },

  async saveRadio() {
    const name = document.getElementById("al-radio").value;
    if (!name) { AM.status("need a name", "warn"); return; }
    await AM.api("/api/alchemy/radios", { json: {
      name, definition: { add: this.add, subtract: this.subtract } } });
    this.refreshSide();
  },

  async refreshSide() {
    const anchors = await AM.api("/api/alchemy/anchors").catch(() => []);
    document.getElementById("al-anchors").innerHTML = anchors.map(a => `
      <li>anchor: ${AM.esc(a)}
        <button onclick="Alchemy.add.push('anchor:${AM.esc(a)}');
          Alchemy.renderSets()">+ mix</button>
        <button onclick="Alchemy.delAnchor('${AM.esc(a)}')">delete</button>
      </li>`).join("") || `<li class="muted">no anchors yet</li>`;
    const radios = await AM.api("/api/alchemy/radios").catch(() => []);
    document.getElementById("al-radios").innerHTML = radios.map(r => `
      <li>radio: ${AM.esc(r.name)}
        <button onclick="Alchemy.playRadio('${AM.esc(r.name)}')">play
        </button></li>`).join("") || `<li class="muted">no radios yet</li>`;
  },

  async delAnchor(name) {
    await AM.api(`/api/alchemy/anchors/${encodeURIComponent(name)}`,
      { method: "DELETE" });
    this.refreshSide();
  },

  async playRadio(name) {
    const out = await AM.api(
      `/api/alchemy/radios/${encodeURIComponent(name)}/play`,
      { method: "POST" });
    this.last = out;
    document.getElementById("al-out").innerHTML =
      out.map(t => AM.trackLi(t)).join("");
  },

  async fingerprint() {
    const ids = this.add.filter(i => !i.startsWith("anchor:"));
    if (!ids.length) { AM.status("add some tracks first", "warn"); return; }
    const qs = ids.map(i => `item_id=${encodeURIComponent(i)}`).join("&");
    const out = await AM.api(`/api/sonic_fingerprint?${qs}&n=20`);
    document.getElementById("al-fp").innerHTML =
      out.map(t => AM.trackLi(t)).join("");
  },
};
AM.register("alchemy", el => Alchemy.render(el));
