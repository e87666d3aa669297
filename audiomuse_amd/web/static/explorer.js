/* Hyperbolic explorer: tree browse over the skeleton/lazy-warm cache
   (reference: app_hyperbolic.py pages over hyperbolic_manager tree). */
"use strict";

const Explorer = {
  trail: [],      // breadcrumb of {id, name}

  render(el) {
    el.innerHTML = `
      <section class="wide"><h2>Hyperbolic explorer</h2>
        <div class="row"><span id="hx-status" class="muted"></span>
          <button onclick="Explorer.build()">Rebuild tree</button></div>
        <div class="crumbs" id="hx-crumbs"></div>
        <ul class="list" id="hx-list" style="max-height:460px"></ul>
        <ul class="list" id="hx-similar"></ul>
      </section>`;
    AM.bindTrackList(document.getElementById("hx-similar"),
      id => { location.hash = "#library"; Library.pick(id); });
    this.open("root", "Explorer");
    this.poll();
  },

  async poll() {
    try {
      const st = await AM.api("/api/hyperbolic_tree/status");
      document.getElementById("hx-status").textContent =
        `${st.track_count} tracks · skeleton ${st.skeleton_loaded ? "resident" : "missing"}` +
        (st.full_loaded
          ? ` · full tree warm (${st.warm_seconds_left}s left)`
          : " · full tree cold (leaf click warms it)");
    } catch (e) { /* not built */ }
  },

  crumbs() {
    document.getElementById("hx-crumbs").innerHTML = this.trail
      .map((t, i) => `<a onclick="Explorer.jump(${i})">${AM.esc(t.name)}</a>`)
      .join(" › ");
  },

  jump(i) {
    const t = this.trail[i];
    this.trail = this.trail.slice(0, i);
    this.open(t.id, t.name);
  },

  async open(nodeId, name) {
    let node;
    try {
      node = await AM.api(
        `/api/hyperbolic_tree/node/${encodeURIComponent(nodeId)}`);
    } catch (e) {
      document.getElementById("hx-list").innerHTML =
        `<li class="muted">${AM.esc(e.message)} — run analysis or press
         Rebuild tree</li>`;
      return;
    }
    this.trail.push({ id: nodeId, name });
    this.crumbs();
    const ul = document.getElementById("hx-list");
    if (node.leaf) {
      ul.innerHTML = node.items.map(t =>
        `<li data-id="${AM.esc(t.item_id)}">${AM.esc(t.title || t.item_id)}
         <span class="tag">r=${t.radius}</span>
         <div class="by">${AM.esc(t.author || "")}</div></li>`).join("");
      ul.onclick = async ev => {
        const li = ev.target.closest("li[data-id]");
        if (!li) return;
        const out = await AM.api(`/api/hyperbolic_similar?item_id=${
          encodeURIComponent(li.dataset.id)}&n=15`);
        document.getElementById("hx-similar").innerHTML =
          `<li class="muted">hyperbolic neighbors of ${
            AM.esc(li.dataset.id)}:</li>` +
          out.map(t => AM.trackLi(t)).join("");
      };
    } else {
      ul.innerHTML = (node.items || []).map(c =>
        `<li data-node="${AM.esc(c.id)}" data-name="${AM.esc(c.name)}">
         ${AM.esc(c.name)} <span class="tag">${c.track_count} tracks</span>
         </li>`).join("");
      ul.onclick = ev => {
        const li = ev.target.closest("li[data-node]");
        if (li) this.open(li.dataset.node, li.dataset.name);
      };
    }
    this.poll();
  },

  async build() {
    AM.status("building hyperbolic tree...", "muted");
    try {
      const out = await AM.api("/api/hyperbolic_tree/build",
        { method: "POST" });
      AM.status(`tree built: ${out.tracks} tracks`, "ok");
      this.trail = [];
      this.open("root", "Explorer");
    } catch (e) { AM.status(e.message, "err"); }
  },
};
AM.register("explorer", el => Explorer.render(el));
