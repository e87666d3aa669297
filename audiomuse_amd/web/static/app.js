/* AudioMuse-AMD SPA core: hash router, API helper, auth/setup barrier.
   (reference L7 analog: app.py blueprint pages + static/script.js) */
"use strict";

const AM = {
  views: {},           // name -> {el, render(el), shown?}
  selected: null,      // last selected track {item_id, title, author}
  previous: null,

  // ---- API helper --------------------------------------------------------
  async api(path, opts = {}) {
    const o = Object.assign({ headers: {} }, opts);
    if (o.json !== undefined) {
      o.method = o.method || "POST";
      o.headers["Content-Type"] = "application/json";
      o.body = JSON.stringify(o.json);
      delete o.json;
    }
    const r = await fetch(path, o);
    if (r.status === 401) { Setup.showLogin(); throw new Error("auth"); }
    if (r.status === 403) {
      const body = await r.json().catch(() => ({}));
      if (body.setup) { Setup.showWizard(); throw new Error("setup"); }
      throw new Error(body.error || "forbidden");
    }
    const ct = r.headers.get("Content-Type") || "";
    const body = ct.includes("json") ? await r.json() : await r.text();
    if (!r.ok) throw new Error((body && body.error) || r.statusText);
    return body;
  },

  status(msg, cls) {
    const el = document.getElementById("status");
    el.textContent = msg || "";
    el.className = cls || "";
    if (msg) setTimeout(() => { if (el.textContent === msg) el.textContent = ""; }, 6000);
  },

  esc(s) {
    return String(s == null ? "" : s).replace(/[&<>"]/g,
      c => ({ "&": "&amp;", "<": "&lt;", ">": "&gt;", '"': "&quot;" }[c]));
  },

  trackLi(t, extra) {
    const d = t.distance !== undefined
      ? `<span class="tag">${(+t.distance).toFixed(3)}</span>` : "";
    return `<li data-id="${this.esc(t.item_id)}">` +
      `${this.esc(t.title || t.item_id)} ${d}` +
      `<div class="by">${this.esc(t.author || "")}${extra || ""}</div></li>`;
  },

  bindTrackList(ul, onpick) {
    ul.addEventListener("click", ev => {
      const li = ev.target.closest("li[data-id]");
      if (!li) return;
      const id = li.dataset.id;
      AM.previous = AM.selected;
      AM.selected = { item_id: id, title: li.firstChild.textContent.trim() };
      (onpick || Library.showSimilar)(id);
    });
  },

  // ---- router ------------------------------------------------------------
  register(name, render) { this.views[name] = { render, shown: false }; },

  show(name) {
    if (!this.views[name]) name = "dashboard";
    document.querySelectorAll("nav a").forEach(a =>
      a.classList.toggle("on", a.hash === "#" + name));
    document.querySelectorAll(".view").forEach(v =>
      v.classList.toggle("on", v.id === "view-" + name));
    const v = this.views[name];
    const el = document.getElementById("view-" + name);
    if (!v.shown) { v.shown = true; v.render(el); }
    else if (v.refresh) v.refresh(el);
  },

  overlay(html) {
    document.getElementById("overlay-box").innerHTML = html;
    document.getElementById("overlay").classList.add("on");
  },
  closeOverlay() {
    document.getElementById("overlay").classList.remove("on");
  },

  async boot() {
    window.addEventListener("hashchange",
      () => this.show(location.hash.slice(1) || "dashboard"));
    try {
      const st = await this.api("/api/setup/status");
      if (st.setup_needed) { Setup.showWizard(); return; }
      await this.api("/api/me").catch(() => {});
    } catch (e) { return; /* wizard/login already shown */ }
    this.show(location.hash.slice(1) || "dashboard");
  },
};
