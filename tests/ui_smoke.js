/* UI smoke harness: loads the SPA modules in a stub DOM and drives every
   view's data path against a live server (node 12, no deps). Run by
   tests/test_ui.py; any rejection exits non-zero. */
"use strict";
const fs = require("fs");
const http = require("http");
const path = require("path");
const vm = require("vm");

const BASE = process.argv[2] || "http://127.0.0.1:5591";

// ---- fetch shim over node http ----
function fetchShim(url, opts) {
  opts = opts || {};
  return new Promise((resolve, reject) => {
    const u = new URL(url.startsWith("http") ? url : BASE + url);
    const req = http.request(u, { method: opts.method || "GET",
                                  headers: opts.headers || {} }, res => {
      let body = "";
      res.on("data", c => { body += c; });
      res.on("end", () => resolve({
        ok: res.statusCode >= 200 && res.statusCode < 300,
        status: res.statusCode,
        statusText: String(res.statusCode),
        headers: { get: k => res.headers[k.toLowerCase()] || "" },
        json: () => Promise.resolve(JSON.parse(body || "null")),
        text: () => Promise.resolve(body),
      }));
    });
    req.on("error", reject);
    if (opts.body) req.write(opts.body);
    req.end();
  });
}

// ---- minimal DOM ----
function makeEl(id) {
  const el = {
    id, innerHTML: "", value: "", textContent: "", className: "",
    checked: false, files: [], dataset: {}, style: {},
    classList: { add() {}, remove() {}, toggle() {},
                 contains() { return true; } },
    addEventListener() {}, insertAdjacentHTML(_, h) { el.innerHTML += h; },
    closest() { return null; }, querySelectorAll() { return []; },
    getBoundingClientRect() {
      return { left: 0, top: 0, width: 800, height: 560 };
    },
    getContext() { return null; },   // map falls back gracefully
    clientWidth: 800, clientHeight: 560, scrollTop: 0, scrollHeight: 0,
    onclick: null, onkeyup: null, firstChild: { textContent: "" },
  };
  return el;
}
const els = {};
const document = {
  getElementById: id => els[id] || (els[id] = makeEl(id)),
  querySelectorAll: () => [],
  querySelector: () => makeEl("q"),
};
const sandbox = {
  console, document, fetch: fetchShim, URL,
  window: { addEventListener() {}, devicePixelRatio: 1 },
  location: { hash: "#dashboard" },
  setTimeout: (fn, ms) => undefined,   // no timers: keep the run finite
  setInterval: () => 0, clearInterval() {},
  Math, JSON, Object, Array, Promise, Map, Set, Float32Array,
  encodeURIComponent, TextDecoder: function () {
    return { decode: b => String(b) };
  },
};
sandbox.global = sandbox;
vm.createContext(sandbox);

const STATIC = path.join(__dirname, "..", "audiomuse_amd", "web", "static");
for (const f of ["app.js", "setup.js", "tasks.js", "library.js", "map.js",
                 "explorer.js", "alchemy.js", "chat.js", "admin.js"]) {
  vm.runInContext(fs.readFileSync(path.join(STATIC, f), "utf8"), sandbox,
                  { filename: f });
}

async function main() {
  // top-level const/let live in the context's lexical scope, not on the
  // sandbox object — read them back through the context
  const g = n => vm.runInContext(n, sandbox);
  const E = id => document.getElementById(id);
  const AM = g("AM"), Tasks = g("Tasks"), Library = g("Library"),
        MusicMap = g("MusicMap"), Explorer = g("Explorer"),
        Alchemy = g("Alchemy"), Admin = g("Admin");

  // every nav view registered
  for (const v of ["dashboard", "library", "map", "explorer", "alchemy",
                   "chat", "admin"]) {
    if (!AM.views[v]) throw new Error(`view ${v} not registered`);
  }

  // first boot: drive the setup wizard exactly as the UI would
  const st = await AM.api("/api/setup/status");
  if (st.setup_needed) {
    const Setup = g("Setup");
    Setup.showWizard();
    E("sw-user").value = "admin";
    E("sw-pass").value = "adminpass123";
    await Setup.createAdmin();
    E("sw-type").value = "synthetic";
    await Setup.testServer();
    if (!String(E("sw-err2").textContent).includes("reachable"))
      throw new Error("wizard probe failed: " + E("sw-err2").textContent);
    await Setup.saveServer();
    await Setup.saveConfig();
    const st2 = await AM.api("/api/setup/status");
    if (st2.setup_needed) throw new Error("wizard did not complete");
  }

  // render every view (binds DOM + fires initial loads)
  for (const v of Object.keys(AM.views)) AM.views[v].render(makeEl("x"));

  // dashboard data path
  E("cron-sched").value = "0 3 * * *";
  await Tasks.refresh();
  if (!E("dash-stats").innerHTML.includes("tracks"))
    throw new Error("dashboard stats empty");

  // library: search -> pick -> similar
  E("lib-q").value = "Song";
  await Library.search();
  if (!E("lib-results").innerHTML.includes("data-id"))
    throw new Error("search returned nothing");
  const id = /data-id="([^"]+)"/.exec(E("lib-results").innerHTML)[1];
  E("lib-n").value = "5";
  AM.selected = { item_id: id };
  await Library.similar();
  if (!E("lib-similar").innerHTML.includes("data-id"))
    throw new Error("similar returned nothing");
  AM.previous = AM.selected;
  const id2 = /data-id="([^"]+)"/.exec(E("lib-similar").innerHTML)[1];
  AM.selected = { item_id: id2 };
  await Library.path();
  await Library.axes().catch(() => {});      // lyrics index optional
  await Library.hyper();

  // map data path (no WebGL in the stub; load still fetches + grids)
  E("map-kind").value = "song";
  E("map-pct").value = "100";
  await MusicMap.load();
  if (!MusicMap.points.length) throw new Error("map empty");
  if (!MusicMap.nearest(MusicMap.points[0].x, MusicMap.points[0].y))
    throw new Error("hover grid broken");

  // explorer: root -> mood folder -> leaf (lazy warm)
  await Explorer.open("root", "Explorer");
  const m = /data-node="([^"]+)"/.exec(E("hx-list").innerHTML);
  if (!m) throw new Error("explorer root has no folders");
  await Explorer.open(m[1], "mood");
  const leaf = /data-node="([^"]+)"/.exec(E("hx-list").innerHTML);
  if (leaf) await Explorer.open(leaf[1], "leaf");
  if (!E("hx-list").innerHTML.includes("data-id"))
    throw new Error("explorer leaf has no tracks");

  // alchemy: two adds -> transmute
  Alchemy.add = [id, id2];
  E("al-n").value = "5";
  E("al-temp").value = "0";
  await Alchemy.run();
  if (!E("al-out").innerHTML.includes("data-id"))
    throw new Error("alchemy returned nothing");
  await Alchemy.refreshSide();

  // chat (non-streaming endpoint exercises the same planner)
  const chat = await AM.api("/chat/api/chatPlaylist",
    { json: { prompt: "10 rock songs" } });
  if (!chat.tracks || !chat.tracks.length)
    throw new Error("chat returned no tracks");

  // admin data paths
  await Admin.refresh();
  await Admin.loadConfig();
  if (!E("ad-config").innerHTML.includes("IVF_NPROBE"))
    throw new Error("config editor empty");

  // migration wizard probe + preview against a synthetic target
  E("mg-type").value = "synthetic";
  E("mg-src").value = "default";
  await Admin.mgProbe();
  if (!E("mg-out").textContent.includes("reachable"))
    throw new Error("migration probe failed");

  // catalogue browser paging
  await Admin.browse(0);
  if (!E("ad-browse").innerHTML.includes("Song"))
    throw new Error("catalogue browser empty");

  // album-by-album review session: open -> dry run -> album table
  await Admin.msStart();
  if (!E("ms-info").textContent.includes("target tracks"))
    throw new Error("migration session did not open");
  await Admin.msDryRun();
  if (!E("ms-albums").innerHTML.includes("decision") &&
      !E("ms-albums").innerHTML.includes("auto"))
    throw new Error("matched-albums table empty");

  console.log("UI_SMOKE_OK views=" + Object.keys(AM.views).length);
}

main().then(() => process.exit(0),
            err => { console.error("UI_SMOKE_FAIL", err); process.exit(1); });
