/* Library: search, similar songs, song path, text/lyrics search, sonic
   fingerprint, playlist creation (reference: app_ivf.py, app_clap_
   search.py, app_lyrics.py, app_path.py, app_sonic_fingerprint.py). */
"use strict";

const Library = {
  lastResults: [],

  render(el) {
    el.innerHTML = `
      <div class="grid">
        <section><h2>Search tracks</h2>
          <div class="row"><input id="lib-q" placeholder="title or artist"
              size="28"><button class="primary" onclick="Library.search()">
              Search</button></div>
          <ul class="list" id="lib-results"></ul></section>

        <section><h2>Selected track</h2>
          <div id="lib-sel" class="muted">pick a track on the left</div>
          <div class="row" style="margin-top:8px">
            <label style="margin:0">n</label>
            <input id="lib-n" value="20" size="3">
            <label style="margin:0"><input type="checkbox" id="lib-radius">
              radius walk</label>
            <label style="margin:0">mood</label>
            <input id="lib-mood" size="8" placeholder="(any)">
          </div>
          <div class="row">
            <button onclick="Library.similar()">Similar songs</button>
            <button onclick="Library.path()">Path from previous</button>
            <button onclick="Library.semgrove()">SemGrove</button>
            <button onclick="Library.axes()">Thematic profile</button>
            <button onclick="Library.hyper()">Hyperbolic</button>
          </div>
          <ul class="list" id="lib-similar"></ul>
          <div class="row">
            <input id="pl-name" placeholder="playlist name" size="16">
            <button onclick="Library.makePlaylist()">Create playlist from
              results</button></div></section>

        <section><h2>CLAP text search</h2>
          <div class="row"><input id="clap-q" size="28"
              placeholder="dreamy synth sunset...">
            <button onclick="Library.clap()">Search</button>
            <button onclick="Library.clapSuggest()">Suggest</button></div>
          <div id="clap-sug" class="muted"></div>
          <ul class="list" id="clap-results"></ul></section>

        <section><h2>Lyrics search</h2>
          <div class="row"><input id="lyr-q" size="22"
              placeholder="semantic text...">
            <button onclick="Library.lyrics()">Search</button>
            <select id="axis-q"></select>
            <button onclick="Library.axis()">Top of axis</button></div>
          <ul class="list" id="lyr-results"></ul></section>

        <section class="wide"><h2>Artist similarity</h2>
          <div class="row"><input id="art-q" size="22"
              placeholder="artist name">
            <button onclick="Library.artists()">Similar artists</button></div>
          <ul class="list" id="art-results"></ul></section>
      </div>`;
    AM.bindTrackList(document.getElementById("lib-results"),
      id => Library.pick(id));
    AM.bindTrackList(document.getElementById("lib-similar"),
      id => Library.pick(id));
    AM.bindTrackList(document.getElementById("clap-results"),
      id => Library.pick(id));
    AM.bindTrackList(document.getElementById("lyr-results"),
      id => Library.pick(id));
    this.loadAxes();
  },

  async loadAxes() {
    const sel = document.getElementById("axis-q");
    try {
      const cfg = await AM.api("/api/config");
      const axes = (cfg.config && cfg.config.LYRICS_AXES) || [];
      sel.innerHTML = ["love", "party", "sadness", "hope", "night"]
        .concat(axes).filter((v, i, a) => a.indexOf(v) === i)
        .map(a => `<option>${AM.esc(a)}</option>`).join("");
    } catch (e) {
      sel.innerHTML = "<option>love</option><option>party</option>";
    }
  },

  fill(ulId, items) {
    this.lastResults = items || [];
    document.getElementById(ulId).innerHTML =
      (items || []).map(t => AM.trackLi(t)).join("") ||
      `<li class="muted">no results</li>`;
  },

  async search() {
    const q = document.getElementById("lib-q").value;
    const out = await AM.api(`/api/search_tracks?q=${encodeURIComponent(q)}`);
    this.fill("lib-results", out);
  },

  pick(id) {
    AM.previous = AM.selected;
    AM.selected = { item_id: id };
    document.getElementById("lib-sel").innerHTML =
      `<b>${AM.esc(id)}</b>` + (AM.previous
        ? `<div class="by">previous: ${AM.esc(AM.previous.item_id)}</div>` : "");
    this.similar();
  },

  showSimilar(id) { Library.pick(id); },

  async similar() {
    if (!AM.selected) return;
    const n = document.getElementById("lib-n").value || 20;
    const radius = document.getElementById("lib-radius").checked ? 1 : 0;
    const mood = document.getElementById("lib-mood").value;
    const out = await AM.api(`/api/similar_tracks?item_id=${
      encodeURIComponent(AM.selected.item_id)}&n=${n}&radius_similarity=${radius}` +
      (mood ? `&mood_filter=${encodeURIComponent(mood)}` : ""));
    this.fill("lib-similar", out);
  },

  async path() {
    if (!AM.selected || !AM.previous) {
      AM.status("pick two tracks first", "warn"); return;
    }
    const out = await AM.api(`/api/path?start=${
      encodeURIComponent(AM.previous.item_id)}&end=${
      encodeURIComponent(AM.selected.item_id)}`);
    this.fill("lib-similar", out);
  },

  async semgrove() {
    if (!AM.selected) return;
    const out = await AM.api(`/api/semgrove?item_id=${
      encodeURIComponent(AM.selected.item_id)}&n=20`);
    this.fill("lib-similar", out);
  },

  async axes() {
    if (!AM.selected) return;
    const out = await AM.api(`/api/lyrics_axes_similar?item_id=${
      encodeURIComponent(AM.selected.item_id)}&n=20`);
    this.fill("lib-similar", out);
  },

  async hyper() {
    if (!AM.selected) return;
    const out = await AM.api(`/api/hyperbolic_similar?item_id=${
      encodeURIComponent(AM.selected.item_id)}&n=20`);
    this.fill("lib-similar", out);
  },

  async clap() {
    const q = document.getElementById("clap-q").value;
    try {
      const out = await AM.api(`/api/clap_search?q=${encodeURIComponent(q)}`);
      this.fill("clap-results", out);
    } catch (e) { AM.status(e.message, "warn"); }
  },

  async clapSuggest() {
    const out = await AM.api("/api/clap_search/suggestions");
    document.getElementById("clap-sug").textContent =
      (out || []).slice(0, 6).join(" · ");
  },

  async lyrics() {
    const q = document.getElementById("lyr-q").value;
    const out = await AM.api(`/api/lyrics_search?q=${encodeURIComponent(q)}`);
    this.fill("lyr-results", out);
  },

  async axis() {
    const a = document.getElementById("axis-q").value;
    const out = await AM.api(`/api/lyrics_axes?axis=${
      encodeURIComponent(a)}&n=20`);
    this.fill("lyr-results", out);
  },

  async artists() {
    const q = document.getElementById("art-q").value;
    const out = await AM.api(`/api/artist_similarity?artist=${
      encodeURIComponent(q)}`);
    document.getElementById("art-results").innerHTML =
      (out || []).map(a => `<li>${AM.esc(a.artist || a.name)}` +
        `<span class="tag">${(+(a.distance || 0)).toFixed(3)}</span></li>`)
        .join("") || `<li class="muted">no results</li>`;
  },

  async makePlaylist() {
    const name = document.getElementById("pl-name").value || "AudioMuse mix";
    const ids = this.lastResults.map(t => t.item_id);
    if (!ids.length) { AM.status("no results to save", "warn"); return; }
    const out = await AM.api("/api/create_playlist",
      { json: { name, item_ids: ids } });
    AM.status(`playlist saved (${out.id || out.tracks || ids.length})`, "ok");
  },
};
AM.register("library", el => Library.render(el));
