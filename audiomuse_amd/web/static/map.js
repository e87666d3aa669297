/* Music map: first-party WebGL scatter (reference: Plotly WebGL map in
   static/sunburst.js over app_map.py buckets; here a dependency-free
   point renderer with pan/zoom, hover picking and click -> similar). */
"use strict";

const MusicMap = {
  gl: null, prog: null, buf: null, n: 0,
  points: [],                 // [{item_id, x, y}]
  centroids: [],              // [{mood, x, y, count}] (song map overlay)
  view: { cx: 0, cy: 0, scale: 1 },
  grid: null, cell: 0,        // uniform hover grid

  render(el) {
    el.innerHTML = `
      <section class="wide"><h2>Music map</h2>
        <div class="row">
          <select id="map-kind"><option value="song">songs</option>
            <option value="artist">artists</option></select>
          <select id="map-pct"><option>100</option><option>75</option>
            <option>50</option><option>25</option></select>
          <button class="primary" onclick="MusicMap.load()">Load</button>
          <span id="map-info" class="muted"></span>
        </div>
        <div style="position:relative">
          <canvas class="map" id="map-canvas"></canvas>
          <div id="map-labels"></div></div>
        <p class="muted">drag to pan · wheel to zoom · hover for the id ·
          click a point for similar songs (opens Library)</p>
        <ul class="list" id="map-similar"></ul>
      </section>`;
    const canvas = document.getElementById("map-canvas");
    this.initGL(canvas);
    this.bindEvents(canvas);
    AM.bindTrackList(document.getElementById("map-similar"),
      id => { location.hash = "#library"; Library.pick(id); });
    this.load();
  },

  initGL(canvas) {
    const dpr = window.devicePixelRatio || 1;
    canvas.width = canvas.clientWidth * dpr;
    canvas.height = 560 * dpr;
    const gl = canvas.getContext("webgl", { antialias: true });
    if (!gl) { AM.status("WebGL unavailable; map disabled", "warn"); return; }
    this.gl = gl;
    const vs = `attribute vec2 p; uniform vec3 view; uniform float ps;
      void main() {
        vec2 q = (p - view.xy) * view.z;
        gl_Position = vec4(q, 0.0, 1.0);
        gl_PointSize = ps;
      }`;
    const fs = `precision mediump float;
      void main() {
        vec2 d = gl_PointCoord - vec2(0.5);
        if (dot(d, d) > 0.25) discard;
        gl_FragColor = vec4(0.50, 0.82, 0.73, 0.75);
      }`;
    const mk = (type, src) => {
      const s = gl.createShader(type);
      gl.shaderSource(s, src); gl.compileShader(s);
      return s;
    };
    const prog = gl.createProgram();
    gl.attachShader(prog, mk(gl.VERTEX_SHADER, vs));
    gl.attachShader(prog, mk(gl.FRAGMENT_SHADER, fs));
    gl.linkProgram(prog);
    gl.useProgram(prog);
    this.prog = prog;
    this.buf = gl.createBuffer();
    gl.enable(gl.BLEND);
    gl.blendFunc(gl.SRC_ALPHA, gl.ONE_MINUS_SRC_ALPHA);
  },

  async load() {
    const kind = document.getElementById("map-kind").value;
    const pct = document.getElementById("map-pct").value;
    try {
      const pts = await AM.api(`/api/map?kind=${kind}&percent=${pct}`);
      this.points = pts;
      this.centroids = kind === "song"
        ? await AM.api("/api/mood_centroids").catch(() => []) : [];
      document.getElementById("map-info").textContent =
        `${pts.length} points`;
      this.upload();
      this.buildGrid();
      this.fit();
      this.draw();
    } catch (e) {
      document.getElementById("map-info").textContent = e.message;
    }
  },

  upload() {
    const gl = this.gl;
    if (!gl) return;
    const arr = new Float32Array(this.points.length * 2);
    this.points.forEach((p, i) => { arr[2 * i] = p.x; arr[2 * i + 1] = p.y; });
    gl.bindBuffer(gl.ARRAY_BUFFER, this.buf);
    gl.bufferData(gl.ARRAY_BUFFER, arr, gl.STATIC_DRAW);
    this.n = this.points.length;
  },

  fit() {
    if (!this.points.length) return;
    let xmin = 1e9, xmax = -1e9, ymin = 1e9, ymax = -1e9;
    for (const p of this.points) {
      xmin = Math.min(xmin, p.x); xmax = Math.max(xmax, p.x);
      ymin = Math.min(ymin, p.y); ymax = Math.max(ymax, p.y);
    }
    this.view.cx = (xmin + xmax) / 2;
    this.view.cy = (ymin + ymax) / 2;
    this.view.scale = 1.8 / Math.max(xmax - xmin, ymax - ymin, 1e-6);
  },

  buildGrid() {
    // uniform grid over data space for O(1) hover picking
    const g = new Map();
    const cell = 0.05;
    for (let i = 0; i < this.points.length; i++) {
      const p = this.points[i];
      const k = `${Math.floor(p.x / cell)}:${Math.floor(p.y / cell)}`;
      if (!g.has(k)) g.set(k, []);
      g.get(k).push(i);
    }
    this.grid = g; this.cell = cell;
  },

  nearest(x, y) {
    if (!this.grid) return null;
    const c = this.cell;
    let best = null, bd = (8 / (this.view.scale * 280)) ** 2; // px threshold
    for (let dx = -1; dx <= 1; dx++)
      for (let dy = -1; dy <= 1; dy++) {
        const k = `${Math.floor(x / c) + dx}:${Math.floor(y / c) + dy}`;
        for (const i of this.grid.get(k) || []) {
          const p = this.points[i];
          const d = (p.x - x) ** 2 + (p.y - y) ** 2;
          if (d < bd) { bd = d; best = p; }
        }
      }
    return best;
  },

  draw() {
    const gl = this.gl;
    if (!gl || !this.n) return;
    gl.viewport(0, 0, gl.canvas.width, gl.canvas.height);
    gl.clearColor(0.05, 0.067, 0.09, 1);
    gl.clear(gl.COLOR_BUFFER_BIT);
    const loc = gl.getAttribLocation(this.prog, "p");
    gl.bindBuffer(gl.ARRAY_BUFFER, this.buf);
    gl.enableVertexAttribArray(loc);
    gl.vertexAttribPointer(loc, 2, gl.FLOAT, false, 0, 0);
    gl.uniform3f(gl.getUniformLocation(this.prog, "view"),
                 this.view.cx, this.view.cy, this.view.scale);
    const ps = Math.max(2, Math.min(9, this.view.scale * 2.2));
    gl.uniform1f(gl.getUniformLocation(this.prog, "ps"),
                 ps * (window.devicePixelRatio || 1));
    gl.drawArrays(gl.POINTS, 0, this.n);
    this.drawLabels();
  },

  /* mood-centroid labels ride an HTML overlay (no GL text): positions
     recompute with the same view transform on every draw */
  drawLabels() {
    const box = document.getElementById("map-labels");
    if (!box || !this.gl) return;
    const w = this.gl.canvas.clientWidth, h = 560;
    box.innerHTML = this.centroids.map(c => {
      const nx = (c.x - this.view.cx) * this.view.scale;
      const ny = (c.y - this.view.cy) * this.view.scale;
      if (nx < -0.98 || nx > 0.98 || ny < -0.98 || ny > 0.98) return "";
      return `<span class="map-label" style="left:${(nx + 1) / 2 * w}px;` +
        `top:${(1 - ny) / 2 * h}px">${AM.esc(c.mood)}</span>`;
    }).join("");
  },

  canvasToData(ev, canvas) {
    const r = canvas.getBoundingClientRect();
    const nx = ((ev.clientX - r.left) / r.width) * 2 - 1;
    const ny = 1 - ((ev.clientY - r.top) / r.height) * 2;
    return [nx / this.view.scale + this.view.cx,
            ny / this.view.scale + this.view.cy];
  },

  bindEvents(canvas) {
    let dragging = false, moved = false, last = null;
    const tip = document.getElementById("tooltip");
    canvas.addEventListener("mousedown", ev => {
      dragging = true; moved = false; last = [ev.clientX, ev.clientY];
    });
    window.addEventListener("mouseup", () => { dragging = false; });
    canvas.addEventListener("mousemove", ev => {
      if (dragging) {
        const r = canvas.getBoundingClientRect();
        const dx = (ev.clientX - last[0]) / r.width * 2 / this.view.scale;
        const dy = (ev.clientY - last[1]) / r.height * 2 / this.view.scale;
        this.view.cx -= dx; this.view.cy += dy;
        last = [ev.clientX, ev.clientY];
        moved = true;
        this.draw();
        return;
      }
      const [x, y] = this.canvasToData(ev, canvas);
      const hit = this.nearest(x, y);
      if (hit) {
        tip.style.display = "block";
        tip.style.left = (ev.clientX + 12) + "px";
        tip.style.top = (ev.clientY + 12) + "px";
        tip.textContent = hit.item_id || hit.artist || "";
      } else tip.style.display = "none";
    });
    canvas.addEventListener("mouseleave",
      () => { tip.style.display = "none"; });
    canvas.addEventListener("wheel", ev => {
      ev.preventDefault();
      const f = ev.deltaY < 0 ? 1.15 : 1 / 1.15;
      const [x, y] = this.canvasToData(ev, canvas);
      this.view.cx = x + (this.view.cx - x) / f;
      this.view.cy = y + (this.view.cy - y) / f;
      this.view.scale *= f;
      this.draw();
    }, { passive: false });
    canvas.addEventListener("click", async ev => {
      if (moved) return;
      const [x, y] = this.canvasToData(ev, canvas);
      const hit = this.nearest(x, y);
      if (!hit || !hit.item_id) return;
      const out = await AM.api(`/api/similar_tracks?item_id=${
        encodeURIComponent(hit.item_id)}&n=15`);
      document.getElementById("map-similar").innerHTML =
        `<li class="muted">similar to ${AM.esc(hit.item_id)}:</li>` +
        out.map(t => AM.trackLi(t)).join("");
    });
  },
};
AM.register("map", el => MusicMap.render(el));
