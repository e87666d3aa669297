/* Setup wizard + login (reference: app_setup.py:1293 wizard flow +
   app_auth.py login barrier). Wizard: admin account -> media server ->
   config review -> finish. */
"use strict";

const Setup = {
  showLogin() {
    AM.overlay(`
      <h2>Sign in</h2>
      <label>Username</label><input id="lg-user">
      <label>Password</label><input id="lg-pass" type="password">
      <div class="row" style="margin-top:14px">
        <button class="primary" onclick="Setup.doLogin()">Sign in</button>
        <span id="lg-err" class="err"></span>
      </div>`);
  },

  async doLogin() {
    try {
      await AM.api("/api/login", { json: {
        username: document.getElementById("lg-user").value,
        password: document.getElementById("lg-pass").value } });
      AM.closeOverlay();
      AM.show(location.hash.slice(1) || "dashboard");
    } catch (e) {
      document.getElementById("lg-err").textContent = "invalid credentials";
    }
  },

  step(n) {
    document.querySelectorAll(".wizard-step").forEach((el, i) =>
      el.classList.toggle("on", i === n));
    document.querySelectorAll(".steps span").forEach((el, i) =>
      el.classList.toggle("on", i === n));
  },

  showWizard() {
    AM.overlay(`
      <h2>Welcome — first-boot setup</h2>
      <div class="steps"><span>Admin</span><span>Media server</span>
        <span>Settings</span><span>Done</span></div>

      <div class="wizard-step on">
        <label>Admin username</label><input id="sw-user">
        <label>Password (min 8 chars)</label>
        <input id="sw-pass" type="password">
        <div class="row" style="margin-top:12px">
          <button class="primary" onclick="Setup.createAdmin()">Create
            account</button><span id="sw-err1" class="err"></span></div>
      </div>

      <div class="wizard-step">
        <label>Server type</label>
        <select id="sw-type">
          <option>navidrome</option><option>jellyfin</option>
          <option>emby</option><option>lyrion</option><option>plex</option>
          <option>synthetic</option></select>
        <label>Base URL</label><input id="sw-url"
          placeholder="http://server:4533">
        <label>Username (if password auth)</label><input id="sw-suser">
        <label>Token / password / API key</label>
        <input id="sw-cred" type="password">
        <div class="row" style="margin-top:12px">
          <button onclick="Setup.testServer()">Test connection</button>
          <button class="primary" onclick="Setup.saveServer()">Save &amp;
            continue</button>
          <span id="sw-err2"></span></div>
      </div>

      <div class="wizard-step">
        <p class="muted">Defaults work out of the box; every parameter can
          be changed later in Admin &gt; Settings (persisted to the
          app_config table, reference Setup Wizard behavior).</p>
        <label>CLAP analysis enabled</label>
        <select id="sw-clap"><option value="1">yes</option>
          <option value="0">no</option></select>
        <label>Lyrics analysis enabled</label>
        <select id="sw-lyr"><option value="0">no</option>
          <option value="1">yes</option></select>
        <div class="row" style="margin-top:12px">
          <button class="primary" onclick="Setup.saveConfig()">Save &amp;
            continue</button></div>
      </div>

      <div class="wizard-step">
        <p>Setup complete. Start the first analysis from the Dashboard —
          workers pick it up from the queue.</p>
        <div class="row"><button class="primary"
          onclick="Setup.finish()">Open dashboard</button></div>
      </div>`);
    this.step(0);
  },

  async createAdmin() {
    try {
      await AM.api("/api/setup/admin", { json: {
        username: document.getElementById("sw-user").value,
        password: document.getElementById("sw-pass").value } });
      await AM.api("/api/login", { json: {
        username: document.getElementById("sw-user").value,
        password: document.getElementById("sw-pass").value } });
      this.step(1);
    } catch (e) {
      document.getElementById("sw-err1").textContent = e.message;
    }
  },

  serverBody() {
    return {
      server_type: document.getElementById("sw-type").value,
      base_url: document.getElementById("sw-url").value,
      username: document.getElementById("sw-suser").value,
      credential: document.getElementById("sw-cred").value,
    };
  },

  async testServer() {
    const el = document.getElementById("sw-err2");
    el.textContent = "probing..."; el.className = "muted";
    try {
      const out = await AM.api("/api/migration/probe", { json: {
        server_type: this.serverBody().server_type,
        server_config: this.serverBody() } });
      el.textContent = out.reachable
        ? `reachable — ${out.libraries.length} libraries` : "unreachable";
      el.className = out.reachable ? "ok" : "err";
    } catch (e) { el.textContent = e.message; el.className = "err"; }
  },

  async saveServer() {
    try {
      await AM.api("/api/servers", { json: this.serverBody() });
      this.step(2);
    } catch (e) {
      const el = document.getElementById("sw-err2");
      el.textContent = e.message; el.className = "err";
    }
  },

  async saveConfig() {
    await AM.api("/api/config", { json: {
      CLAP_ENABLED: document.getElementById("sw-clap").value,
      LYRICS_ENABLED: document.getElementById("sw-lyr").value } })
      .catch(() => {});
    this.step(3);
  },

  finish() {
    AM.closeOverlay();
    location.hash = "#dashboard";
    AM.show("dashboard");
  },
};
