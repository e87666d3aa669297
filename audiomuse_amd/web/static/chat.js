/* Instant playlist chat with SSE streaming (reference: app_chat.py
   chat page + /chat/api/chatPlaylistStream). */
"use strict";

const Chat = {
  render(el) {
    el.innerHTML = `
      <section class="wide"><h2>Instant playlist</h2>
        <div class="chat-log" id="chat-log">
          <div class="chat-msg"><span class="who">AudioMuse</span>
            <div>Describe the playlist you want — e.g. “20 upbeat
            electronic songs for running, no ballads”. The planner turns
            it into library tool calls (seed search, text match, database
            filters) and re-ranks the union.</div></div></div>
        <div class="row" style="margin-top:10px">
          <input id="chat-q" size="60"
            placeholder="a mellow rainy-evening jazz playlist...">
          <button class="primary" onclick="Chat.send()">Send</button>
        </div>
        <ul class="list" id="chat-tracks"></ul>
        <div class="row">
          <input id="chat-pl" size="18" placeholder="playlist name">
          <button onclick="Chat.save()">Create playlist</button></div>
      </section>`;
    AM.bindTrackList(document.getElementById("chat-tracks"),
      id => { location.hash = "#library"; Library.pick(id); });
    document.getElementById("chat-q").addEventListener("keydown",
      ev => { if (ev.key === "Enter") Chat.send(); });
  },

  log(who, html) {
    const box = document.getElementById("chat-log");
    box.insertAdjacentHTML("beforeend",
      `<div class="chat-msg"><span class="who">${AM.esc(who)}</span>
       <div>${html}</div></div>`);
    box.scrollTop = box.scrollHeight;
  },

  async send() {
    const q = document.getElementById("chat-q").value.trim();
    if (!q) return;
    document.getElementById("chat-q").value = "";
    this.log("you", AM.esc(q));
    // SSE over fetch: the endpoint is POST, so parse the stream manually
    const resp = await fetch("/chat/api/chatPlaylistStream", {
      method: "POST", headers: { "Content-Type": "application/json" },
      body: JSON.stringify({ prompt: q }) });
    if (!resp.ok) {
      const err = await resp.json().catch(() => ({}));
      this.log("AudioMuse", `<span class="err">${
        AM.esc(err.error || resp.statusText)}</span>`);
      return;
    }
    const reader = resp.body.getReader();
    const dec = new TextDecoder();
    let buf = "";
    for (;;) {
      const { value, done } = await reader.read();
      if (done) break;
      buf += dec.decode(value, { stream: true });
      let idx;
      while ((idx = buf.indexOf("\n\n")) >= 0) {
        this.handleEvent(buf.slice(0, idx));
        buf = buf.slice(idx + 2);
      }
    }
  },

  handleEvent(chunk) {
    const ev = (chunk.match(/^event: (.*)$/m) || [])[1];
    const data = (chunk.match(/^data: (.*)$/m) || [])[1];
    if (!ev || !data) return;
    let body;
    try { body = JSON.parse(data); } catch (e) { return; }
    if (ev === "plan") {
      this.log("planner", body.map(c =>
        `<span class="tag">${AM.esc(c.tool)}</span> ${
          AM.esc(JSON.stringify(c.args))}`).join("<br>"));
    } else if (ev === "tool") {
      this.log("tool", `${AM.esc(body.tool)} → ${body.n} candidates`);
    } else if (ev === "playlist") {
      this.tracks = Array.isArray(body) ? body : (body.tracks || []);
      this.log("AudioMuse", `${this.tracks.length} tracks selected`);
      document.getElementById("chat-tracks").innerHTML =
        this.tracks.map(t => AM.trackLi(t)).join("");
    } else if (ev === "error") {
      this.log("AudioMuse",
        `<span class="err">${AM.esc(body.error || "failed")}</span>`);
    }
  },

  async save() {
    const name = document.getElementById("chat-pl").value || "Instant mix";
    const ids = (this.tracks || []).map(t => t.item_id);
    if (!ids.length) { AM.status("ask for a playlist first", "warn"); return; }
    await AM.api("/api/create_playlist", { json: { name, item_ids: ids } });
    AM.status("playlist created", "ok");
  },
};
AM.register("chat", el => Chat.render(el));
