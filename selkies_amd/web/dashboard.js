/* selkies_amd dashboard — slide-in panel (settings / stats / files /
 * clipboard), the compact equivalent of the reference dashboard addons
 * (SURVEY.md §2.5). Talks to the core client over the same WS + /api. */
"use strict";

(function () {
  const css = `
  #sk-dash { position:fixed; top:0; right:-320px; width:300px; height:100%;
    background:#1b1e24f2; color:#ddd; font:13px system-ui,sans-serif;
    transition:right .2s; z-index:100; padding:10px; overflow-y:auto; }
  #sk-dash.open { right:0; }
  #sk-dash h3 { margin:12px 0 6px; font-size:13px; color:#8ab4f8; }
  #sk-dash label { display:flex; justify-content:space-between;
    margin:4px 0; align-items:center; }
  #sk-dash input, #sk-dash select { width:130px; }
  #sk-dash-tab { position:fixed; top:45%; right:0; width:22px; height:64px;
    background:#8ab4f8; color:#000; border-radius:6px 0 0 6px;
    cursor:pointer; z-index:101; display:flex; align-items:center;
    justify-content:center; font-weight:bold; }
  #sk-dash pre { background:#0008; padding:6px; border-radius:4px;
    font-size:11px; white-space:pre-wrap; }
  #sk-dash .file { display:flex; justify-content:space-between;
    padding:2px 0; border-bottom:1px solid #333; }
  #sk-dash button { margin:2px; }
  `;
  const style = document.createElement("style");
  style.textContent = css;
  document.head.appendChild(style);

  const tab = document.createElement("div");
  tab.id = "sk-dash-tab";
  tab.textContent = "☰";
  document.body.appendChild(tab);

  const panel = document.createElement("div");
  panel.id = "sk-dash";
  panel.innerHTML = `
    <h3>Stream</h3>
    <div id="sk-settings"></div>
    <h3>Stats</h3>
    <pre id="sk-stats">waiting…</pre>
    <h3>Clipboard</h3>
    <textarea id="sk-clip" rows="3" style="width:100%"></textarea>
    <div>
      <button id="sk-clip-send">send to server</button>
      <button id="sk-clip-get">fetch</button>
    </div>
    <h3>Files</h3>
    <input type="file" id="sk-upload">
    <div id="sk-files"></div>
    <h3>Share</h3>
    <div>
      <button id="sk-share-view">copy viewer link</button>
      <button id="sk-share-rtc">copy WebRTC link</button>
    </div>
    <h3>Touch gamepad</h3>
    <div>
      <select id="sk-touch-profile">
        <option value="modern">modern (dual stick)</option>
        <option value="classic">classic (d-pad)</option>
      </select>
      <button id="sk-touch-toggle">show / hide</button>
    </div>
    <div id="sk-seats"></div>
  `;
  document.body.appendChild(panel);
  tab.onclick = () => panel.classList.toggle("open");

  const send = (msg) => window.skSend && window.skSend(msg);

  /* touch gamepad overlay toggle (reference universal-touch-gamepad) */
  const touchSel = panel.querySelector("#sk-touch-profile");
  touchSel.value = localStorage.getItem("selkies.touchProfile") || "modern";
  touchSel.onchange = () => {
    try {
      localStorage.setItem("selkies.touchProfile", touchSel.value);
    } catch (e) {}
  };
  panel.querySelector("#sk-touch-toggle").onclick = () => {
    if (window.skToggleTouchGamepad)
      window.skToggleTouchGamepad(touchSel.value);
  };

  /* settings widgets built from the server's client-settings contract */
  const WIDGETS = ["encoder", "framerate", "video_crf",
                   "video_bitrate_kbps", "video_cbr_mode", "jpeg_quality",
                   "use_paint_over_quality", "video_fullframe",
                   "capture_scale_div", "enable_audio"];
  window.skOnSettings = (payload) => {
    /* player-seat claim buttons appear when the server enables them
       (reference dashboard PlayerGamepadButton) */
    const seatHost = document.getElementById("sk-seats");
    if (seatHost) {
      seatHost.innerHTML = "";
      for (let p = 2; p <= 4; p++) {
        const def = payload["enable_player" + p];
        if (!def || !def.value) continue;
        const b = document.createElement("button");
        b.textContent = "claim player " + p + " gamepad";
        b.onclick = () => send("CLAIM_SEAT," + (p - 1));
        seatHost.appendChild(b);
      }
    }
    const host = document.getElementById("sk-settings");
    host.innerHTML = "";
    for (const name of WIDGETS) {
      const def = payload[name];
      if (!def) continue;
      const label = document.createElement("label");
      label.textContent = name.replace(/_/g, " ");
      let input;
      if (def.allowed) {
        input = document.createElement("select");
        for (const opt of def.allowed) {
          const o = document.createElement("option");
          o.value = o.textContent = opt;
          input.appendChild(o);
        }
        input.value = def.value;
      } else if (typeof def.value === "boolean") {
        input = document.createElement("input");
        input.type = "checkbox";
        input.checked = def.value;
      } else {
        input = document.createElement("input");
        input.type = "number";
        if (def.range) { input.min = def.range[0]; input.max = def.range[1]; }
        input.value = def.value;
      }
      input.disabled = !!def.locked;
      input.onchange = () => {
        const v = input.type === "checkbox" ? input.checked
            : (input.type === "number" ? +input.value : input.value);
        send("SETTINGS," + JSON.stringify({ [name]: v }));
        /* remember the preference so reconnects replay it */
        try {
          const prefs = JSON.parse(
              localStorage.getItem("selkies.prefs") || "{}");
          prefs[name] = v;
          localStorage.setItem("selkies.prefs", JSON.stringify(prefs));
        } catch (e) { /* private mode */ }
      };
      label.appendChild(input);
      host.appendChild(label);
    }
  };

  window.skOnStats = (stats) => {
    const el = document.getElementById("sk-stats");
    const gpu = (stats.gpus || [])
        .map((g) => `${g.card}: ${g.busy_percent}%`).join("  ");
    el.textContent =
        `cpu ${stats.cpu_percent}%  mem ${stats.mem_percent}%\n` +
        `gpu ${gpu || "n/a"}\n` +
        `encode ${stats.stream ? stats.stream.last_encode_ms.toFixed(2)
                               : "?"} ms  ` +
        `enc=${stats.stream ? stats.stream.encoder : ""}`;
  };

  /* clipboard */
  document.getElementById("sk-clip-send").onclick = () => {
    const text = document.getElementById("sk-clip").value;
    send("cw," + btoa(unescape(encodeURIComponent(text))));
  };
  document.getElementById("sk-clip-get").onclick = () => send("cr,");
  window.skOnClipboard = (text) => {
    document.getElementById("sk-clip").value = text;
  };

  /* files */
  async function refreshFiles() {
    try {
      const r = await fetch("/api/files" + location.search);
      const files = await r.json();
      const host = document.getElementById("sk-files");
      host.innerHTML = "";
      for (const f of files) {
        const row = document.createElement("div");
        row.className = "file";
        const a = document.createElement("a");
        a.textContent = `${f.name} (${f.size}B)`;
        a.href = "/api/download?name=" + encodeURIComponent(f.name);
        a.style.color = "#8ab4f8";
        row.appendChild(a);
        host.appendChild(row);
      }
    } catch (e) { /* server may not be up yet */ }
  }
  document.getElementById("sk-upload").onchange = async (ev) => {
    const file = ev.target.files[0];
    if (!file) return;
    await fetch("/api/upload?name=" + encodeURIComponent(file.name) +
                (location.search ? "&" + location.search.slice(1) : ""),
                { method: "POST", body: file });
    refreshFiles();
  };
  setInterval(refreshFiles, 5000);
  refreshFiles();

  /* sharing links (reference dashboard "sharing links" section) */
  function shareUrl(params) {
    const u = new URL(location.href);
    for (const [k, v] of Object.entries(params)) u.searchParams.set(k, v);
    return u.toString();
  }
  document.getElementById("sk-share-view").onclick = () =>
      navigator.clipboard.writeText(shareUrl({ role: "viewer" }))
          .catch(() => {});
  document.getElementById("sk-share-rtc").onclick = () =>
      navigator.clipboard.writeText(shareUrl({ transport: "webrtc" }))
          .catch(() => {});
})();
