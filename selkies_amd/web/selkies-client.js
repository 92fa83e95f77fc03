/* selkies_amd HTML5 client.
 *
 * Speaks the binary stripe wire protocol (SURVEY.md §3.2):
 *   0x03 JPEG stripe  [tag, flags, frame_id u16, y u16] + JFIF
 *   0x04 H.264 stripe [tag, key, frame_id u16, y u16, w u16, h u16] + AnnexB
 *   0x06 HEVC stripe  same header as 0x04, H.265 AnnexB
 * H.264/HEVC stripes feed one WebCodecs VideoDecoder PER STRIPE ROW (each row is
 * an independent bitstream); decoded rows composite onto the canvas.
 * JPEG stripes decode via createImageBitmap. Input is captured and sent as
 * the text verbs the server's InputDispatcher understands.
 */
"use strict";

const canvas = document.getElementById("screen");
const ctx2d = canvas.getContext("2d");
const stateEl = document.getElementById("state");
const statsEl = document.getElementById("stats");

let ws = null;
let serverSettings = {};
let frameCount = 0, byteCount = 0, lastStats = performance.now();
let lastAckedFrame = -1;

/* ---------------- decode sinks ---------------- */

class H264Row {
  constructor(y, codec) {
    this.y = y;
    this.codec = codec || "avc1.42e028";
    this.decoder = null;
    this.width = 0;
    this.height = 0;
  }
  configure(w, h) {
    if (this.decoder && this.width === w && this.height === h) return;
    if (this.decoder) try { this.decoder.close(); } catch (e) {}
    this.width = w; this.height = h;
    this.decoder = new VideoDecoder({
      output: (frame) => {
        ctx2d.drawImage(frame, 0, this.y);
        frame.close();
      },
      error: (e) => { console.warn("decoder error row", this.y, e);
                      this.decoder = null; requestIdr(); },
    });
    this.decoder.configure({
      codec: this.codec,
      optimizeForLatency: true,
    });
  }
  push(payload, key, frameId) {
    if (!this.decoder) return;
    try {
      this.decoder.decode(new EncodedVideoChunk({
        type: key ? "key" : "delta",
        timestamp: frameId * 16667,
        data: payload,
      }));
    } catch (e) { console.warn(e); requestIdr(); }
  }
}

const h264Rows = new Map();   // y -> H264Row
let jpegDrawQueue = Promise.resolve();

function ensureCanvas(w, hTotal) {
  if (canvas.width !== w) canvas.width = w;
  if (hTotal > canvas.height) canvas.height = hTotal;
}

function onBinary(buf) {
  const d = new Uint8Array(buf);
  byteCount += d.length;
  const tag = d[0];
  if (tag === 0x04 || tag === 0x06) {
    const key = d[1] === 1;
    const frameId = (d[2] << 8) | d[3];
    const y = (d[4] << 8) | d[5];
    const w = (d[6] << 8) | d[7];
    const h = (d[8] << 8) | d[9];
    ensureCanvas(w, y + h);
    // 0x06 = HEVC Main (WebCodecs hev1 codec string); per-row decoders
    const codec = tag === 0x06 ? "hev1.1.6.L123.B0" : "avc1.42e028";
    let row = h264Rows.get(y);
    if (!row || row.codec !== codec) {
      row = new H264Row(y, codec); h264Rows.set(y, row);
    }
    if (key) row.configure(w, h);
    row.push(d.subarray(10), key, frameId);
    noteFrame(frameId);
  } else if (tag === 0x03) {
    const frameId = (d[2] << 8) | d[3];
    const y = (d[4] << 8) | d[5];
    const blob = new Blob([d.subarray(6)], { type: "image/jpeg" });
    jpegDrawQueue = jpegDrawQueue.then(async () => {
      try {
        const img = await createImageBitmap(blob);
        ensureCanvas(img.width, y + img.height);
        ctx2d.drawImage(img, 0, y);
        img.close();
      } catch (e) { console.warn("jpeg decode", e); }
    });
    noteFrame(frameId);
  } else if (tag === 0x01) {
    playAudioFrame(d);
  } else if (tag === 0x05) {
    /* gzip'd control text: rare server->client path; ignore for now */
  }
}

/* ---------------- audio playback ----------------
 * wire: [0x01, n_red] + n_red x (u16 len + redundant payload) + primary.
 * Payload is an Opus (CELT-class) packet by default — decoded by
 * opus-decoder.js (SkOpus) — or s16le PCM when audio_codec=pcm. WS is
 * reliable, so RED payloads are skipped. */
let audioCtx = null, audioTime = 0, gainNode = null, masterVolume = 1,
    audioMuted = false;
const AUDIO_RATE = 48000, AUDIO_CH = 2;
let audioCodec = null;                 /* null = sniff until SETTINGS */
let opusDec = null;

function ensureAudioCtx() {
  if (audioCtx) return true;
  try {
    audioCtx = new AudioContext({ sampleRate: AUDIO_RATE });
    gainNode = audioCtx.createGain();
    gainNode.gain.value = audioMuted ? 0 : masterVolume;
    gainNode.connect(audioCtx.destination);
    return true;
  } catch (e) { return false; }
}

function schedulePlay(buf) {
  const src = audioCtx.createBufferSource();
  src.buffer = buf;
  src.connect(gainNode || audioCtx.destination);
  const now = audioCtx.currentTime;
  if (audioTime < now + 0.02) audioTime = now + 0.04;  // jitter buffer
  src.start(audioTime);
  audioTime += buf.length / AUDIO_RATE;
}

function playAudioFrame(d) {
  let off = 2;
  const nRed = d[1];
  for (let i = 0; i < nRed; i++) {
    const len = (d[off] << 8) | d[off + 1];
    off += 2 + len;
  }
  if (!ensureAudioCtx()) return;
  const payload = d.subarray(off);
  /* codec: explicit from SETTINGS_PAYLOAD, else sniff — a 20 ms PCM
     frame is >= 1920 B while CBR Opus at <= 500 kb/s is < 1300 B */
  const isOpus = audioCodec === "opus" ||
      (audioCodec === null && payload.length < 1500 &&
       (payload[0] >> 3) === 31);
  if (isOpus && typeof SkOpus !== "undefined") {
    try {
      if (!opusDec) opusDec = new SkOpus.OpusDecoder();
      const mono = opusDec.decodePacket(payload);
      const buf = audioCtx.createBuffer(1, mono.length, AUDIO_RATE);
      buf.getChannelData(0).set(mono);
      schedulePlay(buf);
    } catch (e) { console.warn("opus decode", e); }
    return;
  }
  const pcm = new Int16Array(payload.buffer, payload.byteOffset,
                             payload.length >> 1);
  const framesN = pcm.length / AUDIO_CH;
  const buf = audioCtx.createBuffer(AUDIO_CH, framesN, AUDIO_RATE);
  for (let c = 0; c < AUDIO_CH; c++) {
    const chan = buf.getChannelData(c);
    for (let i = 0; i < framesN; i++) chan[i] = pcm[i * AUDIO_CH + c] / 32768;
  }
  schedulePlay(buf);
}

function noteFrame(frameId) {
  frameCount++;
  if (frameId !== lastAckedFrame && ws && ws.readyState === 1) {
    lastAckedFrame = frameId;
    ws.send("CLIENT_FRAME_ACK," + frameId);
  }
}

function requestIdr() {
  if (ws && ws.readyState === 1) ws.send("REQUEST_IDR,");
}

/* ---------------- input capture ---------------- */

/* Browser key -> X11 keysym (reference keysym wire contract). Printable
 * characters map via unicode (latin1 direct, others 0x01000000+cp);
 * specials via this table (X11/keysymdef.h values). */
const KEYSYMS = {
  Backspace: 0xff08, Tab: 0xff09, Enter: 0xff0d, Escape: 0xff1b,
  Delete: 0xffff, Home: 0xff50, End: 0xff57, PageUp: 0xff55,
  PageDown: 0xff56, ArrowLeft: 0xff51, ArrowUp: 0xff52,
  ArrowRight: 0xff53, ArrowDown: 0xff54, Insert: 0xff63,
  ShiftLeft: 0xffe1, ShiftRight: 0xffe2, ControlLeft: 0xffe3,
  ControlRight: 0xffe4, AltLeft: 0xffe9, AltRight: 0xffea,
  MetaLeft: 0xffeb, MetaRight: 0xffec, CapsLock: 0xffe5,
  F1: 0xffbe, F2: 0xffbf, F3: 0xffc0, F4: 0xffc1, F5: 0xffc2,
  F6: 0xffc3, F7: 0xffc4, F8: 0xffc5, F9: 0xffc6, F10: 0xffc7,
  F11: 0xffc8, F12: 0xffc9, Space: 0x20,
};

function keysymOf(ev) {
  if (KEYSYMS[ev.code] !== undefined && ev.key.length !== 1)
    return KEYSYMS[ev.code];
  if (ev.key.length === 1) {
    const cp = ev.key.codePointAt(0);
    return cp < 0x100 ? cp : 0x01000000 + cp;
  }
  return KEYSYMS[ev.code];
}

let buttonMask = 0;

function canvasPos(ev) {
  const r = canvas.getBoundingClientRect();
  const x = Math.round((ev.clientX - r.left) * canvas.width / r.width);
  const y = Math.round((ev.clientY - r.top) * canvas.height / r.height);
  return [Math.max(0, Math.min(canvas.width - 1, x)),
          Math.max(0, Math.min(canvas.height - 1, y))];
}

function send(msg) {
  /* input verbs prefer the WebRTC data channel (lower latency path);
     everything else (and fallback) uses the control WebSocket */
  if (dcInput && dcInput.readyState === "open") {
    try { dcInput.send(msg); return; } catch (e) { /* fall through */ }
  }
  if (ws && ws.readyState === 1) ws.send(msg);
}
window.skSend = send;   /* dashboard hook */

function hookInput() {
  canvas.addEventListener("keydown", (ev) => {
    /* browser auto-repeat is suppressed: the server X autorepeat is the
       single repeat source for XTEST-held keys (double-repeat fix) */
    if (ev.repeat) { ev.preventDefault(); return; }
    const ks = keysymOf(ev);
    if (ks !== undefined) { send("kd," + ks); ev.preventDefault(); }
  });
  canvas.addEventListener("keyup", (ev) => {
    const ks = keysymOf(ev);
    if (ks !== undefined) { send("ku," + ks); ev.preventDefault(); }
  });
  canvas.addEventListener("mousemove", (ev) => {
    if (document.pointerLockElement === canvas) {
      send(`m2,${ev.movementX},${ev.movementY},${buttonMask}`);
    } else {
      const [x, y] = canvasPos(ev);
      send(`m,${x},${y},${buttonMask}`);
    }
  });
  const maskBit = (b) => b === 0 ? 1 : b === 1 ? 2 : b === 2 ? 4 : 0;
  canvas.addEventListener("mousedown", (ev) => {
    canvas.focus();
    buttonMask |= maskBit(ev.button);
    const [x, y] = canvasPos(ev);
    send(`m,${x},${y},${buttonMask}`);
    ev.preventDefault();
  });
  canvas.addEventListener("mouseup", (ev) => {
    buttonMask &= ~maskBit(ev.button);
    const [x, y] = canvasPos(ev);
    send(`m,${x},${y},${buttonMask}`);
  });
  canvas.addEventListener("wheel", (ev) => {
    send("sw," + (ev.deltaY < 0 ? "u" : "d") + ",1");
    ev.preventDefault();
  }, { passive: false });
  canvas.addEventListener("contextmenu", (ev) => ev.preventDefault());
  window.addEventListener("blur", () => send("kr,"));
}

/* ---------------- settings UI ---------------- */

/* ---------------- microphone uplink ----------------
 * getUserMedia -> 48 kHz mono s16 frames over binary [0x02]+PCM
 * (server writes them into the virtual microphone sink when
 * enable_microphone is on). */
let micStream = null, micCtx = null, micNode = null;

async function micStart() {
  if (micStream) return true;
  try {
    micStream = await navigator.mediaDevices.getUserMedia({ audio: {
      channelCount: 1, sampleRate: 48000 } });
    micCtx = new AudioContext({ sampleRate: 48000 });
    const src = micCtx.createMediaStreamSource(micStream);
    micNode = micCtx.createScriptProcessor(2048, 1, 1);
    micNode.onaudioprocess = (ev) => {
      if (!ws || ws.readyState !== 1) return;
      const f = ev.inputBuffer.getChannelData(0);
      const out = new Uint8Array(1 + f.length * 2);
      out[0] = 0x02;
      const dv = new DataView(out.buffer);
      for (let i = 0; i < f.length; i++) {
        let v = Math.max(-1, Math.min(1, f[i]));
        dv.setInt16(1 + i * 2, v * 32767, true);
      }
      ws.send(out);
    };
    src.connect(micNode);
    micNode.connect(micCtx.destination);
    return true;
  } catch (e) { console.warn("mic start failed", e); return false; }
}

function micStop() {
  if (micNode) { micNode.disconnect(); micNode = null; }
  if (micCtx) { micCtx.close().catch(() => {}); micCtx = null; }
  if (micStream) {
    for (const t of micStream.getTracks()) t.stop();
    micStream = null;
  }
}

/* ---------------- file transfers ----------------
 * Upload: streamed POST /api/upload?name= (staging-rename server side);
 * Download: /api/files listing -> /api/download?name= links
 * (reference web-core lib/file-upload.js surface). */
function hookTransfers() {
  const input = document.getElementById("upload-input");
  const btn = document.getElementById("upload-btn");
  const filesBtn = document.getElementById("files-btn");
  const list = document.getElementById("file-list");
  const prog = document.getElementById("upload-progress");
  if (!input || !btn) return;
  btn.onclick = () => input.click();
  input.onchange = async () => {
    for (const f of input.files) {
      prog.style.display = "";
      prog.value = 0;
      try {
        await new Promise((resolve, reject) => {
          /* XHR for upload progress events (fetch lacks them) */
          const xhr = new XMLHttpRequest();
          xhr.open("POST",
                   "/api/upload?name=" + encodeURIComponent(f.name));
          xhr.upload.onprogress = (ev) => {
            if (ev.lengthComputable)
              prog.value = (100 * ev.loaded / ev.total) | 0;
          };
          xhr.onload = () => xhr.status < 300 ? resolve()
                                              : reject(xhr.statusText);
          xhr.onerror = reject;
          xhr.send(f);
        });
      } catch (e) { console.warn("upload failed", e); }
    }
    prog.style.display = "none";
    input.value = "";
  };
  if (filesBtn)
    filesBtn.onclick = async () => {
      if (list.style.display === "") { list.style.display = "none"; return; }
      try {
        const r = await fetch("/api/files");
        const files = await r.json();
        list.innerHTML = "";
        for (const f of files.files || files) {
          const name = f.name || f;
          const li = document.createElement("li");
          const a = document.createElement("a");
          a.href = "/api/download?name=" + encodeURIComponent(name);
          a.textContent = name + (f.size != null ? ` (${f.size} B)` : "");
          li.appendChild(a);
          list.appendChild(li);
        }
        list.style.display = "";
      } catch (e) { console.warn("file list failed", e); }
    };
}

/* clipboard sync worker (reference web-core lib/clipboard-sync.js):
 * on focus, read the local clipboard (permission-gated) and push it to
 * the server when it changed; server->client writes land in
 * lastRemoteClipboard so we don't echo them back. */
let lastSentClipboard = null, lastRemoteClipboard = null;

function hookClipboardSync() {
  const push = async () => {
    if (!navigator.clipboard || !navigator.clipboard.readText) return;
    try {
      const text = await navigator.clipboard.readText();
      if (!text || text === lastSentClipboard ||
          text === lastRemoteClipboard) return;
      lastSentClipboard = text;
      const b64 = btoa(unescape(encodeURIComponent(text)));
      if (b64.length < 64 * 1024) {
        send("cw," + b64);
      } else {
        /* multipart protocol for large payloads (cws/cwd/cwe) */
        const tid = Date.now().toString(36);
        const rawLen = Math.ceil(b64.length / 4) * 3 -
            (b64.endsWith("==") ? 2 : b64.endsWith("=") ? 1 : 0);
        send(`cws,${tid},${rawLen}`);
        for (let i = 0; i < b64.length; i += 64 * 1024)
          send(`cwd,${tid},${b64.slice(i, i + 64 * 1024)}`);
        send(`cwe,${tid}`);
      }
    } catch (e) { /* permission denied: stay quiet */ }
  };
  window.addEventListener("focus", push);
  canvas.addEventListener("mouseenter", push);
}

function hookHud() {
  hookTransfers();
  hookClipboardSync();
  const enc = document.getElementById("encoder");
  const fps = document.getElementById("fps");
  const crf = document.getElementById("crf");
  enc.onchange = () => send('SETTINGS,' +
      JSON.stringify({ encoder: enc.value }));
  fps.onchange = () => send('SETTINGS,' +
      JSON.stringify({ framerate: +fps.value }));
  crf.onchange = () => send('SETTINGS,' +
      JSON.stringify({ video_crf: +crf.value }));
  document.getElementById("idr").onclick = requestIdr;
  const micBtn = document.getElementById("mic");
  if (micBtn)
    micBtn.onclick = async () => {
      if (micStream) { micStop(); micBtn.textContent = "mic off"; }
      else if (await micStart()) micBtn.textContent = "mic on";
    };
}

function applyServerSettings(payload) {
  serverSettings = payload;
  const set = (id, name) => {
    if (payload[name]) document.getElementById(id).value =
        payload[name].value;
  };
  set("encoder", "encoder");
  set("fps", "framerate");
  set("crf", "video_crf");
  /* UI contract knobs (reference ui_*): title + chrome visibility */
  if (payload.ui_title && payload.ui_title.value)
    document.title = payload.ui_title.value;
  const hud = document.getElementById("hud");
  if (hud && payload.ui_show_sidebar)
    hud.style.display = payload.ui_show_sidebar.value ? "" : "none";
  const statsEl = document.getElementById("stats");
  if (statsEl && payload.ui_sidebar_show_stats)
    statsEl.style.display =
        payload.ui_sidebar_show_stats.value ? "" : "none";
  if (payload.use_css_scaling) {
    /* CSS fit-to-window vs native 1:1 (reference use_css_scaling) */
    canvas.style.width = payload.use_css_scaling.value ? "100%" : "";
    canvas.style.height = payload.use_css_scaling.value ? "100%" : "";
  }
  const trEl = document.getElementById("transfers");
  if (trEl && payload.ui_sidebar_show_files)
    trEl.style.display =
        payload.ui_sidebar_show_files.value ? "" : "none";
  if (payload.audio_codec && payload.audio_codec.value) {
    const c = payload.audio_codec.value;
    if (c !== audioCodec) opusDec = null;   /* codec switch: fresh state */
    audioCodec = c;
  }
  const res = payload.resolution && payload.resolution.value;
  if (res) {
    const [w, h] = res.split("x").map(Number);
    canvas.width = w; canvas.height = h;
  }
}

/* server-pushed cursor shape -> CSS cursor on the canvas */
function applyCursor(c) {
  const bin = atob(c.argb_b64);
  const n = c.width * c.height;
  const oc = document.createElement("canvas");
  oc.width = c.width; oc.height = c.height;
  const octx = oc.getContext("2d");
  const img = octx.createImageData(c.width, c.height);
  for (let i = 0; i < n; i++) {
    // little-endian u32 ARGB -> bytes B,G,R,A
    const b = bin.charCodeAt(i * 4), g = bin.charCodeAt(i * 4 + 1);
    const r = bin.charCodeAt(i * 4 + 2), a = bin.charCodeAt(i * 4 + 3);
    img.data[i * 4] = r; img.data[i * 4 + 1] = g;
    img.data[i * 4 + 2] = b; img.data[i * 4 + 3] = a;
  }
  octx.putImageData(img, 0, 0);
  canvas.style.cursor =
      `url(${oc.toDataURL()}) ${c.hot_x} ${c.hot_y}, auto`;
}

/* ---------------- transport ---------------- */

function connect() {
  const proto = location.protocol === "https:" ? "wss" : "ws";
  ws = new WebSocket(`${proto}://${location.host}/websockets` +
                     location.search);
  ws.binaryType = "arraybuffer";

  ws.onclose = () => {
    stateEl.textContent = "reconnecting…";
    for (const row of h264Rows.values())
      if (row.decoder) try { row.decoder.close(); } catch (e) {}
    h264Rows.clear();
    setTimeout(connect, 1000);
  };
  ws.onopen = () => {
    stateEl.textContent = "connected";
    /* replay persisted client preferences (unlocked settings only:
       the server sanitizes every proposal) */
    try {
      const saved = JSON.parse(
          localStorage.getItem("selkies.prefs") || "{}");
      for (const [k, v] of Object.entries(saved)) {
        ws.send("SETTINGS," + JSON.stringify({ [k]: v }));
      }
    } catch (e) {}
  };
  ws.onmessage = (ev) => {
    if (typeof ev.data === "string") {
      const i = ev.data.indexOf(",");
      const verb = i < 0 ? ev.data : ev.data.slice(0, i);
      const rest = i < 0 ? "" : ev.data.slice(i + 1);
      if (verb === "SETTINGS_PAYLOAD") {
        try {
          const payload = JSON.parse(rest);
          applyServerSettings(payload);
          /* persist client-changeable settings across sessions
             (namespaced, quota-safe — reference selkies-core.js:14-40) */
          try {
            localStorage.setItem("selkies.settings",
                                 JSON.stringify(payload));
          } catch (e) { /* quota exceeded / privacy mode: skip */ }
          if (window.skOnSettings) window.skOnSettings(payload);
        } catch (e) {}
      } else if (verb === "SYSTEM_STATS") {
        try {
          if (window.skOnStats) window.skOnStats(JSON.parse(rest));
        } catch (e) {}
      } else if (verb === "RTC_CONFIG") {
        /* hot-reloaded ICE config (server file monitor); used by the
           next RTCPeerConnection */
        try { window.skRtcConfig = JSON.parse(rest); } catch (e) {}
      } else if (ev.data.startsWith("PIPELINE_RESETTING")) {
        /* encoder restarted: drop per-row decoders + frame tracking so
           the next IDR re-primes cleanly */
        for (const row of h264Rows.values())
          if (row.decoder) try { row.decoder.close(); } catch (e) {}
        h264Rows.clear();
        lastAckedFrame = -1;
      } else if (verb === "DISPLAY_CONFIG_UPDATE") {
        try {
          if (window.skOnDisplayConfig)
            window.skOnDisplayConfig(JSON.parse(rest));
        } catch (e) {}
      } else if (verb === "SEAT") {
        /* player-seat grant/deny (reference PlayerGamepadButton) */
        if (window.skOnSeat) window.skOnSeat(parseInt(rest, 10));
      } else if (verb === "CURSOR") {
        try { applyCursor(JSON.parse(rest)); } catch (e) {}
      } else if (verb === "clipboard") {
        try {
          const text = decodeURIComponent(escape(atob(rest)));
          lastRemoteClipboard = text;
          navigator.clipboard.writeText(text).catch(() => {});
          if (window.skOnClipboard) window.skOnClipboard(text);
        } catch (e) {}
      }
    } else {
      onBinary(ev.data);
    }
  };
}

setInterval(() => {
  const now = performance.now();
  const dt = (now - lastStats) / 1000;
  statsEl.textContent =
      `stripes/s ${(frameCount / dt).toFixed(0)}  ` +
      `mbps ${(byteCount * 8 / dt / 1e6).toFixed(2)}`;
  frameCount = 0; byteCount = 0; lastStats = now;
}, 2000);

/* resize -> ask the server to match our window */
let resizeTimer = null;
window.addEventListener("resize", () => {
  clearTimeout(resizeTimer);
  resizeTimer = setTimeout(() => {
    const el = document.getElementById("stage");
    send(`r,${el.clientWidth}x${el.clientHeight}`);
  }, 400);
});

/* ---------------- WebRTC transport (?transport=webrtc) ----------------
 * Video (and audio, G.711) arrive over DTLS-SRTP from the ice-lite
 * endpoint; input rides the SCTP data channel when it opens, with the
 * control WebSocket (display=none) as fallback + settings/clipboard
 * channel. */
let dcInput = null;            /* open RTCDataChannel, used by sendInput */
async function connectWebRTC() {
  let rtcCfg = {};
  try {
    const tr = await fetch("/api/turn" + location.search);
    if (tr.ok) {
      const cfg = await tr.json();
      if (cfg.iceServers) rtcCfg.iceServers = cfg.iceServers;
    }
  } catch (e) { /* no TURN configured: host candidates only */ }
  const pc = new RTCPeerConnection(rtcCfg);
  pc.addTransceiver("video", { direction: "recvonly" });
  pc.addTransceiver("audio", { direction: "recvonly" });
  const dc = pc.createDataChannel("input", { ordered: true });
  dc.onopen = () => { dcInput = dc; };
  dc.onclose = () => { if (dcInput === dc) dcInput = null; };
  pc.ontrack = (ev) => {
    const vid = document.createElement("video");
    vid.autoplay = true; vid.playsInline = true; vid.muted = true;
    vid.srcObject = ev.streams[0] || new MediaStream([ev.track]);
    vid.play().catch(() => {});
    const draw = () => {
      if (vid.videoWidth) {
        ensureCanvas(vid.videoWidth, vid.videoHeight);
        ctx2d.drawImage(vid, 0, 0);
      }
      requestAnimationFrame(draw);
    };
    draw();
    stateEl.textContent = "connected (webrtc)";
  };
  const offer = await pc.createOffer();
  await pc.setLocalDescription(offer);
  await new Promise((res) => {
    if (pc.iceGatheringState === "complete") return res();
    pc.onicegatheringstatechange = () =>
        pc.iceGatheringState === "complete" && res();
    setTimeout(res, 1000);
  });
  const r = await fetch("/api/webrtc/offer" + location.search, {
    method: "POST",
    headers: { "Content-Type": "application/json" },
    body: JSON.stringify({ sdp: pc.localDescription.sdp }),
  });
  const ans = await r.json();
  await pc.setRemoteDescription({ type: "answer", sdp: ans.sdp });
  /* opt-in stats recording: flatten getStats and ship to the server */
  setInterval(async () => {
    try {
      const stats = [];
      (await pc.getStats()).forEach((report) => {
        const row = { id: report.id, type: report.type };
        for (const [k, v] of Object.entries(report)) {
          if (typeof v === "number" || typeof v === "string") row[k] = v;
        }
        stats.push(row);
      });
      if (stats.length)
        fetch("/api/webrtc-stats" + location.search, {
          method: "POST",
          headers: { "Content-Type": "application/json" },
          body: JSON.stringify(stats),
        }).catch(() => {});
    } catch (e) {}
  }, 5000);
}

/* ---- gamepads: physical polling + touch overlay (gamepad.js) ---- */
let touchPad = null;
function hookGamepads() {
  const G = window.SelkiesGamepad;
  if (!G) return;
  const enc = new G.GamepadEncoder(send);
  let polling = false;
  function poll() {
    G.pollPhysical(enc, navigator.getGamepads ? navigator.getGamepads() : []);
    if (polling) requestAnimationFrame(poll);
  }
  window.addEventListener("gamepadconnected", () => {
    if (!polling) { polling = true; poll(); }
  });
  window.addEventListener("gamepaddisconnected", () => {
    /* keep polling; encoder emits js,d when the slot empties */
  });
  /* touch overlay: ?touch=1 forces it on; dashboard can toggle.
     Touch pads claim index 3 so they never fight a physical pad. */
  window.skToggleTouchGamepad = (profile) => {
    if (touchPad) { touchPad.destroy(); touchPad = null; return false; }
    touchPad = G.attachOverlay(document.body, send,
                               profile || localStorage.getItem(
                                   "selkies.touchProfile") || "modern", 3);
    return true;
  };
  if (new URLSearchParams(location.search).get("touch") === "1")
    window.skToggleTouchGamepad();
}

hookInput();
hookHud();
hookGamepads();
const params = new URLSearchParams(location.search);
if (params.get("transport") === "webrtc") {
  connectWebRTC().catch((e) => {
    stateEl.textContent = "webrtc failed: " + e;
  });
  /* control-only WS (no video fan-out) */
  params.set("display", "none");
  history.replaceState(null, "", "?" + params.toString());
}
connect();
canvas.focus();

/* ---- dashboard postMessage bridge (contract:
   selkies_amd/web/postmessage-bridge.js) ---- */
let lastServerStats = null, lastServerSettings = null;
if (window.installPostMessageBridge) {
  const bridge = window.installPostMessageBridge({
    send: send,
    b64encode: (t) => btoa(unescape(encodeURIComponent(t))),
    getStats: () => ({
      clientFps: Number(
          (frameCount / Math.max(0.001,
              (performance.now() - lastStats) / 1000)).toFixed(1)),
      videoBuffer: h264Rows.size,
      audioBuffer: 0,
      encoderName: (lastServerSettings &&
                    lastServerSettings.encoder &&
                    lastServerSettings.encoder.value) || "h264enc-striped",
      server: lastServerStats,
      isVideoPipelineActive: !!(ws && ws.readyState === 1),
      isAudioPipelineActive: !!audioCtx,
      isMicrophoneActive: false,
    }),
    setVolume: (v) => {
      masterVolume = Math.max(0, Math.min(1, v));
      if (gainNode && !audioMuted) gainNode.gain.value = masterVolume;
    },
    setMute: (m) => {
      audioMuted = m;
      if (gainNode) gainNode.gain.value = m ? 0 : masterVolume;
    },
    setGamepadEnabled: (on) => { window.skGamepadEnabled = on; },
    resizeToWindow: () => {
      const el = document.getElementById("stage");
      send(`r,${el.clientWidth}x${el.clientHeight}`);
    },
    showVirtualKeyboard: () => canvas.focus(),
    setRenderFlag: (k, v) => {
      if (k === "setAntiAliasing")
        canvas.style.imageRendering = v ? "auto" : "pixelated";
    },
  }, window);
  const prevSettings = window.skOnSettings;
  window.skOnSettings = (p) => {
    lastServerSettings = p;
    bridge.onServerSettings(p);
    if (prevSettings) prevSettings(p);
  };
  const prevStats = window.skOnStats;
  window.skOnStats = (st) => {
    lastServerStats = st;
    bridge.onStats(st);
    if (prevStats) prevStats(st);
  };
  const prevClip = window.skOnClipboard;
  window.skOnClipboard = (t) => {
    bridge.onClipboard(t);
    if (prevClip) prevClip(t);
  };
}
