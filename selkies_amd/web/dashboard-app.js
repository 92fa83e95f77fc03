/* Standalone dashboard driving the embedded core client EXCLUSIVELY via
 * the window-messaging contract (postmessage-bridge.js; surveyed from
 * reference addons/selkies-web-core/README.md:44-170 and the
 * selkies-dashboard addon's role). */
"use strict";

(function () {
  const frame = document.getElementById("frame");
  const origin = window.location.origin;
  const post = (msg) => frame.contentWindow.postMessage(msg, origin);
  const $ = (id) => document.getElementById(id);

  $("d-apply").onclick = () => post({
    type: "settings",
    settings: {
      encoder: $("d-encoder").value,
      framerate: Number($("d-fps").value),
      rate_control_mode: $("d-rc").value,
      video_crf: Number($("d-crf").value),
      video_bitrate: Number($("d-br").value),
      video_deblock: $("d-deblock").checked,
    },
  });
  $("d-setres").onclick = () => post({
    type: "setManualResolution",
    width: Number($("d-w").value), height: Number($("d-h").value),
  });
  $("d-fitres").onclick = () => post({ type: "resetResolutionToWindow" });
  $("d-vol").oninput = () =>
      post({ type: "setVolume", value: Number($("d-vol").value) / 100 });
  $("d-mute").onchange = () =>
      post({ type: "setMute", value: $("d-mute").checked });

  for (const [id, pipeline] of [["d-video", "video"], ["d-audio", "audio"]]) {
    $(id).onclick = () => {
      const on = $(id).dataset.on !== "1";
      $(id).dataset.on = on ? "1" : "0";
      $(id).textContent = `${pipeline}: ${on ? "on" : "off"}`;
      post({ type: "pipelineControl", pipeline, enabled: on });
    };
  }
  $("d-gamepad").onclick = () => {
    const on = $("d-gamepad").dataset.on !== "1";
    $("d-gamepad").dataset.on = on ? "1" : "0";
    $("d-gamepad").textContent = `gamepad: ${on ? "on" : "off"}`;
    post({ type: "gamepadControl", enabled: on });
  };
  $("d-clipsend").onclick = () =>
      post({ type: "clipboardUpdateFromUI", text: $("d-clip").value });
  $("d-share-view").onclick = () => navigator.clipboard.writeText(
      `${origin}/?role=viewer`).catch(() => {});
  $("d-share-ctl").onclick = () => navigator.clipboard.writeText(
      `${origin}/?role=controller`).catch(() => {});

  /* client -> dashboard */
  window.addEventListener("message", (ev) => {
    if (ev.origin !== origin) return;
    const m = ev.data;
    if (!m || !m.type) return;
    if (m.type === "stats") {
      $("stats").textContent = JSON.stringify(m.data, null, 1)
          .replace(/[{}",]/g, "");
    } else if (m.type === "clipboardContentUpdate") {
      $("d-clip").value = m.text;
    } else if (m.type === "serverSettings") {
      const enc = m.payload && m.payload.encoder;
      if (enc && enc.value) $("d-encoder").value = enc.value;
    } else if (m.type === "pipelineStatusUpdate") {
      for (const p of ["video", "audio", "gamepad"]) {
        if (m[p] !== undefined) {
          const b = $(`d-${p}`);
          b.dataset.on = m[p] ? "1" : "0";
          b.textContent = `${p}: ${m[p] ? "on" : "off"}`;
        }
      }
    }
  });

  /* poll stats only while visible (contract note: the dashboard side
     stops polling when closed) */
  post({ type: "sidebarVisibilityChanged", isOpen: true });
  setInterval(() => {
    if (!document.hidden) post({ type: "getStats" });
  }, 2000);
})();
