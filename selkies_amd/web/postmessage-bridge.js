/* Window-messaging (postMessage) API between an embedding dashboard and
 * the core client — the contract surveyed from the reference
 * (addons/selkies-web-core/README.md:44-170): same-origin only,
 * `{type: ...}` objects in, telemetry/state objects out.
 *
 * Dashboard -> client: settings, setManualResolution,
 * resetResolutionToWindow, setScaleLocally, setUseCssScaling,
 * setAntiAliasing, pipelineControl, setVolume, setMute, gamepadControl,
 * showVirtualKeyboard, clipboardUpdateFromUI, command, getStats,
 * sidebarVisibilityChanged.
 * Client -> dashboard: stats, serverSettings, clipboardContentUpdate,
 * pipelineStatusUpdate, sidebarButtonStatusUpdate, clientRoleUpdate.
 *
 * Factored out of the client so the contract is testable under node
 * (tests/test_dashboard_api.py) with a stub core/window. */
"use strict";

function installPostMessageBridge(core, win) {
  const origin = win.location.origin;
  const post = (msg) => {
    if (win.parent && win.parent !== win) {
      win.parent.postMessage(msg, origin);
    }
  };

  const state = {
    sidebarOpen: false,
    pipelines: { video: true, audio: true, microphone: false,
                 gamepad: true },
  };

  const postPipelines = () => {
    post(Object.assign({ type: "pipelineStatusUpdate" }, state.pipelines));
    post(Object.assign({ type: "sidebarButtonStatusUpdate" },
                       state.pipelines));
  };

  const SETTING_KEYS = [
    "framerate", "rate_control_mode", "video_bitrate", "video_crf",
    "encoder", "audio_bitrate", "scaling_dpi", "use_cpu",
    "video_fullcolor", "video_streaming_mode", "jpeg_quality",
    "use_paint_over_quality", "video_deblock",
  ];

  win.addEventListener("message", (ev) => {
    if (ev.origin !== origin) return;            /* same-origin only */
    const m = ev.data;
    if (!m || typeof m !== "object" || !m.type) return;
    switch (m.type) {
      case "settings": {
        const out = {};
        for (const k of SETTING_KEYS) {
          if (m.settings && m.settings[k] !== undefined) {
            out[k === "video_bitrate" ? "video_bitrate_kbps" : k] =
                m.settings[k];
          }
        }
        if (Object.keys(out).length) {
          core.send("SETTINGS," + JSON.stringify(out));
        }
        if (m.settings && m.settings.scaling_dpi !== undefined) {
          core.send("s," + m.settings.scaling_dpi);
        }
        break;
      }
      case "setManualResolution":
        core.send("r," + m.width + "x" + m.height);
        if (core.setManualResolution) core.setManualResolution(true);
        break;
      case "resetResolutionToWindow":
        if (core.resizeToWindow) core.resizeToWindow();
        break;
      case "setScaleLocally":
      case "setUseCssScaling":
      case "setAntiAliasing":
        if (core.setRenderFlag) core.setRenderFlag(m.type, m.value);
        break;
      case "pipelineControl":
        if (m.pipeline === "video") {
          core.send(m.enabled ? "START_VIDEO" : "STOP_VIDEO");
        } else if (m.pipeline === "audio") {
          core.send(m.enabled ? "START_AUDIO" : "STOP_AUDIO");
        } else if (m.pipeline === "microphone" && core.setMicrophone) {
          core.setMicrophone(!!m.enabled);
        }
        state.pipelines[m.pipeline] = !!m.enabled;
        postPipelines();
        break;
      case "setVolume":
        if (core.setVolume) core.setVolume(m.value);
        break;
      case "setMute":
        if (core.setMute) core.setMute(!!m.value);
        break;
      case "gamepadControl":
        if (core.setGamepadEnabled) core.setGamepadEnabled(!!m.enabled);
        state.pipelines.gamepad = !!m.enabled;
        postPipelines();
        break;
      case "showVirtualKeyboard":
        if (core.showVirtualKeyboard) core.showVirtualKeyboard();
        break;
      case "clipboardUpdateFromUI": {
        const b64 = core.b64encode
            ? core.b64encode(m.text)
            : Buffer.from(String(m.text), "utf-8").toString("base64");
        core.send("cw," + b64);
        break;
      }
      case "command":
        core.send("cmd," + m.value);
        break;
      case "getStats":
        post({ type: "stats", data: core.getStats() });
        break;
      case "sidebarVisibilityChanged":
        state.sidebarOpen = !!m.isOpen;
        break;
      default:
        break;
    }
  });

  return {
    post: post,
    state: state,
    /* core-side notifications -> dashboard */
    onServerSettings: (payload) =>
        post({ type: "serverSettings", payload: payload }),
    onClipboard: (text) =>
        post({ type: "clipboardContentUpdate", text: text }),
    onRole: (role) => post({ type: "clientRoleUpdate", role: role }),
    onStats: (data) => {
      if (state.sidebarOpen) post({ type: "stats", data: data });
    },
    postPipelines: postPipelines,
  };
}

/* browser global + node export */
if (typeof window !== "undefined") {
  window.installPostMessageBridge = installPostMessageBridge;
}
if (typeof module !== "undefined" && module.exports) {
  module.exports = { installPostMessageBridge };
}
