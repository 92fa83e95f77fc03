/* Gamepad support: physical pad polling + universal touch overlay.
 *
 * Reference parity: addons/universal-touch-gamepad/universalTouchGamepad.js
 * (on-screen touch gamepad: sticks with tap->L3/R3, d-pad, face buttons,
 * shoulders/triggers, profile layouts) and addons/selkies-web-core/lib/
 * gamepad.js (navigator.getGamepads polling). Fresh implementation: a
 * pure-logic core (encoder, layout geometry, touch state machine) that
 * runs under node for tests, plus a thin DOM layer for the browser.
 *
 * Wire protocol (server: selkies_amd/gamepad.py GamepadHub):
 *   js,c,<idx>,<b64 name>,<numButtons>,<numAxes>   connect
 *   js,d,<idx>                                     disconnect
 *   js,b,<idx>,<button>,<0|1>                      button state
 *   js,a,<idx>,<axis>,<value>                      axis value [-1..1]
 *
 * Standard-mapping indices: 0..3 face (A B X Y), 4/5 L1/R1, 6/7 L2/R2,
 * 8 select, 9 start, 10/11 L3/R3, 12..15 dpad U D L R, 16 home.
 */
(function (root, factory) {
  if (typeof module === "object" && module.exports) module.exports = factory();
  else root.SelkiesGamepad = factory();
})(typeof self !== "undefined" ? self : this, function () {
  "use strict";

  var MAX_BUTTONS = 18;
  var MAX_AXES = 4;
  var AXIS_EPS = 0.01;        // resend threshold
  var DEADZONE = 0.05;
  var TAP_MS = 250;           // stick tap -> L3/R3
  var TAP_MOVE_FRACTION = 0.25;
  var STICK_PRESS_MS = 60;

  function b64(s) {
    if (typeof btoa === "function") return btoa(s);
    return Buffer.from(s, "utf-8").toString("base64");
  }

  // ---- wire encoder with change detection ---------------------------------
  function GamepadEncoder(send) {
    this.send = send;
    this.pads = {};             // idx -> {buttons: [], axes: []}
  }
  GamepadEncoder.prototype.connect = function (idx, name, nb, na) {
    if (this.pads[idx]) return;
    nb = nb || MAX_BUTTONS;
    na = na || MAX_AXES;
    this.pads[idx] = { buttons: [], axes: [] };
    this.send("js,c," + idx + "," + b64(name || "Selkies Touch Gamepad") +
              "," + nb + "," + na);
  };
  GamepadEncoder.prototype.disconnect = function (idx) {
    if (!this.pads[idx]) return;
    delete this.pads[idx];
    this.send("js,d," + idx);
  };
  GamepadEncoder.prototype.button = function (idx, n, pressed) {
    var p = this.pads[idx];
    if (!p) return;
    var v = pressed ? 1 : 0;
    if (p.buttons[n] === v) return;
    p.buttons[n] = v;
    this.send("js,b," + idx + "," + n + "," + v);
  };
  GamepadEncoder.prototype.axis = function (idx, n, value) {
    var p = this.pads[idx];
    if (!p) return;
    if (Math.abs(value) < DEADZONE) value = 0;
    value = Math.max(-1, Math.min(1, value));
    var q = Math.round(value * 1000) / 1000;
    if (p.axes[n] !== undefined && Math.abs(p.axes[n] - q) < AXIS_EPS &&
        !(q === 0 && p.axes[n] !== 0)) return;
    p.axes[n] = q;
    this.send("js,a," + idx + "," + n + "," + q);
  };

  // ---- physical pads: navigator.getGamepads() -> encoder ------------------
  // `pads` is the array from getGamepads() (entries may be null).
  function pollPhysical(enc, pads) {
    for (var i = 0; i < pads.length && i < 4; i++) {
      var gp = pads[i];
      if (!gp || !gp.connected) {
        enc.disconnect(i);
        continue;
      }
      enc.connect(i, gp.id || "Gamepad", gp.buttons.length, gp.axes.length);
      for (var b = 0; b < gp.buttons.length && b < MAX_BUTTONS; b++) {
        var bt = gp.buttons[b];
        enc.button(i, b, typeof bt === "object" ? bt.pressed : bt > 0.5);
      }
      for (var a = 0; a < gp.axes.length && a < MAX_AXES; a++)
        enc.axis(i, a, gp.axes[a]);
    }
  }

  // ---- layout geometry (normalized viewport coords, 0..1) -----------------
  // Controls: {id, kind: stick|dpad|button, cx, cy, r, button?, axes?}
  // r is a fraction of min(viewW, viewH).
  function layoutControls(profile) {
    var face = function (cx, cy, r) {
      // diamond: A bottom(0) B right(1) X left(2) Y top(3)
      return [
        { id: "A", kind: "button", button: 0, cx: cx, cy: cy + r, r: 0.055 },
        { id: "B", kind: "button", button: 1, cx: cx + r, cy: cy, r: 0.055 },
        { id: "X", kind: "button", button: 2, cx: cx - r, cy: cy, r: 0.055 },
        { id: "Y", kind: "button", button: 3, cx: cx, cy: cy - r, r: 0.055 },
      ];
    };
    var common = [
      { id: "L1", kind: "button", button: 4, cx: 0.07, cy: 0.16, r: 0.06 },
      { id: "R1", kind: "button", button: 5, cx: 0.93, cy: 0.16, r: 0.06 },
      { id: "L2", kind: "button", button: 6, cx: 0.07, cy: 0.04, r: 0.06 },
      { id: "R2", kind: "button", button: 7, cx: 0.93, cy: 0.04, r: 0.06 },
      { id: "SELECT", kind: "button", button: 8, cx: 0.42, cy: 0.06, r: 0.045 },
      { id: "START", kind: "button", button: 9, cx: 0.58, cy: 0.06, r: 0.045 },
      { id: "HOME", kind: "button", button: 16, cx: 0.50, cy: 0.06, r: 0.04 },
    ];
    if (profile === "classic") {
      // d-pad left, face cluster right; no sticks
      return common.concat([
        { id: "DPAD", kind: "dpad", cx: 0.16, cy: 0.62, r: 0.16 },
      ], face(0.84, 0.62, 0.10));
    }
    // "modern": dual sticks low, d-pad + face above them
    return common.concat([
      { id: "LS", kind: "stick", axes: [0, 1], tapButton: 10,
        cx: 0.17, cy: 0.74, r: 0.12 },
      { id: "RS", kind: "stick", axes: [2, 3], tapButton: 11,
        cx: 0.83, cy: 0.74, r: 0.12 },
      { id: "DPAD", kind: "dpad", cx: 0.13, cy: 0.40, r: 0.13 },
    ], face(0.87, 0.40, 0.085));
  }

  // Hit test in normalized coords; slop widens every control a little.
  function hitTest(controls, x, y, aspect, slop) {
    slop = slop || 0.02;
    var best = null, bestD = 1e9;
    for (var i = 0; i < controls.length; i++) {
      var c = controls[i];
      // aspect-correct distance: x spans `aspect` units when y spans 1
      var dx = (x - c.cx) * aspect, dy = y - c.cy;
      var d = Math.sqrt(dx * dx + dy * dy);
      if (d <= c.r * (c.kind === "button" ? 1 : 1.15) + slop && d < bestD) {
        best = c;
        bestD = d;
      }
    }
    return best;
  }

  // Stick displacement -> axes pair, clamped to the unit circle.
  function stickValue(c, x, y, aspect) {
    var dx = (x - c.cx) * aspect / c.r, dy = (y - c.cy) / c.r;
    var m = Math.sqrt(dx * dx + dy * dy);
    if (m > 1) { dx /= m; dy /= m; }
    return [dx, dy];
  }

  // D-pad touch position -> pressed direction buttons (12 U,13 D,14 L,15 R).
  // 8-way: diagonals press two.
  function dpadButtons(c, x, y, aspect) {
    var dx = (x - c.cx) * aspect, dy = y - c.cy;
    var m = Math.sqrt(dx * dx + dy * dy);
    var out = [];
    if (m < c.r * 0.25) return out;       // center dead zone
    var ang = Math.atan2(dy, dx);         // 0 = right, pi/2 = down
    var deg = ang * 180 / Math.PI;
    if (deg > -157.5 && deg < -22.5) out.push(12);   // up
    if (deg > 22.5 && deg < 157.5) out.push(13);     // down
    if (deg > 112.5 || deg < -112.5) out.push(14);   // left
    if (deg > -67.5 && deg < 67.5) out.push(15);     // right
    return out;
  }

  // ---- touch state machine -------------------------------------------------
  // Feed normalized touch events; emits encoder calls on pad `idx`.
  // opts: {now: ()->ms, setTimeout: fn, aspect: w/h}
  function TouchGamepadCore(enc, idx, profile, opts) {
    opts = opts || {};
    this.enc = enc;
    this.idx = idx;
    this.controls = layoutControls(profile || "modern");
    this.aspect = opts.aspect || (16 / 9);
    this.now = opts.now || function () { return Date.now(); };
    this.setTimeout = opts.setTimeout ||
      (typeof setTimeout === "function" ? setTimeout : null);
    this.touches = {};          // touchId -> state
    enc.connect(idx, "Selkies Touch Gamepad", MAX_BUTTONS, MAX_AXES);
  }
  TouchGamepadCore.prototype._applyDpad = function (st, dirs) {
    var have = {};
    var i;
    for (i = 0; i < dirs.length; i++) have[dirs[i]] = true;
    for (i = 12; i <= 15; i++)
      this.enc.button(this.idx, i, !!have[i]);
    st.dpadDirs = dirs;
  };
  TouchGamepadCore.prototype.touchStart = function (id, x, y) {
    var c = hitTest(this.controls, x, y, this.aspect);
    if (!c) return false;
    var st = { control: c, x0: x, y0: y, t0: this.now(), maxDist: 0 };
    this.touches[id] = st;
    if (c.kind === "button") this.enc.button(this.idx, c.button, true);
    else if (c.kind === "stick") {
      var v = stickValue(c, x, y, this.aspect);
      this.enc.axis(this.idx, c.axes[0], v[0]);
      this.enc.axis(this.idx, c.axes[1], v[1]);
    } else if (c.kind === "dpad")
      this._applyDpad(st, dpadButtons(c, x, y, this.aspect));
    return true;
  };
  TouchGamepadCore.prototype.touchMove = function (id, x, y) {
    var st = this.touches[id];
    if (!st) return;
    var c = st.control;
    var dx = (x - st.x0) * this.aspect, dy = y - st.y0;
    st.maxDist = Math.max(st.maxDist, Math.sqrt(dx * dx + dy * dy));
    if (c.kind === "stick") {
      var v = stickValue(c, x, y, this.aspect);
      this.enc.axis(this.idx, c.axes[0], v[0]);
      this.enc.axis(this.idx, c.axes[1], v[1]);
    } else if (c.kind === "dpad")
      this._applyDpad(st, dpadButtons(c, x, y, this.aspect));
  };
  TouchGamepadCore.prototype.touchEnd = function (id) {
    var st = this.touches[id];
    if (!st) return;
    delete this.touches[id];
    var c = st.control, enc = this.enc, idx = this.idx;
    if (c.kind === "button") enc.button(idx, c.button, false);
    else if (c.kind === "stick") {
      enc.axis(idx, c.axes[0], 0);
      enc.axis(idx, c.axes[1], 0);
      // quick tap with little travel = stick click (L3/R3)
      if (this.now() - st.t0 < TAP_MS &&
          st.maxDist < c.r * TAP_MOVE_FRACTION) {
        enc.button(idx, c.tapButton, true);
        if (this.setTimeout)
          this.setTimeout(function () { enc.button(idx, c.tapButton, false); },
                          STICK_PRESS_MS);
        else enc.button(idx, c.tapButton, false);
      }
    } else if (c.kind === "dpad") this._applyDpad(st, []);
  };
  TouchGamepadCore.prototype.detach = function () {
    for (var id in this.touches) this.touchEnd(id);
    this.enc.disconnect(this.idx);
  };

  // ---- DOM overlay (browser only) -----------------------------------------
  function attachOverlay(container, sendFn, profile, padIndex) {
    var doc = container.ownerDocument;
    var overlay = doc.createElement("div");
    overlay.className = "selkies-touch-gamepad";
    overlay.style.cssText =
      "position:absolute;inset:0;z-index:40;touch-action:none;" +
      "user-select:none;-webkit-user-select:none;";
    var enc = new GamepadEncoder(sendFn);
    var core = new TouchGamepadCore(enc, padIndex || 0, profile, {
      aspect: Math.max(0.5, container.clientWidth /
                       Math.max(1, container.clientHeight)),
    });
    // faint visual hints for each control
    core.controls.forEach(function (c) {
      var el = doc.createElement("div");
      var rpct = c.r * 100;
      el.style.cssText =
        "position:absolute;border:1.5px solid rgba(255,255,255,.35);" +
        "border-radius:50%;background:rgba(255,255,255,.08);color:#fff;" +
        "display:flex;align-items:center;justify-content:center;" +
        "font:600 11px sans-serif;opacity:.7;pointer-events:none;";
      var d = rpct * 2;
      el.style.width = d + "vmin";
      el.style.height = d + "vmin";
      el.style.left = "calc(" + c.cx * 100 + "% - " + rpct + "vmin)";
      el.style.top = "calc(" + c.cy * 100 + "% - " + rpct + "vmin)";
      el.textContent = c.id;
      overlay.appendChild(el);
    });
    function norm(ev) {
      var r = overlay.getBoundingClientRect();
      return [(ev.clientX - r.left) / r.width,
              (ev.clientY - r.top) / r.height];
    }
    overlay.addEventListener("pointerdown", function (ev) {
      var p = norm(ev);
      if (core.touchStart(ev.pointerId, p[0], p[1])) {
        overlay.setPointerCapture(ev.pointerId);
        ev.preventDefault();
      }
    });
    overlay.addEventListener("pointermove", function (ev) {
      var p = norm(ev);
      core.touchMove(ev.pointerId, p[0], p[1]);
    });
    function up(ev) { core.touchEnd(ev.pointerId); }
    overlay.addEventListener("pointerup", up);
    overlay.addEventListener("pointercancel", up);
    container.appendChild(overlay);
    return {
      element: overlay,
      core: core,
      destroy: function () {
        core.detach();
        if (overlay.parentNode) overlay.parentNode.removeChild(overlay);
      },
    };
  }

  return {
    GamepadEncoder: GamepadEncoder,
    pollPhysical: pollPhysical,
    layoutControls: layoutControls,
    hitTest: hitTest,
    stickValue: stickValue,
    dpadButtons: dpadButtons,
    TouchGamepadCore: TouchGamepadCore,
    attachOverlay: attachOverlay,
    MAX_BUTTONS: MAX_BUTTONS,
    MAX_AXES: MAX_AXES,
  };
});
