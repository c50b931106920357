// sofa.js — shared helpers for the sofaboard pages: CSV loading and a
// self-contained canvas scatter/line plotter (no CDN dependencies — the
// reference's Highcharts/d3/Plotly pages break on air-gapped clusters).

"use strict";

function sofaFetchText(path) {
  return fetch(path).then(function (r) {
    if (!r.ok) throw new Error(path + ": " + r.status);
    return r.text();
  });
}

function sofaParseCSV(text) {
  // minimal CSV with quoted-field support
  var rows = [];
  var row = [];
  var cur = "";
  var inq = false;
  for (var i = 0; i < text.length; i++) {
    var c = text[i];
    if (inq) {
      if (c === '"') {
        if (text[i + 1] === '"') { cur += '"'; i++; }
        else inq = false;
      } else cur += c;
    } else if (c === '"') inq = true;
    else if (c === ",") { row.push(cur); cur = ""; }
    else if (c === "\n") { row.push(cur); cur = ""; rows.push(row); row = []; }
    else if (c !== "\r") cur += c;
  }
  if (cur.length || row.length) { row.push(cur); rows.push(row); }
  if (!rows.length) return { header: [], rows: [] };
  var header = rows[0];
  return { header: header, rows: rows.slice(1).filter(function (r) { return r.length === header.length; }) };
}

function sofaCSVObjects(text) {
  var p = sofaParseCSV(text);
  return p.rows.map(function (r) {
    var o = {};
    p.header.forEach(function (h, i) { o[h] = r[i]; });
    return o;
  });
}

var SOFA_COLORS = [
  "#4363d8", "#e6194b", "#3cb44b", "#ffa500", "#911eb4", "#46f0f0",
  "#f032e6", "#9a6324", "#800000", "#808000", "#008080", "#000075",
  "#f58231", "#bcf60c", "#fabebe", "#e6beff",
];

// ---------------------------------------------------------------- plotter

function SofaPlot(canvasId, opts) {
  this.canvas = document.getElementById(canvasId);
  this.ctx = this.canvas.getContext("2d");
  this.series = [];
  this.opts = opts || {};
  this.logy = !!this.opts.logy;
  this.margin = { l: 70, r: 20, t: 10, b: 40 };
  this.zoom = null; // {x0,x1}
  this.tooltip = null;
  this._bindEvents();
}

SofaPlot.prototype.addSeries = function (name, color, points, mode) {
  this.series.push({ name: name, color: color, points: points, mode: mode || "scatter", visible: true });
};

SofaPlot.prototype._dataRange = function () {
  var x0 = Infinity, x1 = -Infinity, y0 = Infinity, y1 = -Infinity;
  this.series.forEach(function (s) {
    if (!s.visible) return;
    s.points.forEach(function (p) {
      if (!isFinite(p.x) || !isFinite(p.y)) return;
      if (p.x < x0) x0 = p.x;
      if (p.x > x1) x1 = p.x;
      if (p.y < y0) y0 = p.y;
      if (p.y > y1) y1 = p.y;
    });
  });
  if (!isFinite(x0)) { x0 = 0; x1 = 1; y0 = 0; y1 = 1; }
  if (x0 === x1) x1 = x0 + 1;
  if (y0 === y1) y1 = y0 + 1;
  return { x0: x0, x1: x1, y0: y0, y1: y1 };
};

SofaPlot.prototype._yval = function (y) {
  return this.logy ? Math.log10(Math.max(y, 1e-12)) : y;
};

SofaPlot.prototype.draw = function () {
  var ctx = this.ctx, W = this.canvas.width, H = this.canvas.height;
  var m = this.margin;
  ctx.clearRect(0, 0, W, H);
  var r = this._dataRange();
  if (this.zoom) { r.x0 = this.zoom.x0; r.x1 = this.zoom.x1; }
  var self = this;
  var ly0 = this._yval(r.y0), ly1 = this._yval(r.y1);
  if (ly0 === ly1) ly1 = ly0 + 1;
  var xs = function (x) { return m.l + (x - r.x0) / (r.x1 - r.x0) * (W - m.l - m.r); };
  var ys = function (y) { return H - m.b - (self._yval(y) - ly0) / (ly1 - ly0) * (H - m.t - m.b); };
  this._xs = xs; this._ys = ys; this._range = r;

  // axes
  ctx.strokeStyle = "#888"; ctx.lineWidth = 1;
  ctx.strokeRect(m.l, m.t, W - m.l - m.r, H - m.t - m.b);
  ctx.fillStyle = "#444"; ctx.font = "11px sans-serif";
  for (var i = 0; i <= 8; i++) {
    var xv = r.x0 + (r.x1 - r.x0) * i / 8;
    var xp = xs(xv);
    ctx.fillText(xv.toFixed(2), xp - 12, H - m.b + 14);
    ctx.strokeStyle = "#eee";
    ctx.beginPath(); ctx.moveTo(xp, m.t); ctx.lineTo(xp, H - m.b); ctx.stroke();
  }
  for (var j = 0; j <= 6; j++) {
    var lv = ly0 + (ly1 - ly0) * j / 6;
    var yv = this.logy ? Math.pow(10, lv) : lv;
    var yp = H - m.b - (lv - ly0) / (ly1 - ly0) * (H - m.t - m.b);
    ctx.fillStyle = "#444";
    ctx.fillText(yv.toExponential(1), 4, yp + 3);
    ctx.strokeStyle = "#eee";
    ctx.beginPath(); ctx.moveTo(m.l, yp); ctx.lineTo(W - m.r, yp); ctx.stroke();
  }
  ctx.fillStyle = "#444";
  ctx.fillText(this.opts.xlabel || "time (s)", W / 2 - 20, H - 6);

  // cross-page time cursor (set by click on any page; shared via
  // localStorage so every open sofaboard page shows the same instant)
  var cur = parseFloat(localStorage.getItem("sofa_cursor_t"));
  if (isFinite(cur) && cur >= r.x0 && cur <= r.x1) {
    ctx.strokeStyle = "#d22"; ctx.lineWidth = 1.5;
    ctx.beginPath(); ctx.moveTo(xs(cur), m.t); ctx.lineTo(xs(cur), H - m.b); ctx.stroke();
    ctx.fillStyle = "#d22";
    ctx.fillText("t=" + cur.toFixed(4), xs(cur) + 4, m.t + 12);
    ctx.lineWidth = 1;
  }

  // series
  this.series.forEach(function (s) {
    if (!s.visible) return;
    ctx.fillStyle = s.color; ctx.strokeStyle = s.color;
    if (s.mode === "line") {
      ctx.beginPath();
      var started = false;
      s.points.forEach(function (p) {
        if (p.x < r.x0 || p.x > r.x1) return;
        var xp = xs(p.x), yp = ys(p.y);
        if (!started) { ctx.moveTo(xp, yp); started = true; } else ctx.lineTo(xp, yp);
      });
      ctx.stroke();
    } else {
      s.points.forEach(function (p) {
        if (p.x < r.x0 || p.x > r.x1) return;
        ctx.fillRect(xs(p.x) - 1.5, ys(p.y) - 1.5, 3, 3);
      });
    }
  });
};

SofaPlot.prototype._bindEvents = function () {
  var self = this;
  var dragStart = null;
  this.canvas.addEventListener("mousedown", function (e) { dragStart = e.offsetX; });
  this.canvas.addEventListener("mouseup", function (e) {
    if (dragStart === null) return;
    var a = dragStart, b = e.offsetX;
    dragStart = null;
    if (Math.abs(a - b) < 8) {
      // plain click: place the shared time cursor at this instant
      var rr = self._range;
      var t = rr.x0 + (b - self.margin.l) /
        (self.canvas.width - self.margin.l - self.margin.r) * (rr.x1 - rr.x0);
      if (isFinite(t)) {
        localStorage.setItem("sofa_cursor_t", String(t));
        self.draw();
      }
      return;
    }
    var r = self._range;
    var inv = function (px) {
      return r.x0 + (px - self.margin.l) / (self.canvas.width - self.margin.l - self.margin.r) * (r.x1 - r.x0);
    };
    self.zoom = { x0: inv(Math.min(a, b)), x1: inv(Math.max(a, b)) };
    self.draw();
  });
  this.canvas.addEventListener("dblclick", function () {
    self.zoom = null;
    localStorage.removeItem("sofa_cursor_t");
    self.draw();
  });
  // other pages moved the cursor -> redraw live
  window.addEventListener("storage", function (ev) {
    if (ev.key === "sofa_cursor_t") self.draw();
  });
  this.canvas.addEventListener("mousemove", function (e) {
    var tip = document.getElementById(self.opts.tooltipId || "");
    if (!tip || !self._xs) return;
    // nearest point
    var best = null, bestd = 400;
    self.series.forEach(function (s) {
      if (!s.visible) return;
      s.points.forEach(function (p) {
        var dx = self._xs(p.x) - e.offsetX, dy = self._ys(p.y) - e.offsetY;
        var d = dx * dx + dy * dy;
        if (d < bestd) { bestd = d; best = { s: s, p: p }; }
      });
    });
    if (best) {
      tip.textContent = "[" + best.s.name + "] x=" + best.p.x.toFixed(4) +
        " y=" + best.p.y.toPrecision(4) + (best.p.name ? "  " + best.p.name : "");
    }
  });
};

function sofaLegend(containerId, plot) {
  var el = document.getElementById(containerId);
  el.innerHTML = "";
  plot.series.forEach(function (s) {
    var span = document.createElement("span");
    span.style.cssText = "margin-right:12px;cursor:pointer;user-select:none;";
    span.innerHTML = '<span style="color:' + s.color + ';">&#9632;</span> ' + s.name +
      " (" + s.points.length + ")";
    span.onclick = function () {
      s.visible = !s.visible;
      span.style.opacity = s.visible ? 1.0 : 0.35;
      plot.draw();
    };
    el.appendChild(span);
  });
}
