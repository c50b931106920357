// timeline.js — render every series in report.js (`sofa_traces`) on the
// shared canvas plotter.  Parity with reference sofaboard/timeline.js:1-64
// (Highcharts scatter, log y, zoom) without the CDN dependency.

"use strict";

(function () {
  var plot = new SofaPlot("timeline", { logy: true, tooltipId: "tooltip", xlabel: "time since record start (s)" });
  if (typeof sofa_traces === "undefined") {
    document.getElementById("legend").textContent =
      "report.js not found — run `sofa preprocess` first";
    return;
  }
  sofa_traces.forEach(function (t, i) {
    if (!t || !t.data) return;
    var color = t.color || SOFA_COLORS[i % SOFA_COLORS.length];
    plot.addSeries(t.name, color, t.data, "scatter");
  });
  sofaLegend("legend", plot);
  plot.draw();
})();
