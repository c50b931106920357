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

  // summary bar from the analyzer's feature vector
  sofaFetchText("features.csv").then(function (t) {
    var f = {};
    sofaCSVObjects(t).forEach(function (r) { f[r.name] = parseFloat(r.value); });
    var parts = [];
    if (f.elapsed_time) parts.push("elapsed " + f.elapsed_time.toFixed(2) + " s");
    if (f.n_gpu_events) parts.push(f.n_gpu_events.toLocaleString() + " GPU events");
    if (f.n_cpu_samples) parts.push(f.n_cpu_samples.toLocaleString() + " CPU samples");
    if (f.gpu_time) parts.push("GPU busy " + f.gpu_time.toFixed(2) + " s");
    if (f.rccl_time) parts.push("RCCL " + f.rccl_time.toFixed(3) + " s");
    if (f.iter_step_time) parts.push("step " + (1e3 * f.iter_step_time).toFixed(1) + " ms (AISI)");
    if (f.launch_latency_us_p50) parts.push("launch p50 " + f.launch_latency_us_p50.toFixed(1) + " µs");
    document.getElementById("summary").textContent = parts.join("  ·  ");
  }).catch(function () {});
})();
