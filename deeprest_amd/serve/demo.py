"""Browser demo (the reference's web-demo/ equivalent, L8).

The reference ships a Dash UI reading precomputed results.pkl
(reference: web-demo/app.py).  Ours is a single self-contained HTML page on
the live REST API: ingest stats, what-if traffic mix sliders per API
endpoint, and per-metric quantile-band charts rendered with <canvas> — no
JS build chain, no extra dependencies.  Mount with
``app.include_router`` via serve.api.create_app (route GET /demo).
"""

DEMO_HTML = """<!DOCTYPE html>
<html>
<head>
<meta charset="utf-8"/>
<title>deeprest-amd demo</title>
<style>
 body { font-family: system-ui, sans-serif; margin: 2rem; background:#fafafa; }
 h1 { font-size: 1.3rem; } h2 { font-size: 1.05rem; margin-top: 1.5rem; }
 .row { display: flex; gap: 2rem; flex-wrap: wrap; }
 .card { background: white; border: 1px solid #ddd; border-radius: 8px;
         padding: 1rem; margin: .5rem 0; }
 canvas { border: 1px solid #eee; background: white; }
 input[type=number] { width: 5rem; }
 #status { color: #666; font-size: .9rem; }
 button { padding: .4rem .9rem; border-radius: 6px; border: 1px solid #888;
          background: #eee; cursor: pointer; }
</style>
</head>
<body>
<h1>deeprest-amd &mdash; per-endpoint resource estimation (MI355X)</h1>
<div id="status">loading&hellip;</div>

<div class="card">
  <h2>1 &middot; What-if traffic mix (calls per window)</h2>
  <div id="apis"></div>
  <p>
    windows: <input id="nwin" type="number" value="120"/>
    <button onclick="estimate()">Estimate</button>
  </p>
</div>

<div class="card">
  <h2>2 &middot; Predicted utilization (q05 / median / q95 band)</h2>
  <div id="charts" class="row"></div>
</div>

<script>
let apis = [];
async function init() {
  const h = await (await fetch('health')).json();
  let msg = 'model loaded: ' + h.model_loaded + ' | native HIP extension: ' + h.native_extension;
  try {
    const a = await (await fetch('apis')).json();
    apis = a.apis;
  } catch (e) { msg += ' | featurize first (POST /ingest then /featurize)'; }
  document.getElementById('status').textContent = msg;
  const box = document.getElementById('apis');
  box.innerHTML = apis.map((a, i) =>
    `<label style="display:inline-block;margin:.2rem .8rem .2rem 0">${a}
     <input type="number" id="api${i}" value="${i < 2 ? 20 : 0}"/></label>`).join('');
}
function drawBand(canvas, q05, q50, q95, title) {
  const ctx = canvas.getContext('2d');
  const W = canvas.width, H = canvas.height, pad = 28;
  ctx.clearRect(0, 0, W, H);
  const all = q05.concat(q95);
  const lo = Math.min(...all), hi = Math.max(...all) || 1;
  const x = i => pad + (W - 2 * pad) * i / (q50.length - 1 || 1);
  const y = v => H - pad - (H - 2 * pad) * (v - lo) / (hi - lo || 1);
  ctx.fillStyle = 'rgba(70,130,220,0.25)';
  ctx.beginPath();
  q95.forEach((v, i) => i ? ctx.lineTo(x(i), y(v)) : ctx.moveTo(x(i), y(v)));
  for (let i = q05.length - 1; i >= 0; i--) ctx.lineTo(x(i), y(q05[i]));
  ctx.closePath(); ctx.fill();
  ctx.strokeStyle = 'rgb(30,80,180)'; ctx.lineWidth = 1.5;
  ctx.beginPath();
  q50.forEach((v, i) => i ? ctx.lineTo(x(i), y(v)) : ctx.moveTo(x(i), y(v)));
  ctx.stroke();
  ctx.fillStyle = '#333'; ctx.font = '11px sans-serif';
  ctx.fillText(title, 6, 13);
  ctx.fillText(hi.toFixed(1), 2, pad); ctx.fillText(lo.toFixed(1), 2, H - pad);
}
async function estimate() {
  const plan = {};
  apis.forEach((a, i) => {
    const v = parseInt(document.getElementById('api' + i).value) || 0;
    if (v > 0) plan[a] = v;
  });
  const n = parseInt(document.getElementById('nwin').value) || 120;
  document.getElementById('status').textContent = 'estimating…';
  const r = await fetch('estimate', {
    method: 'POST', headers: {'Content-Type': 'application/json'},
    body: JSON.stringify({traffic_plan: Array(n).fill(plan), seed: 0}),
  });
  if (!r.ok) {
    document.getElementById('status').textContent = 'error: ' + (await r.text());
    return;
  }
  const body = await r.json();
  const charts = document.getElementById('charts');
  charts.innerHTML = '';
  const names = Object.keys(body.predictions).slice(0, 24);
  for (const name of names) {
    const w = body.predictions[name][0];   // first window: (T, 3)
    const c = document.createElement('canvas');
    c.width = 300; c.height = 150;
    charts.appendChild(c);
    drawBand(c, w.map(q => q[0]), w.map(q => q[1]), w.map(q => q[2]), name);
  }
  document.getElementById('status').textContent =
    'estimated ' + names.length + ' metrics for mix ' + JSON.stringify(plan) +
    ' — showing first window quantile bands';
}
init();
</script>
</body>
</html>"""


# The reference web-demo's core function — browsing precomputed results.pkl
# (web-demo/app.py:51-193: experiment/component/metric selection, ground
# truth vs all four estimators, per-component scale factors) — as a second
# self-contained page on the /results REST surface (route GET /demo/results).
RESULTS_HTML = """<!DOCTYPE html>
<html>
<head>
<meta charset="utf-8"/>
<title>deeprest-amd results</title>
<style>
 body { font-family: system-ui, sans-serif; margin: 2rem; background:#fafafa; }
 h1 { font-size: 1.3rem; }
 .card { background: white; border: 1px solid #ddd; border-radius: 8px;
         padding: 1rem; margin: .5rem 0; }
 canvas { border: 1px solid #eee; background: white; }
 select { padding: .3rem; margin-right: .8rem; }
 #status { color: #666; font-size: .9rem; }
 .legend span { margin-right: 1rem; font-size: .85rem; }
</style>
</head>
<body>
<h1>deeprest-amd &mdash; results browser</h1>
<div id="status">loading&hellip;</div>
<div class="card">
  experiment <select id="exp"></select>
  component <select id="comp"></select>
  metric <select id="metric"></select>
  query window <select id="qwin"></select>
</div>
<div class="card">
  <div class="legend">
    <span style="color:red">ground truth</span>
    <span style="color:green">bl-resrc</span>
    <span style="color:orange">bl-api</span>
    <span style="color:purple">bl-trace</span>
    <span style="color:mediumblue">ours</span>
  </div>
  <canvas id="chart" width="900" height="320"></canvas>
  <div id="scales"></div>
</div>
<div class="card">
  <b>Scaling factors</b> &mdash; all 4 estimators vs ground truth
  (magenta dashed), every metric of this component
  <!-- the reference's per-component bar view, web-demo/app.py:150-180 -->
  <div><canvas id="bars" width="900" height="220"></canvas></div>
</div>
<script>
const COLORS = {'bl-resrc':'green','bl-api':'orange','bl-trace':'purple','ours':'mediumblue'};
async function init() {
  const r = await fetch('../results');
  if (!r.ok) {
    document.getElementById('status').textContent =
      'no results loaded (serve with --results results.pkl)';
    return;
  }
  const exps = (await r.json()).experiments;
  fill('exp', exps);
  await onExp();
  document.getElementById('exp').onchange = onExp;
  document.getElementById('comp').onchange = onComp;
  document.getElementById('metric').onchange = draw;
  document.getElementById('status').textContent = exps.length + ' experiments';
}
function fill(id, values) {
  const s = document.getElementById(id);
  s.innerHTML = '';
  for (const v of values) {
    const o = document.createElement('option'); o.value = v; o.textContent = v;
    s.appendChild(o);
  }
}
async function onExp() {
  const exp = document.getElementById('exp').value;
  const comps = await (await fetch('../results/' + encodeURIComponent(exp))).json();
  window._comps = comps;
  fill('comp', Object.keys(comps));
  onComp();
}
function onComp() {
  const comp = document.getElementById('comp').value;
  fill('metric', window._comps[comp] || []);
  draw();
  drawBars();
}
async function fetchEntry(exp, comp, met) {
  return await (await fetch('../results/' + encodeURIComponent(exp) + '/' +
      encodeURIComponent(comp) + '/' + encodeURIComponent(met))).json();
}
// per-component grouped scaling-factor bars: 4 estimators per metric, with
// the ground-truth scale as a magenta dashed line per metric group
// (the reference's view, web-demo/app.py:150-180)
async function drawBars() {
  const exp = document.getElementById('exp').value;
  const comp = document.getElementById('comp').value;
  const mets = window._comps[comp] || [];
  if (!exp || !comp || !mets.length) return;
  const entries = {};
  for (const m of mets) entries[m] = await fetchEntry(exp, comp, m);
  const nwin = (entries[mets[0]].scale_groundtruth || []).length;
  const qsel = document.getElementById('qwin');
  if (qsel.options.length !== nwin) {
    fill('qwin', Array.from({length: nwin}, (_, i) => i));
    qsel.onchange = drawBars;
  }
  const qi = parseInt(qsel.value) || 0;
  const ests = ['bl-resrc', 'bl-api', 'bl-trace', 'ours'];
  const ctx = document.getElementById('bars').getContext('2d');
  const W = 900, H = 220, pad = 34;
  ctx.clearRect(0, 0, W, H);
  let hi = 1.0;
  for (const m of mets) {
    for (const est of ests) {
      const s = entries[m]['scale_' + est];
      if (s && s[qi] !== undefined) hi = Math.max(hi, s[qi]);
    }
    const g = entries[m].scale_groundtruth;
    if (g && g[qi] !== undefined) hi = Math.max(hi, g[qi]);
  }
  const y = v => H - pad - (H - 2 * pad) * v / hi;
  const groupW = (W - 2 * pad) / mets.length;
  const barW = groupW / (ests.length + 1.5);
  mets.forEach((m, gi) => {
    const x0 = pad + gi * groupW;
    ests.forEach((est, bi) => {
      const s = entries[m]['scale_' + est];
      const v = (s && s[qi] !== undefined) ? s[qi] : 0;
      ctx.fillStyle = COLORS[est];
      ctx.fillRect(x0 + bi * barW, y(v), barW - 2, H - pad - y(v));
    });
    const g = entries[m].scale_groundtruth;
    if (g && g[qi] !== undefined) {
      ctx.strokeStyle = 'magenta'; ctx.setLineDash([4, 3]); ctx.lineWidth = 2;
      ctx.beginPath();
      ctx.moveTo(x0 - 2, y(g[qi]));
      ctx.lineTo(x0 + ests.length * barW + 2, y(g[qi]));
      ctx.stroke(); ctx.setLineDash([]); ctx.lineWidth = 1;
    }
    ctx.fillStyle = '#333'; ctx.font = '11px sans-serif';
    ctx.fillText(m, x0, H - pad + 14);
  });
  ctx.fillStyle = '#333'; ctx.font = '11px sans-serif';
  ctx.fillText(hi.toFixed(2) + 'x', 2, pad);
  ctx.fillText('0', 2, H - pad);
}
async function draw() {
  const exp = document.getElementById('exp').value;
  const comp = document.getElementById('comp').value;
  const met = document.getElementById('metric').value;
  if (!exp || !comp || !met) return;
  const e = await (await fetch('../results/' + encodeURIComponent(exp) + '/' +
      encodeURIComponent(comp) + '/' + encodeURIComponent(met))).json();
  const ctx = document.getElementById('chart').getContext('2d');
  const W = 900, H = 320, pad = 30;
  ctx.clearRect(0, 0, W, H);
  const meas = e.measurement;
  let all = meas.slice();
  for (const est in COLORS)
    if (e['prediction_' + est]) all = all.concat(e['prediction_' + est]);
  const lo = Math.min(...all), hi = Math.max(...all) || 1;
  const n = Math.max(meas.length,
      ...Object.keys(COLORS).map(k => (e['prediction_' + k] || []).length));
  const x = i => pad + (W - 2 * pad) * i / (n - 1 || 1);
  const y = v => H - pad - (H - 2 * pad) * (v - lo) / (hi - lo || 1);
  const line = (arr, color, dash) => {
    if (!arr || !arr.length) return;
    ctx.strokeStyle = color; ctx.setLineDash(dash || []);
    ctx.beginPath(); ctx.moveTo(x(0), y(arr[0]));
    for (let i = 1; i < arr.length; i++) ctx.lineTo(x(i), y(arr[i]));
    ctx.stroke(); ctx.setLineDash([]);
  };
  line(meas, 'red', [5, 3]);
  for (const est in COLORS) line(e['prediction_' + est], COLORS[est]);
  const sc = [];
  if (e.scale_groundtruth) sc.push('scale gt: ' +
      e.scale_groundtruth.map(v => v.toFixed(2)).join(', '));
  for (const est in COLORS)
    if (e['scale_' + est]) sc.push('scale ' + est + ': ' +
        e['scale_' + est].map(v => v.toFixed(2)).join(', '));
  document.getElementById('scales').textContent = sc.join('  |  ');
}
init();
</script>
</body>
</html>"""
