"""Web dashboard (reference: dashboard/src/app — Next.js pages /clusters,
/jobs, /jobs/new, /new, /history over the APIServer; 5.1k LoC TS).

Here: a dependency-free multi-page app (hash-routed, same page set) served
by the same FastAPI apiserver, using the v1 endpoints + the v2 ray.io proxy
via fetch(). Pages: clusters list/detail/create, jobs list/submit, services
list/detail, job submissions browser.
"""

DASHBOARD_HTML = """<!doctype html>
<html>
<head>
<meta charset="utf-8">
<title>KubeRay-AMD — MI355X Ray clusters</title>
<style>
  body { font-family: system-ui, sans-serif; margin: 0; background: #0d1117; color: #e6edf3; }
  header { display: flex; align-items: baseline; gap: 1.2rem; padding: 1rem 2rem;
           border-bottom: 1px solid #30363d; }
  header h1 { color: #ff6b35; margin: 0; font-size: 1.3rem; }
  nav a { color: #8b949e; text-decoration: none; margin-right: 1rem; }
  nav a.active { color: #ff6b35; border-bottom: 2px solid #ff6b35; }
  main { padding: 1.5rem 2rem; }
  h2 { border-bottom: 1px solid #30363d; padding-bottom: .3rem; }
  table { border-collapse: collapse; width: 100%; margin-bottom: 1.5rem; }
  th, td { text-align: left; padding: .4rem .8rem; border-bottom: 1px solid #21262d; }
  th { color: #8b949e; font-weight: 600; }
  tr.rowlink { cursor: pointer; } tr.rowlink:hover { background: #161b22; }
  .ready, .RUNNING, .SUCCEEDED, .Complete { color: #3fb950; }
  .notready, .DEPLOYING, .PENDING, .Initializing, .Running { color: #d29922; }
  .FAILED, .Failed, .DEPLOY_FAILED { color: #f85149; }
  button { background: #21262d; color: #e6edf3; border: 1px solid #30363d;
           border-radius: 6px; padding: .3rem .8rem; cursor: pointer; }
  button:hover { border-color: #ff6b35; }
  button.danger:hover { border-color: #f85149; color: #f85149; }
  input, select, textarea { background: #0d1117; color: #e6edf3; border: 1px solid #30363d;
                  border-radius: 6px; padding: .3rem; margin-right: .5rem; }
  .muted { color: #8b949e; font-size: .85rem; }
  .card { background: #161b22; border: 1px solid #30363d; border-radius: 8px;
          padding: 1rem; margin-bottom: 1rem; }
  .grid { display: grid; grid-template-columns: repeat(auto-fit, minmax(280px, 1fr));
          gap: 1rem; }
  pre { background: #161b22; border: 1px solid #30363d; border-radius: 8px;
        padding: .8rem; overflow-x: auto; }
  .crumbs a { color: #58a6ff; text-decoration: none; }
  form.stack label { display: block; margin-bottom: .6rem; }
  form.stack input, form.stack textarea { width: 24rem; max-width: 90%; }
</style>
</head>
<body>
<header>
  <h1>KubeRay-AMD</h1>
  <nav id="nav">
    <a href="#/clusters" data-page="clusters">Clusters</a>
    <a href="#/jobs" data-page="jobs">Jobs</a>
    <a href="#/services" data-page="services">Services</a>
    <a href="#/new" data-page="new">New cluster</a>
    <a href="#/jobs/new" data-page="jobsnew">Submit job</a>
  </nav>
  <span class="muted">namespace
    <input id="ns" value="default" size="10" onchange="route()"></span>
</header>
<main id="main"><p class="muted">loading…</p></main>

<script>
const $ = (id) => document.getElementById(id);
const ns = () => $('ns').value || 'default';
const esc = (s) => String(s === undefined || s === null ? '' : s).replace(/[&<>"]/g,
  (c) => ({'&':'&amp;','<':'&lt;','>':'&gt;','"':'&quot;'}[c]));
async function api(method, path, body) {
  const resp = await fetch(path, {
    method, headers: {'Content-Type': 'application/json'},
    body: body === undefined ? undefined : JSON.stringify(body)});
  if (!resp.ok) throw new Error(await resp.text());
  const text = await resp.text();
  return text ? JSON.parse(text) : {};
}
function statusCell(s) { return `<span class="${esc(s)}">${esc(s || '—')}</span>`; }

// ---------------------------------------------------------------- clusters
async function pageClusters() {
  const data = await api('GET', `/apis/v1/namespaces/${ns()}/clusters`);
  const rows = (data.clusters || []).map(c => `
    <tr class="rowlink" onclick="location.hash='#/clusters/${esc(c.name)}'">
      <td>${esc(c.name)}</td><td>${statusCell(c.clusterState)}</td>
      <td>${esc(c.createdAt || '')}</td>
      <td><button class="danger" onclick="event.stopPropagation();
        delCluster('${esc(c.name)}')">delete</button></td></tr>`).join('');
  $('main').innerHTML = `<h2>RayClusters</h2>
    <table><thead><tr><th>name</th><th>state</th><th>created</th><th></th>
    </tr></thead><tbody>${rows ||
      '<tr><td colspan=4 class=muted>none</td></tr>'}</tbody></table>`;
}
async function delCluster(name) {
  if (!confirm(`delete RayCluster ${name}?`)) return;
  await api('DELETE', `/apis/v1/namespaces/${ns()}/clusters/${name}`);
  route();
}
async function pageClusterDetail(name) {
  const obj = await api('GET',
    `/apis/ray.io/v1/namespaces/${ns()}/rayclusters/${name}`);
  const st = obj.status || {};
  const groups = (obj.spec.workerGroupSpecs || []).map(g => `
    <tr><td>${esc(g.groupName)}</td><td>${esc(g.replicas)}</td>
    <td>${esc(g.minReplicas)}–${esc(g.maxReplicas)}</td>
    <td>${esc((((((g.template.spec.containers||[])[0]||{}).resources||{})
      .limits)||{})['amd.com/gpu'] || '0')}</td></tr>`).join('');
  const conds = (st.conditions || []).map(c => `
    <tr><td>${esc(c.type)}</td><td>${statusCell(c.status)}</td>
    <td class=muted>${esc(c.reason || '')}</td></tr>`).join('');
  $('main').innerHTML = `
    <p class="crumbs"><a href="#/clusters">clusters</a> / ${esc(name)}</p>
    <h2>${esc(name)} ${statusCell(st.state)}</h2>
    <div class="grid">
    <div class="card"><b>Status</b><table>
      <tr><td>desired workers</td><td>${esc(st.desiredWorkerReplicas)}</td></tr>
      <tr><td>available workers</td><td>${esc(st.availableWorkerReplicas)}</td></tr>
      <tr><td>desired GPUs</td><td>${esc(st.desiredGPU)}</td></tr>
      <tr><td>head pod IP</td><td>${esc((st.head||{}).podIP || '')}</td></tr>
    </table></div>
    <div class="card"><b>Conditions</b>
      <table>${conds || '<tr><td class=muted>none</td></tr>'}</table></div>
    </div>
    <h2>Worker groups</h2>
    <table><thead><tr><th>group</th><th>replicas</th><th>min–max</th>
      <th>amd.com/gpu</th></tr></thead><tbody>${groups}</tbody></table>
    <h2>Raw spec</h2><pre>${esc(JSON.stringify(obj.spec, null, 2))}</pre>`;
}
async function pageNewCluster() {
  $('main').innerHTML = `<h2>New RayCluster</h2>
    <form class="stack" onsubmit="return createCluster(event)">
      <label>name <input id="cname" required></label>
      <label>workers <input id="workers" type="number" value="1" min="0"></label>
      <label>amd.com/gpu per worker
        <input id="gpus" type="number" value="1" min="0" max="8"></label>
      <button type="submit">create</button>
    </form>`;
}
async function createCluster(ev) {
  ev.preventDefault();
  const gpus = parseInt($('gpus').value || '0');
  const worker = {groupName: 'mi355x-group',
                  replicas: parseInt($('workers').value || '1'),
                  minReplicas: 0, maxReplicas: 8,
                  computeTemplate: '', rayStartParams: {}};
  await api('POST', `/apis/v1/namespaces/${ns()}/clusters`, {
    name: $('cname').value, version: '2.46.0',
    clusterSpec: {headGroupSpec: {rayStartParams: {}},
                  workerGroupSpec: [{...worker, gpu: gpus}]}});
  location.hash = '#/clusters';
  return false;
}

// -------------------------------------------------------------------- jobs
async function pageJobs() {
  const data = await api('GET', `/apis/v1/namespaces/${ns()}/jobs`);
  const rows = (data.jobs || []).map(j => `
    <tr><td>${esc(j.name)}</td>
      <td>${statusCell(j.jobDeploymentStatus)}</td>
      <td>${statusCell(j.jobStatus)}</td>
      <td>${esc(j.rayClusterName || '')}</td>
      <td><button class="danger"
        onclick="delJob('${esc(j.name)}')">delete</button></td></tr>`).join('');
  $('main').innerHTML = `<h2>RayJobs</h2>
    <table><thead><tr><th>name</th><th>deployment</th><th>app status</th>
    <th>cluster</th><th></th></tr></thead><tbody>${rows ||
      '<tr><td colspan=5 class=muted>none</td></tr>'}</tbody></table>`;
}
async function delJob(name) {
  if (!confirm(`delete RayJob ${name}?`)) return;
  await api('DELETE', `/apis/v1/namespaces/${ns()}/jobs/${name}`);
  route();
}
async function pageJobsNew() {
  $('main').innerHTML = `<h2>Submit RayJob</h2>
    <form class="stack" onsubmit="return submitJob(event)">
      <label>name <input id="jname" required></label>
      <label>entrypoint <input id="entrypoint"
        placeholder="python train.py" required></label>
      <label>workers <input id="jworkers" type="number" value="1" min="0"></label>
      <label>amd.com/gpu per worker
        <input id="jgpus" type="number" value="1" min="0" max="8"></label>
      <label><input id="jshutdown" type="checkbox" checked
        style="width:auto"> shutdown cluster after job finishes</label>
      <button type="submit">submit</button>
    </form>`;
}
async function submitJob(ev) {
  ev.preventDefault();
  await api('POST', `/apis/v1/namespaces/${ns()}/jobs`, {
    name: $('jname').value, entrypoint: $('entrypoint').value,
    shutdownAfterJobFinishes: $('jshutdown').checked,
    clusterSpec: {headGroupSpec: {rayStartParams: {}},
      workerGroupSpec: [{groupName: 'mi355x-group',
        replicas: parseInt($('jworkers').value || '1'),
        minReplicas: 0, maxReplicas: 8,
        gpu: parseInt($('jgpus').value || '0'), rayStartParams: {}}]}});
  location.hash = '#/jobs';
  return false;
}

// ---------------------------------------------------------------- services
async function pageServices() {
  const data = await api('GET', `/apis/v1/namespaces/${ns()}/services`);
  const rows = (data.services || []).map(s => `
    <tr class="rowlink" onclick="location.hash='#/services/${esc(s.name)}'">
      <td>${esc(s.name)}</td><td>${statusCell(s.serviceStatus)}</td>
      <td>${esc(s.activeRayClusterName || '')}</td>
      <td>${esc(s.pendingRayClusterName || '')}</td>
      <td><button class="danger" onclick="event.stopPropagation();
        delService('${esc(s.name)}')">delete</button></td></tr>`).join('');
  $('main').innerHTML = `<h2>RayServices</h2>
    <table><thead><tr><th>name</th><th>status</th><th>active cluster</th>
    <th>pending cluster</th><th></th></tr></thead><tbody>${rows ||
      '<tr><td colspan=5 class=muted>none</td></tr>'}</tbody></table>`;
}
async function delService(name) {
  if (!confirm(`delete RayService ${name}?`)) return;
  await api('DELETE', `/apis/v1/namespaces/${ns()}/services/${name}`);
  route();
}
async function pageServiceDetail(name) {
  const obj = await api('GET',
    `/apis/ray.io/v1/namespaces/${ns()}/rayservices/${name}`);
  const st = obj.status || {};
  const active = st.activeServiceStatus || {};
  const apps = Object.entries(active.applications || {}).map(([n, a]) => `
    <tr><td>${esc(n)}</td><td>${statusCell(a.status)}</td>
    <td class=muted>${esc(a.message || '')}</td></tr>`).join('');
  $('main').innerHTML = `
    <p class="crumbs"><a href="#/services">services</a> / ${esc(name)}</p>
    <h2>${esc(name)} ${statusCell(st.serviceStatus)}</h2>
    <div class="card"><table>
      <tr><td>serve endpoints</td><td>${esc(st.numServeEndpoints)}</td></tr>
      <tr><td>active cluster</td><td>${esc(active.rayClusterName || '')}</td></tr>
      <tr><td>pending cluster</td>
        <td>${esc((st.pendingServiceStatus||{}).rayClusterName || '')}</td></tr>
    </table></div>
    <h2>Serve applications</h2>
    <table><thead><tr><th>app</th><th>status</th><th>message</th></tr>
    </thead><tbody>${apps ||
      '<tr><td colspan=3 class=muted>none</td></tr>'}</tbody></table>
    <h2>serveConfigV2</h2><pre>${esc(obj.spec.serveConfigV2 || '')}</pre>`;
}

// ------------------------------------------------------------------ router
const ROUTES = [
  [/^#\\/clusters$/, () => pageClusters(), 'clusters'],
  [/^#\\/clusters\\/([^/]+)$/, (m) => pageClusterDetail(m[1]), 'clusters'],
  [/^#\\/jobs$/, () => pageJobs(), 'jobs'],
  [/^#\\/jobs\\/new$/, () => pageJobsNew(), 'jobsnew'],
  [/^#\\/services$/, () => pageServices(), 'services'],
  [/^#\\/services\\/([^/]+)$/, (m) => pageServiceDetail(m[1]), 'services'],
  [/^#\\/new$/, () => pageNewCluster(), 'new'],
];
async function route() {
  const hash = location.hash || '#/clusters';
  for (const [re, fn, page] of ROUTES) {
    const m = hash.match(re);
    if (m) {
      document.querySelectorAll('#nav a').forEach(a =>
        a.classList.toggle('active', a.dataset.page === page));
      try { await fn(m); } catch (e) {
        $('main').innerHTML = `<p class="FAILED">${esc(e.message)}</p>`;
      }
      return;
    }
  }
  location.hash = '#/clusters';
}
window.addEventListener('hashchange', route);
route();
setInterval(() => { // live refresh for list pages
  if (/^#\\/(clusters|jobs|services)$/.test(location.hash || '#/clusters'))
    route();
}, 5000);
</script>
</body>
</html>
"""
