"""Web dashboard (reference: dashboard/ — Next.js list/create/delete UI over
the APIServer). Here a dependency-free single-page app served at ``/`` by
the same FastAPI apiserver, using the v1 endpoints via fetch()."""

DASHBOARD_HTML = """<!doctype html>
<html>
<head>
<meta charset="utf-8">
<title>KubeRay-AMD — MI355X Ray clusters</title>
<style>
  body { font-family: system-ui, sans-serif; margin: 2rem; background: #0d1117; color: #e6edf3; }
  h1 { color: #ff6b35; } h2 { border-bottom: 1px solid #30363d; padding-bottom: .3rem; }
  table { border-collapse: collapse; width: 100%; margin-bottom: 1.5rem; }
  th, td { text-align: left; padding: .4rem .8rem; border-bottom: 1px solid #21262d; }
  th { color: #8b949e; font-weight: 600; }
  .ready { color: #3fb950; } .notready { color: #d29922; }
  button { background: #21262d; color: #e6edf3; border: 1px solid #30363d;
           border-radius: 6px; padding: .3rem .8rem; cursor: pointer; }
  button:hover { border-color: #ff6b35; }
  input, select { background: #0d1117; color: #e6edf3; border: 1px solid #30363d;
                  border-radius: 6px; padding: .3rem; margin-right: .5rem; }
  .muted { color: #8b949e; font-size: .85rem; }
</style>
</head>
<body>
<h1>KubeRay-AMD</h1>
<p class="muted">MI355X-native Ray operator — namespace
  <input id="ns" value="default" size="10"> <button onclick="refresh()">refresh</button></p>

<h2>RayClusters</h2>
<p>
  <input id="cname" placeholder="name">
  <input id="workers" type="number" value="1" min="0" style="width:4rem" title="workers">
  <input id="gpus" type="number" value="1" min="0" max="8" style="width:4rem" title="amd.com/gpu per worker">
  <button onclick="createCluster()">create</button>
</p>
<table id="clusters"><thead><tr>
  <th>name</th><th>state</th><th>workers</th><th>GPUs</th><th>created</th><th></th>
</tr></thead><tbody></tbody></table>

<h2>RayJobs</h2>
<p>
  <input id="jname" placeholder="job name">
  <input id="entrypoint" placeholder="entrypoint (python train.py)" size="32">
  <input id="jgpus" type="number" value="1" min="0" max="8" style="width:4rem"
         title="amd.com/gpu per worker">
  <button onclick="submitJob()">submit job</button>
</p>
<table id="jobs"><thead><tr>
  <th>name</th><th>deployment status</th><th>job status</th><th>cluster</th><th></th>
</tr></thead><tbody></tbody></table>

<h2>RayServices</h2>
<p>
  <input id="sname" placeholder="service name">
  <input id="sgpus" type="number" value="1" min="0" max="8" style="width:4rem"
         title="amd.com/gpu per worker">
  <button onclick="createService()">create service</button><br>
  <textarea id="serveconfig" rows="4" cols="60"
            placeholder="serveConfigV2 YAML (applications: ...)"></textarea>
</p>
<table id="services"><thead><tr>
  <th>name</th><th>status</th><th>endpoints</th><th>active cluster</th><th></th>
</tr></thead><tbody></tbody></table>

<script>
const ns = () => document.getElementById('ns').value || 'default';
const api = (p, o) => fetch(`/apis/v1/namespaces/${ns()}${p}`, o).then(r => r.json());

function row(tds, delFn) {
  const tr = document.createElement('tr');
  tds.forEach(t => { const td = document.createElement('td');
    if (t instanceof Node) td.appendChild(t); else td.innerHTML = t;
    tr.appendChild(td); });
  const td = document.createElement('td');
  const b = document.createElement('button'); b.textContent = 'delete';
  b.onclick = delFn; td.appendChild(b); tr.appendChild(td);
  return tr;
}

async function refresh() {
  const cb = document.querySelector('#clusters tbody'); cb.innerHTML = '';
  const cl = await api('/clusters');
  (cl.clusters || []).forEach(c => {
    const state = c.clusterState === 'ready'
      ? '<span class="ready">ready</span>'
      : `<span class="notready">${c.clusterState || 'pending'}</span>`;
    const gpus = (c.clusterSpec.workerGroupSpec || [])
      .map(g => `${g.groupName}:${g.replicas}`).join(' ');
    cb.appendChild(row([c.name, state, gpus,
      c.serviceEndpoint ? Object.keys(c.serviceEndpoint).length : 0,
      c.createdAt || ''],
      async () => { await fetch(`/apis/v1/namespaces/${ns()}/clusters/${c.name}`,
                                {method: 'DELETE'}); refresh(); }));
  });
  const jb = document.querySelector('#jobs tbody'); jb.innerHTML = '';
  const jl = await api('/jobs');
  (jl.jobs || []).forEach(j => jb.appendChild(row(
    [j.name, j.jobDeploymentStatus || '-', j.jobStatus || '-', j.rayClusterName || '-'],
    async () => { await fetch(`/apis/v1/namespaces/${ns()}/jobs/${j.name}`,
                              {method: 'DELETE'}); refresh(); })));
  const sb = document.querySelector('#services tbody'); sb.innerHTML = '';
  const sl = await api('/services');
  (sl.services || []).forEach(s => sb.appendChild(row(
    [s.name, s.serviceStatus || '-', s.numServeEndpoints, s.activeRayClusterName || '-'],
    async () => { await fetch(`/apis/v1/namespaces/${ns()}/services/${s.name}`,
                              {method: 'DELETE'}); refresh(); })));
}

async function createCluster() {
  const name = document.getElementById('cname').value;
  if (!name) return alert('name required');
  const workers = +document.getElementById('workers').value;
  const gpus = +document.getElementById('gpus').value;
  await fetch(`/apis/v1/namespaces/${ns()}/compute_templates`, {
    method: 'POST', headers: {'Content-Type': 'application/json'},
    body: JSON.stringify({name: `${name}-tpl`, cpu: 4, memory: 8, gpu: gpus})});
  await fetch(`/apis/v1/namespaces/${ns()}/clusters`, {
    method: 'POST', headers: {'Content-Type': 'application/json'},
    body: JSON.stringify({name, version: '2.46.0', clusterSpec: {
      headGroupSpec: {computeTemplate: `${name}-tpl`},
      workerGroupSpec: [{groupName: 'default-group',
        computeTemplate: `${name}-tpl`, replicas: workers,
        minReplicas: 0, maxReplicas: Math.max(workers, 8)}]}})});
  refresh();
}
async function submitJob() {
  const name = document.getElementById('jname').value;
  const entrypoint = document.getElementById('entrypoint').value;
  if (!name || !entrypoint) return alert('name and entrypoint required');
  const gpus = +document.getElementById('jgpus').value;
  await fetch(`/apis/v1/namespaces/${ns()}/compute_templates`, {
    method: 'POST', headers: {'Content-Type': 'application/json'},
    body: JSON.stringify({name: `${name}-tpl`, cpu: 4, memory: 8, gpu: gpus})});
  await fetch(`/apis/v1/namespaces/${ns()}/jobs`, {
    method: 'POST', headers: {'Content-Type': 'application/json'},
    body: JSON.stringify({name, entrypoint, clusterSpec: {
      headGroupSpec: {computeTemplate: `${name}-tpl`},
      workerGroupSpec: [{groupName: 'default-group',
        computeTemplate: `${name}-tpl`, replicas: 1, minReplicas: 0,
        maxReplicas: 4}]}})});
  refresh();
}

async function createService() {
  const name = document.getElementById('sname').value;
  const cfg = document.getElementById('serveconfig').value;
  if (!name || !cfg) return alert('name and serveConfigV2 required');
  const gpus = +document.getElementById('sgpus').value;
  await fetch(`/apis/v1/namespaces/${ns()}/compute_templates`, {
    method: 'POST', headers: {'Content-Type': 'application/json'},
    body: JSON.stringify({name: `${name}-tpl`, cpu: 4, memory: 8, gpu: gpus})});
  await fetch(`/apis/v1/namespaces/${ns()}/services`, {
    method: 'POST', headers: {'Content-Type': 'application/json'},
    body: JSON.stringify({name, serveConfig_V2: cfg, clusterSpec: {
      headGroupSpec: {computeTemplate: `${name}-tpl`},
      workerGroupSpec: [{groupName: 'default-group',
        computeTemplate: `${name}-tpl`, replicas: 1, minReplicas: 0,
        maxReplicas: 4}]}})});
  refresh();
}

refresh();
setInterval(refresh, 5000);
</script>
</body>
</html>
"""
