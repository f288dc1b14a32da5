"""kray — the kubectl-plugin analog CLI
(reference: kubectl-plugin/pkg/cmd/ray.go:27-57: get / create / delete /
scale / session / log / job submit / version).

Targets a kuberay-amd apiserver (``--server``) or a real K8s apiserver via
its ray.io/v1 surface. The ``--gpu`` flag requests ``amd.com/gpu`` — there
is no NVIDIA resource path (reference: kubectl-plugin generation.go:175-214
defaulted to nvidia.com/gpu; dropped).
"""
from __future__ import annotations

import os
from typing import Optional

import click
import yaml

import kuberay_amd
from ..kube.httpclient import HttpKubeClient
from ..models import RayCluster, RayCronJob, RayJob, RayService
from ..utils import constants as C


def make_client(server: Optional[str]):
    server = server or os.environ.get("KURAY_SERVER", "http://127.0.0.1:8888")
    return HttpKubeClient(server)


@click.group()
@click.option("--server", envvar="KURAY_SERVER", default=None,
              help="kuberay-amd apiserver URL (default http://127.0.0.1:8888)")
@click.option("--namespace", "-n", default="default")
@click.pass_context
def cli(ctx, server, namespace):
    """kray — manage Ray clusters on MI355X Kubernetes."""
    ctx.ensure_object(dict)
    ctx.obj["server"] = server
    ctx.obj["namespace"] = namespace
    ctx.obj["client"] = None


def client_of(ctx):
    if ctx.obj["client"] is None:
        ctx.obj["client"] = make_client(ctx.obj["server"])
    return ctx.obj["client"]


@cli.command()
def version():
    """Print the kuberay-amd version."""
    click.echo(f"kuberay-amd {kuberay_amd.__version__} (ray.io/v1)")


# ---------------------------------------------------------------------------
# get
# ---------------------------------------------------------------------------
@cli.group()
def get():
    """Get Ray resources."""


def _emit_objects(objs, output):
    """kubectl-style -o yaml|json rendering; returns True when handled."""
    import json as _json
    if output == "yaml":
        docs = [o.to_dict() | {"kind": o.kind} for o in objs]
        click.echo(yaml.safe_dump_all(docs, sort_keys=False).rstrip())
        return True
    if output == "json":
        items = [o.to_dict() | {"kind": o.kind} for o in objs]
        out = items[0] if len(items) == 1 else {
            "apiVersion": "v1", "kind": "List", "items": items}
        click.echo(_json.dumps(out, indent=2))
        return True
    if output:
        raise click.ClickException(f"unsupported output format '{output}' "
                                   "(want yaml or json)")
    return False


_OUTPUT_OPT = click.option("-o", "--output", default=None,
                           help="output format: yaml | json")


@get.command("cluster")
@click.argument("name", required=False)
@_OUTPUT_OPT
@click.pass_context
def get_cluster(ctx, name, output):
    client = client_of(ctx)
    ns = ctx.obj["namespace"]
    rows = []
    clusters = ([client.get(RayCluster, ns, name)] if name
                else client.list(RayCluster, ns))
    if _emit_objects(clusters, output):
        return
    for rc in clusters:
        rows.append((rc.metadata.name, rc.status.state or "-",
                     rc.status.desired_worker_replicas,
                     rc.status.ready_worker_replicas,
                     rc.status.desired_cpu or "-", rc.status.desired_gpu or "-"))
    _table(["NAME", "STATE", "DESIRED", "READY", "CPUS", "GPUS"], rows)


@get.command("workergroup")
@click.argument("cluster_name")
@click.pass_context
def get_workergroup(ctx, cluster_name):
    client = client_of(ctx)
    rc = client.get(RayCluster, ctx.obj["namespace"], cluster_name)
    rows = [(g.group_name, g.replicas, g.min_replicas, g.max_replicas,
             g.num_of_hosts, "yes" if g.suspend else "no")
            for g in rc.spec.worker_group_specs]
    _table(["GROUP", "REPLICAS", "MIN", "MAX", "HOSTS", "SUSPENDED"], rows)


@get.command("job")
@click.argument("name", required=False)
@_OUTPUT_OPT
@click.pass_context
def get_job(ctx, name, output):
    client = client_of(ctx)
    ns = ctx.obj["namespace"]
    jobs = [client.get(RayJob, ns, name)] if name else client.list(RayJob, ns)
    if _emit_objects(jobs, output):
        return
    rows = [(j.metadata.name, j.status.job_deployment_status or "-",
             j.status.job_status or "-", j.status.ray_cluster_name or "-")
            for j in jobs]
    _table(["NAME", "DEPLOYMENT STATUS", "JOB STATUS", "CLUSTER"], rows)


@get.command("node")
@click.pass_context
def get_node(ctx, ):
    """List nodes with their MI355X capacity and xGMI labels."""
    client = client_of(ctx)
    server = getattr(client, "server", None)
    raw = getattr(client, "raw_list", None)
    if server is not None:
        nodes = server.list("Node")
    elif raw is not None:
        nodes = raw("Node")
    else:
        raise click.ClickException("nodes not available over this server")
    rows = []
    for n in nodes:
        labels = (n.get("metadata") or {}).get("labels") or {}
        cap = ((n.get("status") or {}).get("capacity") or {})
        rows.append((n["metadata"]["name"],
                     cap.get("amd.com/gpu", labels.get("amd.com/gpu.count", "-")),
                     labels.get("amd.com/xgmi-island", "-"),
                     labels.get("amd.com/xgmi-fully-connected", "-")))
    _table(["NODE", "AMD GPUS", "XGMI ISLAND", "FULLY CONNECTED"], rows)


@get.command("token")
@click.argument("cluster_name")
@click.pass_context
def get_token(ctx, cluster_name):
    """Print the auth token of a token-auth RayCluster."""
    import base64
    client = client_of(ctx)
    ns = ctx.obj["namespace"]
    rc = client.get(RayCluster, ns, cluster_name)
    from ..common.pod import is_auth_enabled
    if not is_auth_enabled(rc.spec):
        raise click.ClickException(f"cluster {cluster_name} does not use token auth")
    from ..utils import names as _names
    secret_name = (rc.spec.auth_options.secret_name
                   if rc.spec.auth_options and rc.spec.auth_options.secret_name
                   else _names.auth_secret_name(cluster_name))
    server = getattr(client, "server", None)
    raw = getattr(client, "raw_try_get", None)
    if server is not None:
        secret = server.try_get("Secret", ns, secret_name)
    elif raw is not None:
        secret = raw("Secret", ns, secret_name)
    else:
        raise click.ClickException("secrets not available over this server")
    if secret is None:
        raise click.ClickException(f"auth secret {secret_name} not found")
    token = base64.b64decode((secret.get("data") or {}).get("auth_token", "")).decode()
    click.echo(token)


@get.command("events")
@click.argument("name", required=False)
@click.pass_context
def get_events(ctx, name):
    """List operator events (optionally for one object)."""
    client = client_of(ctx)
    ns = ctx.obj["namespace"]
    # events are core objects — fetch through the raw surface when available
    raw = getattr(client, "raw_list", None)
    server = getattr(client, "server", None)
    if server is not None:
        events = server.list("Event", ns)
    elif raw is not None:
        events = raw("Event", ns)
    else:
        raise click.ClickException("events not available over this server")
    rows = []
    for e in events:
        involved = e.get("involvedObject") or {}
        if name and involved.get("name") != name:
            continue
        rows.append((e.get("lastTimestamp", ""), e.get("type", ""),
                     e.get("reason", ""),
                     f"{involved.get('kind','')}/{involved.get('name','')}",
                     e.get("count", 1), (e.get("message") or "")[:60]))
    rows.sort()
    _table(["LAST SEEN", "TYPE", "REASON", "OBJECT", "COUNT", "MESSAGE"], rows)


@get.command("cronjob")
@click.argument("name", required=False)
@click.pass_context
def get_cronjob(ctx, name):
    client = client_of(ctx)
    ns = ctx.obj["namespace"]
    crons = ([client.get(RayCronJob, ns, name)] if name
             else client.list(RayCronJob, ns))
    rows = [(c.metadata.name, c.spec.schedule,
             c.spec.time_zone or "UTC",
             "yes" if c.spec.suspend else "no",
             c.status.last_schedule_time or "-")
            for c in crons]
    _table(["NAME", "SCHEDULE", "TIMEZONE", "SUSPENDED", "LAST SCHEDULE"],
           rows)


@get.command("service")
@click.argument("name", required=False)
@_OUTPUT_OPT
@click.pass_context
def get_service(ctx, name, output):
    client = client_of(ctx)
    ns = ctx.obj["namespace"]
    svcs = ([client.get(RayService, ns, name)] if name
            else client.list(RayService, ns))
    if _emit_objects(svcs, output):
        return
    rows = [(s.metadata.name, s.status.service_status or "-",
             s.status.num_serve_endpoints,
             s.status.active_service_status.ray_cluster_name or "-")
            for s in svcs]
    _table(["NAME", "STATUS", "ENDPOINTS", "ACTIVE CLUSTER"], rows)


# ---------------------------------------------------------------------------
# create
# ---------------------------------------------------------------------------
@cli.group()
def create():
    """Create Ray resources."""


def _cluster_spec(image, head_cpu, head_memory, worker_replicas, worker_cpu,
                  worker_memory, worker_gpu, autoscaler):
    worker_limits = {"cpu": worker_cpu, "memory": worker_memory}
    if worker_gpu:
        worker_limits[C.AMD_GPU_RESOURCE_NAME] = str(worker_gpu)
    spec = {
        "rayVersion": "2.46.0",
        "headGroupSpec": {
            "rayStartParams": {},
            "template": {"spec": {"containers": [{
                "name": "ray-head", "image": image,
                "resources": {"limits": {"cpu": head_cpu, "memory": head_memory},
                              "requests": {"cpu": head_cpu, "memory": head_memory}},
            }]}},
        },
        "workerGroupSpecs": [{
            "groupName": "default-group",
            "replicas": worker_replicas,
            "minReplicas": 0,
            "maxReplicas": max(worker_replicas, 8),
            "rayStartParams": {},
            "template": {"spec": {"containers": [{
                "name": "ray-worker", "image": image,
                "resources": {"limits": dict(worker_limits),
                              "requests": dict(worker_limits)},
            }]}},
        }],
    }
    if autoscaler:
        spec["enableInTreeAutoscaling"] = True
    return spec


@create.command("cluster")
@click.argument("name")
@click.option("--image", default=C.DEFAULT_RAY_ROCM_IMAGE, show_default=True)
@click.option("--head-cpu", default="2")
@click.option("--head-memory", default="4Gi")
@click.option("--worker-replicas", default=1, type=int)
@click.option("--worker-cpu", default="4")
@click.option("--worker-memory", default="8Gi")
@click.option("--worker-gpu", default=0, type=int,
              help="amd.com/gpu per worker (MI355X)")
@click.option("--autoscaler", is_flag=True)
@click.option("--dry-run", is_flag=True, help="print YAML instead of creating")
@click.pass_context
def create_cluster(ctx, name, image, head_cpu, head_memory, worker_replicas,
                   worker_cpu, worker_memory, worker_gpu, autoscaler, dry_run):
    spec = _cluster_spec(image, head_cpu, head_memory, worker_replicas,
                         worker_cpu, worker_memory, worker_gpu, autoscaler)
    obj = {"apiVersion": "ray.io/v1", "kind": "RayCluster",
           "metadata": {"name": name, "namespace": ctx.obj["namespace"]},
           "spec": spec}
    if dry_run:
        click.echo(yaml.safe_dump(obj, sort_keys=False))
        return
    client_of(ctx).create(RayCluster.from_dict(obj))
    click.echo(f"raycluster.ray.io/{name} created")


@create.command("service")
@click.argument("name")
@click.option("--serve-config", "serve_config", required=True,
              type=click.Path(exists=True),
              help="path to a serveConfigV2 YAML file")
@click.option("--image", default=C.DEFAULT_RAY_ROCM_IMAGE, show_default=True)
@click.option("--worker-replicas", default=1, type=int)
@click.option("--worker-gpu", default=1, type=int,
              help="amd.com/gpu per worker (MI355X)")
@click.option("--upgrade-strategy", default=None,
              type=click.Choice(["NewCluster",
                                 "NewClusterWithIncrementalUpgrade", "None"]))
@click.option("--dry-run", is_flag=True)
@click.pass_context
def create_service(ctx, name, serve_config, image, worker_replicas,
                   worker_gpu, upgrade_strategy, dry_run):
    """Create a RayService from a serve-config file."""
    with open(serve_config) as f:
        cfg = f.read()
    spec = {
        "serveConfigV2": cfg,
        "rayClusterConfig": _cluster_spec(image, "2", "4Gi", worker_replicas,
                                          "4", "8Gi", worker_gpu, False),
    }
    if upgrade_strategy:
        spec["upgradeStrategy"] = {"type": upgrade_strategy}
    obj = {"apiVersion": "ray.io/v1", "kind": "RayService",
           "metadata": {"name": name, "namespace": ctx.obj["namespace"]},
           "spec": spec}
    if dry_run:
        click.echo(yaml.safe_dump(obj, sort_keys=False))
        return
    client_of(ctx).create(RayService.from_dict(obj))
    click.echo(f"rayservice.ray.io/{name} created")


@create.command("cronjob")
@click.argument("name")
@click.option("--schedule", required=True, help='5-field cron, e.g. "0 9 * * *"')
@click.option("--timezone", "time_zone", default=None,
              help="IANA zone for the schedule (default UTC)")
@click.option("--entrypoint", required=True)
@click.option("--image", default=C.DEFAULT_RAY_ROCM_IMAGE, show_default=True)
@click.option("--worker-replicas", default=1, type=int)
@click.option("--worker-gpu", default=0, type=int)
@click.option("--dry-run", is_flag=True)
@click.pass_context
def create_cronjob(ctx, name, schedule, time_zone, entrypoint, image,
                   worker_replicas, worker_gpu, dry_run):
    """Create a RayCronJob that submits a RayJob on a schedule."""
    spec = {
        "schedule": schedule,
        "jobTemplate": {
            "entrypoint": entrypoint,
            "shutdownAfterJobFinishes": True,
            "rayClusterSpec": _cluster_spec(image, "2", "4Gi",
                                            worker_replicas, "4", "8Gi",
                                            worker_gpu, False),
        },
    }
    if time_zone:
        spec["timeZone"] = time_zone
    obj = {"apiVersion": "ray.io/v1", "kind": "RayCronJob",
           "metadata": {"name": name, "namespace": ctx.obj["namespace"]},
           "spec": spec}
    if dry_run:
        click.echo(yaml.safe_dump(obj, sort_keys=False))
        return
    client_of(ctx).create(RayCronJob.from_dict(obj))
    click.echo(f"raycronjob.ray.io/{name} created")


@create.command("workergroup")
@click.argument("cluster_name")
@click.option("--group-name", required=True)
@click.option("--image", default=None)
@click.option("--worker-replicas", default=1, type=int)
@click.option("--worker-cpu", default="4")
@click.option("--worker-memory", default="8Gi")
@click.option("--worker-gpu", default=0, type=int)
@click.pass_context
def create_workergroup(ctx, cluster_name, group_name, image, worker_replicas,
                       worker_cpu, worker_memory, worker_gpu):
    client = client_of(ctx)
    limits = {"cpu": worker_cpu, "memory": worker_memory}
    if worker_gpu:
        limits[C.AMD_GPU_RESOURCE_NAME] = str(worker_gpu)
    from ..models.raycluster import WorkerGroupSpec

    def add_group(rc):
        if any(g.group_name == group_name for g in rc.spec.worker_group_specs):
            raise click.ClickException(
                f"worker group '{group_name}' already exists")
        head_image = rc.spec.head_group_spec.template.spec.containers[0].image
        rc.spec.worker_group_specs.append(WorkerGroupSpec.from_dict({
            "groupName": group_name, "replicas": worker_replicas,
            "minReplicas": 0, "maxReplicas": max(worker_replicas, 8),
            "rayStartParams": {},
            "template": {"spec": {"containers": [{
                "name": "ray-worker", "image": image or head_image,
                "resources": {"limits": limits, "requests": dict(limits)}}]}},
        }))

    client.update_with_retry(RayCluster, ctx.obj["namespace"], cluster_name,
                             add_group)
    click.echo(f"worker group {group_name} added to {cluster_name}")


# ---------------------------------------------------------------------------
# delete / scale / suspend
# ---------------------------------------------------------------------------
@cli.command()
@click.argument("kind", type=click.Choice(["cluster", "job", "service", "cronjob"]))
@click.argument("name")
@click.pass_context
def delete(ctx, kind, name):
    """Delete a Ray resource."""
    model = {"cluster": RayCluster, "job": RayJob,
             "service": RayService, "cronjob": RayCronJob}[kind]
    client_of(ctx).delete(model, ctx.obj["namespace"], name)
    click.echo(f"{kind} {name} deleted")


@cli.group()
def scale():
    """Scale Ray resources."""


@scale.command("cluster")
@click.argument("name")
@click.option("--worker-group", default=None)
@click.option("--replicas", required=True, type=int)
@click.pass_context
def scale_cluster(ctx, name, worker_group, replicas):
    client = client_of(ctx)
    scaled_group = {}

    def set_replicas(rc):
        groups = rc.spec.worker_group_specs
        if worker_group:
            target = next((g for g in groups
                           if g.group_name == worker_group), None)
            if target is None:
                raise click.ClickException(
                    f"worker group '{worker_group}' not found "
                    f"(have: {[g.group_name for g in groups]})")
        elif len(groups) == 1:
            target = groups[0]
        else:
            raise click.ClickException(
                "--worker-group required (multiple groups)")
        target.replicas = replicas
        scaled_group["name"] = target.group_name

    client.update_with_retry(RayCluster, ctx.obj["namespace"], name,
                             set_replicas)
    click.echo(f"cluster {name}/{scaled_group['name']} scaled to {replicas}")


# ---------------------------------------------------------------------------
# job submit
# ---------------------------------------------------------------------------
@cli.group()
def job():
    """Ray job operations."""


@job.command("submit")
@click.option("--name", required=True)
@click.option("--entrypoint", required=True)
@click.option("--image", default=C.DEFAULT_RAY_ROCM_IMAGE)
@click.option("--worker-replicas", default=1, type=int)
@click.option("--worker-gpu", default=0, type=int)
@click.option("--runtime-env", default=None, help="runtime env YAML file")
@click.option("--shutdown-after-finish/--keep-cluster", default=True)
@click.option("--dry-run", is_flag=True)
@click.pass_context
def job_submit(ctx, name, entrypoint, image, worker_replicas, worker_gpu,
               runtime_env, shutdown_after_finish, dry_run):
    spec = {
        "entrypoint": entrypoint,
        "shutdownAfterJobFinishes": shutdown_after_finish,
        "rayClusterSpec": _cluster_spec(image, "2", "4Gi", worker_replicas,
                                        "4", "8Gi", worker_gpu, False),
    }
    if runtime_env:
        with open(runtime_env) as f:
            spec["runtimeEnvYAML"] = f.read()
    obj = {"apiVersion": "ray.io/v1", "kind": "RayJob",
           "metadata": {"name": name, "namespace": ctx.obj["namespace"]},
           "spec": spec}
    if dry_run:
        click.echo(yaml.safe_dump(obj, sort_keys=False))
        return
    client_of(ctx).create(RayJob.from_dict(obj))
    click.echo(f"rayjob.ray.io/{name} submitted")


# ---------------------------------------------------------------------------
# session / log
# ---------------------------------------------------------------------------
@cli.command()
@click.argument("cluster_name")
@click.option("--forward", "-f", is_flag=True,
              help="forward dashboard/client/serve ports to localhost and "
                   "block (Ctrl-C to stop)")
@click.option("--port", "ports", multiple=True,
              help="local:remote pair to forward (repeatable; default "
                   "8265:8265 10001:10001 8000:8000)")
@click.option("--target", default=None,
              help="override the forward target host (defaults to the "
                   "head pod IP from cluster status)")
@click.pass_context
def session(ctx, cluster_name, forward, ports, target):
    """Connect to a cluster: print endpoints, or --forward its ports."""
    client = client_of(ctx)
    rc = client.get(RayCluster, ctx.obj["namespace"], cluster_name)
    head_svc = f"{cluster_name}-head-svc.{ctx.obj['namespace']}.svc"
    click.echo(f"cluster:   {cluster_name} ({rc.status.state})")
    click.echo(f"dashboard: http://{head_svc}:8265")
    click.echo(f"client:    ray://{head_svc}:10001")
    click.echo(f"serve:     http://{head_svc}:8000")
    if not forward:
        click.echo("kubectl port-forward "
                   f"svc/{cluster_name}-head-svc 8265:8265 10001:10001 "
                   "8000:8000   (or: kray session --forward)")
        return

    from .portforward import PortForwarder
    host = target or rc.status.head.pod_ip
    if not host:
        raise click.ClickException(
            f"cluster {cluster_name} has no head pod IP yet (state="
            f"{rc.status.state}); pass --target to forward anyway")
    mappings = []
    for pair in ports or ("8265:8265", "10001:10001", "8000:8000"):
        local, _, remote = pair.partition(":")
        mappings.append((int(local), int(remote or local)))
    fwd = PortForwarder(host, mappings).start()
    for (local, remote), bound in zip(mappings, fwd.local_ports):
        click.echo(f"forwarding 127.0.0.1:{bound} -> {host}:{remote}")
    fwd.wait()


@cli.command()
@click.argument("kind", type=click.Choice(["cluster", "job", "service",
                                           "cronjob"]))
@click.argument("name")
@click.option("--resume/--suspend", "resume", default=False)
@click.pass_context
def suspend(ctx, kind, name, resume):
    """Suspend (or --resume) a Ray resource."""
    client = client_of(ctx)
    ns = ctx.obj["namespace"]
    model = {"cluster": RayCluster, "job": RayJob, "service": RayService,
             "cronjob": RayCronJob}[kind]
    def set_suspend(obj):
        obj.spec.suspend = not resume

    client.update_with_retry(model, ns, name, set_suspend)
    click.echo(f"{kind} {name} {'resumed' if resume else 'suspended'}")


@cli.command()
@click.argument("cluster_name")
@click.option("--pods", is_flag=True,
              help="also fetch each pod's container logs (REST backend)")
@click.option("--tail", type=int, default=100, help="lines per pod with --pods")
@click.pass_context
def log(ctx, cluster_name, pods, tail):
    """Show cluster state/conditions; --pods downloads pod logs."""
    client = client_of(ctx)
    ns = ctx.obj["namespace"]
    rc = client.get(RayCluster, ns, cluster_name)
    click.echo(f"state={rc.status.state} head={rc.status.head.pod_name} "
               f"ready={rc.status.ready_worker_replicas}/"
               f"{rc.status.desired_worker_replicas}")
    for cond in rc.status.conditions or []:
        click.echo(f"  {cond.type}={cond.status} ({cond.reason})")
    if not pods:
        return
    fetch = getattr(client, "pod_logs", None)
    if fetch is None:
        raise click.ClickException(
            "pod logs need the kubernetes REST backend (this server has no "
            "kubelet, so there are no container logs to download)")
    for view in client.list_pod_views(ns, {C.RAY_CLUSTER_LABEL_KEY:
                                           cluster_name}):
        click.echo(f"----- {view.name} -----")
        try:
            click.echo(fetch(ns, view.name, tail_lines=tail))
        except Exception as exc:  # keep going across pods
            click.echo(f"(failed to fetch logs: {exc})")


def _table(headers, rows):
    widths = [max(len(str(h)), *(len(str(r[i])) for r in rows)) if rows else len(h)
              for i, h in enumerate(headers)]
    click.echo("  ".join(str(h).ljust(w) for h, w in zip(headers, widths)))
    for r in rows:
        click.echo("  ".join(str(c).ljust(w) for c, w in zip(r, widths)))


def main(argv=None):
    return cli(args=argv, standalone_mode=True)


if __name__ == "__main__":
    main()
