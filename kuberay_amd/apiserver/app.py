"""APIServer (reference: apiserver/ gRPC:8887 + HTTP-gateway:8888, and
apiserversdk/proxy.go).

One FastAPI app serves both surfaces:

* **v1** (``/apis/v1/namespaces/{ns}/...``) — the grpc-gateway HTTP surface
  of the reference's ClusterService / ComputeTemplateService / RayJobService
  / RayServeService / RayJobSubmissionService, speaking simplified api.*
  JSON expanded through kuberay_amd.apiserver.converters,
* **v2** (``/apis/ray.io/v1/...``) — the apiserversdk-style path-restricted
  proxy straight onto the (in-memory or real) Kubernetes API.

Run standalone: ``python -m kuberay_amd.apiserver``.
"""
from __future__ import annotations

from typing import Any, Dict, Optional

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse

from ..kube import objects as k8s
from ..kube.client import InMemoryClient, KubeClient, model_for_kind
from ..kube.store import ApiError
from ..models import RayCluster, RayJob, RayService
from ..utils import constants as C
from ..utils.validation import (
    validate_raycluster_spec,
    validate_rayjob_spec,
    validate_rayservice_spec,
)
from . import converters as conv


def create_app(client: Optional[KubeClient] = None,
               dashboard_factory=None) -> FastAPI:
    client = client or InMemoryClient()
    app = FastAPI(title="kuberay-amd-apiserver", version="1.0")
    app.state.client = client
    app.state.dashboard_factory = dashboard_factory

    def _templates(namespace: str) -> Dict[str, Dict[str, Any]]:
        cms = client.list(k8s.ConfigMap, namespace)
        out = {}
        for cm in cms:
            if (cm.metadata.labels or {}).get(conv.COMPUTE_TEMPLATE_LABEL):
                t = conv.configmap_to_compute_template(cm)
                out[t["name"]] = t
        return out

    @app.exception_handler(ApiError)
    async def api_error_handler(request: Request, exc: ApiError):
        return JSONResponse(status_code=exc.code, content={"message": exc.message})

    # ------------------------------------------------------------------
    # v1: compute templates (ConfigMaps)
    # ------------------------------------------------------------------
    @app.post("/apis/v1/namespaces/{ns}/compute_templates")
    def create_compute_template(ns: str, body: Dict[str, Any]):
        if not body.get("name"):
            raise HTTPException(400, "name is required")
        cm = conv.compute_template_to_configmap(ns, body)
        client.create(cm)
        return body

    @app.get("/apis/v1/namespaces/{ns}/compute_templates")
    def list_compute_templates(ns: str):
        return {"computeTemplates": list(_templates(ns).values())}

    @app.get("/apis/v1/compute_templates")
    def list_all_compute_templates():
        """grpc-gateway ListAllComputeTemplates route (config.proto:50)."""
        return {"computeTemplates": list(_templates(None).values())}

    @app.get("/apis/v1/namespaces/{ns}/compute_templates/{name}")
    def get_compute_template(ns: str, name: str):
        t = _templates(ns).get(name)
        if t is None:
            raise HTTPException(404, f"compute template {name} not found")
        return t

    @app.delete("/apis/v1/namespaces/{ns}/compute_templates/{name}")
    def delete_compute_template(ns: str, name: str):
        client.delete(k8s.ConfigMap, ns, name)
        return {}

    # ------------------------------------------------------------------
    # v1: image templates (config.proto ImageTemplateService :138-231,
    # grpc-gateway HTTP mapping; stored as ConfigMaps like the gRPC layer)
    # ------------------------------------------------------------------
    IMAGE_TEMPLATE_LABEL = "ray.io/image-template"

    def _image_template_of(cm: k8s.ConfigMap) -> Dict[str, Any]:
        import json as _json
        data = cm.data or {}
        return {"name": data.get("name", ""),
                "namespace": cm.metadata.namespace or "",
                "baseImage": data.get("baseImage", ""),
                "pipPackages": _json.loads(data.get("pipPackages", "[]")),
                "environmentVariables": _json.loads(
                    data.get("environmentVariables", "{}") or "{}")}

    @app.post("/apis/v1/namespaces/{ns}/image_templates")
    def create_image_template(ns: str, body: Dict[str, Any]):
        import json as _json
        if not body.get("name"):
            raise HTTPException(400, "name is required")
        cm = k8s.ConfigMap(
            metadata=k8s.ObjectMeta(
                name=f"imagetpl-{body['name']}", namespace=ns,
                labels={IMAGE_TEMPLATE_LABEL: body["name"]}),
            data={"name": body["name"],
                  "baseImage": body.get("baseImage", ""),
                  "pipPackages": _json.dumps(body.get("pipPackages") or []),
                  "environmentVariables": _json.dumps(
                      body.get("environmentVariables") or {})})
        client.create(cm)
        return body

    @app.get("/apis/v1/namespaces/{ns}/image_templates")
    def list_image_templates(ns: str):
        out = [_image_template_of(cm) for cm in client.list(k8s.ConfigMap, ns)
               if (cm.metadata.labels or {}).get(IMAGE_TEMPLATE_LABEL)]
        return {"imageTemplates": out}

    @app.get("/apis/v1/namespaces/{ns}/image_templates/{name}")
    def get_image_template(ns: str, name: str):
        cm = client.try_get(k8s.ConfigMap, ns, f"imagetpl-{name}")
        if cm is None:
            raise HTTPException(404, f"image template {name} not found")
        return _image_template_of(cm)

    @app.delete("/apis/v1/namespaces/{ns}/image_templates/{name}")
    def delete_image_template(ns: str, name: str):
        client.delete(k8s.ConfigMap, ns, f"imagetpl-{name}")
        return {}

    # ------------------------------------------------------------------
    # v1: clusters
    # ------------------------------------------------------------------
    @app.post("/apis/v1/namespaces/{ns}/clusters")
    def create_cluster(ns: str, body: Dict[str, Any]):
        if not body.get("name"):
            raise HTTPException(400, "name is required")
        rc = conv.api_cluster_to_raycluster(ns, body, _templates(ns))
        errs = validate_raycluster_spec(rc)
        if errs:
            raise HTTPException(400, "; ".join(errs))
        created = client.create(rc)
        return conv.raycluster_to_api_cluster(created)

    @app.get("/apis/v1/namespaces/{ns}/clusters")
    def list_clusters(ns: str):
        return {"clusters": [conv.raycluster_to_api_cluster(rc)
                             for rc in client.list(RayCluster, ns)]}

    @app.get("/apis/v1/clusters")
    def list_all_clusters():
        """grpc-gateway ListAllClusters route (cluster.proto:50)."""
        return {"clusters": [conv.raycluster_to_api_cluster(rc)
                             for rc in client.list(RayCluster, None)]}

    @app.get("/apis/v1/namespaces/{ns}/clusters/{name}")
    def get_cluster(ns: str, name: str):
        rc = client.try_get(RayCluster, ns, name)
        if rc is None:
            raise HTTPException(404, f"cluster {name} not found")
        out = conv.raycluster_to_api_cluster(rc)
        # operator events for this cluster (reference returns cluster
        # events); typed list works on both backends
        events = [e.to_dict() for e in client.list(k8s.Event, ns)]
        out["events"] = [
            {"reason": e.get("reason"), "type": e.get("type"),
             "message": e.get("message"), "count": e.get("count", 1),
             "lastTimestamp": e.get("lastTimestamp")}
            for e in events
            if (e.get("involvedObject") or {}).get("name") == name]
        return out

    @app.delete("/apis/v1/namespaces/{ns}/clusters/{name}")
    def delete_cluster(ns: str, name: str):
        client.delete(RayCluster, ns, name)
        return {}

    # ------------------------------------------------------------------
    # v1: jobs
    # ------------------------------------------------------------------
    @app.post("/apis/v1/namespaces/{ns}/jobs")
    def create_job(ns: str, body: Dict[str, Any]):
        if not body.get("name"):
            raise HTTPException(400, "name is required")
        job = conv.api_job_to_rayjob(ns, body, _templates(ns))
        errs = validate_rayjob_spec(job)
        if errs:
            raise HTTPException(400, "; ".join(errs))
        created = client.create(job)
        return conv.rayjob_to_api_job(created)

    @app.get("/apis/v1/namespaces/{ns}/jobs")
    def list_jobs(ns: str):
        return {"jobs": [conv.rayjob_to_api_job(j)
                         for j in client.list(RayJob, ns)]}

    @app.get("/apis/v1/jobs")
    def list_all_jobs():
        """grpc-gateway ListAllRayJobs route (job.proto:52)."""
        return {"jobs": [conv.rayjob_to_api_job(j)
                         for j in client.list(RayJob, None)]}

    @app.get("/apis/v1/namespaces/{ns}/jobs/{name}")
    def get_job(ns: str, name: str):
        job = client.try_get(RayJob, ns, name)
        if job is None:
            raise HTTPException(404, f"job {name} not found")
        return conv.rayjob_to_api_job(job)

    @app.delete("/apis/v1/namespaces/{ns}/jobs/{name}")
    def delete_job(ns: str, name: str):
        client.delete(RayJob, ns, name)
        return {}

    # ------------------------------------------------------------------
    # v1: services
    # ------------------------------------------------------------------
    @app.post("/apis/v1/namespaces/{ns}/services")
    def create_service(ns: str, body: Dict[str, Any]):
        if not body.get("name"):
            raise HTTPException(400, "name is required")
        svc = conv.api_service_to_rayservice(ns, body, _templates(ns))
        errs = validate_rayservice_spec(svc)
        if errs:
            raise HTTPException(400, "; ".join(errs))
        created = client.create(svc)
        return conv.rayservice_to_api_service(created)

    @app.get("/apis/v1/namespaces/{ns}/services")
    def list_services(ns: str):
        return {"services": [conv.rayservice_to_api_service(s)
                             for s in client.list(RayService, ns)]}

    @app.get("/apis/v1/services")
    def list_all_services():
        """grpc-gateway ListAllRayServices route (service.proto:49)."""
        return {"services": [conv.rayservice_to_api_service(x)
                             for x in client.list(RayService, None)]}

    @app.get("/apis/v1/namespaces/{ns}/services/{name}")
    def get_service(ns: str, name: str):
        svc = client.try_get(RayService, ns, name)
        if svc is None:
            raise HTTPException(404, f"service {name} not found")
        return conv.rayservice_to_api_service(svc)

    @app.delete("/apis/v1/namespaces/{ns}/services/{name}")
    def delete_service(ns: str, name: str):
        client.delete(RayService, ns, name)
        return {}

    # ------------------------------------------------------------------
    # v1: live job submission proxy (RayJobSubmissionService)
    # ------------------------------------------------------------------
    def _dashboard_for_cluster(ns: str, cluster_name: str):
        rc = client.try_get(RayCluster, ns, cluster_name)
        if rc is None:
            raise HTTPException(404, f"cluster {cluster_name} not found")
        factory = app.state.dashboard_factory
        if factory is None:
            from ..utils import names
            from ..utils.dashboard_client import RayDashboardClient
            url = (f"{names.head_service_name(C.KIND_RAYCLUSTER, rc.spec, cluster_name)}"
                   f".{ns}.svc.{names.cluster_domain_name()}:{C.DEFAULT_DASHBOARD_PORT}")
            return RayDashboardClient(url)
        return factory(cluster_name)

    @app.post("/apis/v1/namespaces/{ns}/jobsubmissions/{cluster}")
    def submit_job(ns: str, cluster: str, body: Dict[str, Any]):
        dashboard = _dashboard_for_cluster(ns, cluster)
        submission_id = dashboard.submit_job(body)
        return {"submissionId": submission_id}

    @app.get("/apis/v1/namespaces/{ns}/jobsubmissions/{cluster}")
    def list_job_submissions(ns: str, cluster: str):
        return {"submissions": _dashboard_for_cluster(ns, cluster).list_jobs()}

    @app.get("/apis/v1/namespaces/{ns}/jobsubmissions/{cluster}/{submission_id}")
    def get_job_submission(ns: str, cluster: str, submission_id: str):
        info = _dashboard_for_cluster(ns, cluster).get_job_info(submission_id)
        if info is None:
            raise HTTPException(404, f"submission {submission_id} not found")
        return info

    @app.delete("/apis/v1/namespaces/{ns}/jobsubmissions/{cluster}/{submission_id}")
    def stop_job_submission(ns: str, cluster: str, submission_id: str):
        _dashboard_for_cluster(ns, cluster).stop_job(submission_id)
        return {}

    # ------------------------------------------------------------------
    # v2: apiserversdk-style restricted proxy onto the K8s API
    # (proxy.go:28-68 — only ray.io/v1 resources are reachable)
    # ------------------------------------------------------------------
    KIND_BY_PLURAL = {"rayclusters": "RayCluster", "rayjobs": "RayJob",
                      "rayservices": "RayService", "raycronjobs": "RayCronJob"}

    @app.get("/apis/ray.io/v1/namespaces/{ns}/{plural}")
    def v2_list(ns: str, plural: str):
        kind = KIND_BY_PLURAL.get(plural)
        if kind is None:
            raise HTTPException(404, f"resource {plural} is not proxied")
        model = model_for_kind(kind)
        items = [o.to_dict() for o in client.list(model, ns)]
        return {"apiVersion": "ray.io/v1", "kind": f"{kind}List", "items": items}

    @app.get("/apis/ray.io/v1/namespaces/{ns}/{plural}/{name}")
    def v2_get(ns: str, plural: str, name: str):
        kind = KIND_BY_PLURAL.get(plural)
        if kind is None:
            raise HTTPException(404, f"resource {plural} is not proxied")
        obj = client.try_get(model_for_kind(kind), ns, name)
        if obj is None:
            raise HTTPException(404, f"{kind} {name} not found")
        return obj.to_dict()

    def _expand_compute_templates(ns: str, body: Dict[str, Any]) -> None:
        """apiserversdk compute-template middleware analog
        (apiserversdk/util/template.go): group specs naming a
        ``computeTemplate`` get its resources injected."""
        templates = _templates(ns)
        spec = body.get("spec") or {}
        # RayJob nests the cluster under rayClusterSpec, RayService under
        # rayClusterConfig (template.go:54-79)
        cluster_spec = (spec.get("rayClusterSpec")
                        or spec.get("rayClusterConfig") or spec)
        groups = [cluster_spec.get("headGroupSpec") or {}]
        groups += list(cluster_spec.get("workerGroupSpecs") or [])
        for g in groups:
            tpl_name = g.pop("computeTemplate", None)
            if not tpl_name:
                continue
            tpl = templates.get(tpl_name)
            if tpl is None:
                raise HTTPException(400, f"compute template '{tpl_name}' not found")
            containers = (((g.get("template") or {}).get("spec") or {})
                          .get("containers") or [])
            if not containers:
                g.setdefault("template", {}).setdefault("spec", {})[
                    "containers"] = [conv._container_from_template(
                        "ray", C.DEFAULT_RAY_ROCM_IMAGE, tpl)]
            elif not containers[0].get("resources"):
                limits = {"cpu": str(tpl["cpu"]), "memory": f"{tpl['memory']}Gi"}
                if tpl.get("gpu"):
                    limits[tpl.get("gpuAccelerator", C.AMD_GPU_RESOURCE_NAME)] =                         str(tpl["gpu"])
                containers[0]["resources"] = {"limits": limits,
                                              "requests": dict(limits)}

    @app.post("/apis/ray.io/v1/namespaces/{ns}/{plural}")
    def v2_create(ns: str, plural: str, body: Dict[str, Any]):
        kind = KIND_BY_PLURAL.get(plural)
        if kind is None:
            raise HTTPException(404, f"resource {plural} is not proxied")
        model = model_for_kind(kind)
        body.setdefault("metadata", {})["namespace"] = ns
        _expand_compute_templates(ns, body)
        obj = model.from_dict(body)
        return client.create(obj).to_dict()

    @app.put("/apis/ray.io/v1/namespaces/{ns}/{plural}/{name}")
    def v2_update(ns: str, plural: str, name: str, body: Dict[str, Any]):
        kind = KIND_BY_PLURAL.get(plural)
        if kind is None:
            raise HTTPException(404, f"resource {plural} is not proxied")
        model = model_for_kind(kind)
        body.setdefault("metadata", {})["namespace"] = ns
        body["metadata"]["name"] = name
        return client.update(model.from_dict(body)).to_dict()

    @app.put("/apis/ray.io/v1/namespaces/{ns}/{plural}/{name}/status")
    def v2_update_status(ns: str, plural: str, name: str, body: Dict[str, Any]):
        kind = KIND_BY_PLURAL.get(plural)
        if kind is None:
            raise HTTPException(404, f"resource {plural} is not proxied")
        model = model_for_kind(kind)
        body.setdefault("metadata", {})["namespace"] = ns
        body["metadata"]["name"] = name
        return client.update_status(model.from_dict(body)).to_dict()

    @app.patch("/apis/ray.io/v1/namespaces/{ns}/{plural}/{name}")
    def v2_patch(ns: str, plural: str, name: str, body: Dict[str, Any]):
        kind = KIND_BY_PLURAL.get(plural)
        if kind is None:
            raise HTTPException(404, f"resource {plural} is not proxied")
        return client.patch(model_for_kind(kind), ns, name, body).to_dict()

    @app.delete("/apis/ray.io/v1/namespaces/{ns}/{plural}/{name}")
    def v2_delete(ns: str, plural: str, name: str):
        kind = KIND_BY_PLURAL.get(plural)
        if kind is None:
            raise HTTPException(404, f"resource {plural} is not proxied")
        client.delete(model_for_kind(kind), ns, name)
        return {"status": "Success"}

    # ------------------------------------------------------------------
    # v1: serve applications proxy (RayServeService analog)
    # ------------------------------------------------------------------
    @app.get("/apis/v1/namespaces/{ns}/serveapplications/{cluster}")
    def get_serve_applications(ns: str, cluster: str):
        return _dashboard_for_cluster(ns, cluster).get_serve_applications()

    @app.put("/apis/v1/namespaces/{ns}/serveapplications/{cluster}")
    def update_serve_applications(ns: str, cluster: str, body: Dict[str, Any]):
        _dashboard_for_cluster(ns, cluster).update_serve_applications(body)
        return {}

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    # web dashboard (reference: dashboard/ Next.js UI)
    from fastapi.responses import HTMLResponse

    from .dashboard import DASHBOARD_HTML

    @app.get("/", response_class=HTMLResponse)
    def dashboard():
        return DASHBOARD_HTML

    return app


def main(argv=None) -> int:
    import argparse
    import uvicorn

    parser = argparse.ArgumentParser(prog="kuberay-amd-apiserver")
    parser.add_argument("--port", type=int, default=8888,
                        help="HTTP gateway port (reference :8888)")
    parser.add_argument("--grpc-port", type=int, default=8887,
                        help="gRPC port (reference :8887); 0 disables")
    parser.add_argument("--host", default="0.0.0.0")
    args = parser.parse_args(argv)
    from ..kube.client import InMemoryClient
    client = InMemoryClient()
    grpc_server = None
    if args.grpc_port:
        from .grpc_api import create_grpc_server
        grpc_server = create_grpc_server(client, port=args.grpc_port)
        grpc_server.start()
    try:
        uvicorn.run(create_app(client), host=args.host, port=args.port)
    finally:
        if grpc_server is not None:
            grpc_server.stop(2)
    return 0
