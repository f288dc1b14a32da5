"""gRPC API surface (reference: apiserver gRPC :8887 + proto/ definitions).

The reference serves its v1 API over gRPC with an HTTP gateway; this module
provides the same dual surface. There is no protoc in the image, so the
message types are built at import time from a FileDescriptorProto via the
protobuf runtime (semantically equivalent to compiling
deploy/proto/kuberayamd.proto, which is shipped for external clients).

Services (reference proto/cluster.proto:26, config.proto:26, job.proto:28):
  ClusterService          Create/Get/List/Delete
  ComputeTemplateService  Create/Get/List/Delete
  RayJobService           Create/Get/List/Delete
  RayServeService         Create/Get/List/Delete (serve.proto:25 analog)

Complex cluster specs travel as a JSON payload field (`spec_json`) — the
simplified scalar fields match the HTTP v1 surface.
"""
from __future__ import annotations

import json
from concurrent import futures
from typing import Any, Dict

import grpc
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

# ---------------------------------------------------------------------------
# dynamic proto definitions (kept in sync with deploy/proto/kuberayamd.proto)
# ---------------------------------------------------------------------------

_FDP = descriptor_pb2.FileDescriptorProto()
_FDP.name = "kuberayamd/v1/api.proto"
_FDP.package = "kuberayamd.v1"
_FDP.syntax = "proto3"


def _msg(name, fields):
    m = _FDP.message_type.add()
    m.name = name
    for i, (fname, ftype) in enumerate(fields, start=1):
        f = m.field.add()
        f.name = fname
        f.number = i
        f.label = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED \
            if ftype.startswith("repeated:") else \
            descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
        ftype = ftype.split(":")[-1]
        if ftype == "string":
            f.type = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
        elif ftype == "int32":
            f.type = descriptor_pb2.FieldDescriptorProto.TYPE_INT32
        elif ftype == "bool":
            f.type = descriptor_pb2.FieldDescriptorProto.TYPE_BOOL
        else:  # message reference
            f.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
            f.type_name = f".kuberayamd.v1.{ftype}"


_msg("Cluster", [("name", "string"), ("namespace", "string"),
                 ("version", "string"), ("user", "string"),
                 ("cluster_state", "string"), ("created_at", "string"),
                 ("spec_json", "string")])
_msg("ComputeTemplate", [("name", "string"), ("namespace", "string"),
                         ("cpu", "int32"), ("memory", "int32"),
                         ("gpu", "int32"), ("gpu_accelerator", "string")])
_msg("RayJob", [("name", "string"), ("namespace", "string"),
                ("entrypoint", "string"), ("job_status", "string"),
                ("job_deployment_status", "string"),
                ("ray_cluster_name", "string"), ("spec_json", "string")])
_msg("RayServiceMsg", [("name", "string"), ("namespace", "string"),
                       ("serve_config_v2", "string"),
                       ("service_status", "string"),
                       ("active_ray_cluster_name", "string"),
                       ("pending_ray_cluster_name", "string"),
                       ("spec_json", "string")])
_msg("GetRequest", [("name", "string"), ("namespace", "string")])
_msg("ListRequest", [("namespace", "string")])
_msg("DeleteRequest", [("name", "string"), ("namespace", "string")])
_msg("Empty", [])
# error.proto Status analog (error codes ride gRPC status; this shape is for
# HTTP-gateway error bodies)
_msg("Status", [("error", "string"), ("code", "int32")])
# job_submission.proto:73-179 shapes
_msg("RayJobSubmission", [("entrypoint", "string"),
                          ("submission_id", "string"),
                          ("runtime_env", "string"),
                          ("metadata_json", "string"),
                          ("entrypoint_num_cpus", "int32"),
                          ("entrypoint_num_gpus", "int32")])
_msg("SubmitRayJobRequest", [("namespace", "string"),
                             ("clustername", "string"),
                             ("jobsubmission", "RayJobSubmission")])
_msg("SubmitRayJobReply", [("submission_id", "string")])
_msg("JobSubmissionQuery", [("namespace", "string"),
                            ("clustername", "string"),
                            ("submissionid", "string")])
_msg("JobSubmissionInfo", [("submission_id", "string"),
                           ("status", "string"), ("entrypoint", "string"),
                           ("message", "string"), ("error_type", "string"),
                           ("start_time", "string"), ("end_time", "string")])
_msg("ListJobSubmissionInfo",
     [("submissions", "repeated:JobSubmissionInfo")])
_msg("GetJobLogReply", [("log", "string")])
# config.proto ImageTemplateService shapes (:138-231)
_msg("ImageTemplate", [("name", "string"), ("namespace", "string"),
                       ("base_image", "string"),
                       ("pip_packages", "repeated:string"),
                       ("environment_variables_json", "string")])
_msg("ListImageTemplateResponse",
     [("image_templates", "repeated:ImageTemplate")])
_msg("ListClusterResponse", [("clusters", "repeated:Cluster")])
_msg("ListComputeTemplateResponse",
     [("compute_templates", "repeated:ComputeTemplate")])
_msg("ListRayJobResponse", [("jobs", "repeated:RayJob")])
_msg("ListRayServiceResponse", [("services", "repeated:RayServiceMsg")])

_pool = descriptor_pool.DescriptorPool()
_file_desc = _pool.Add(_FDP)


def _cls(name):
    return message_factory.GetMessageClass(
        _pool.FindMessageTypeByName(f"kuberayamd.v1.{name}"))


Cluster = _cls("Cluster")
ComputeTemplate = _cls("ComputeTemplate")
RayJobMsg = _cls("RayJob")
GetRequest = _cls("GetRequest")
ListRequest = _cls("ListRequest")
DeleteRequest = _cls("DeleteRequest")
Empty = _cls("Empty")
ListClusterResponse = _cls("ListClusterResponse")
ListComputeTemplateResponse = _cls("ListComputeTemplateResponse")
ListRayJobResponse = _cls("ListRayJobResponse")
RayServiceMsg = _cls("RayServiceMsg")
ListRayServiceResponse = _cls("ListRayServiceResponse")
Status = _cls("Status")
RayJobSubmission = _cls("RayJobSubmission")
SubmitRayJobRequest = _cls("SubmitRayJobRequest")
SubmitRayJobReply = _cls("SubmitRayJobReply")
JobSubmissionQuery = _cls("JobSubmissionQuery")
JobSubmissionInfo = _cls("JobSubmissionInfo")
ListJobSubmissionInfo = _cls("ListJobSubmissionInfo")
GetJobLogReply = _cls("GetJobLogReply")
ImageTemplate = _cls("ImageTemplate")
ListImageTemplateResponse = _cls("ListImageTemplateResponse")


# ---------------------------------------------------------------------------
# service implementation over the same converters as the HTTP surface
# ---------------------------------------------------------------------------

class _Service:
    """Shared backend for the six gRPC services."""

    def __init__(self, client, dashboard_factory=None):
        from ..kube.client import InMemoryClient
        self.client = client or InMemoryClient()
        self.dashboard_factory = dashboard_factory

    def _dashboard_for_cluster(self, context, namespace: str,
                               cluster_name: str):
        """Resolve a RayCluster's dashboard (same seam as the HTTP
        job-submission proxy in app.py)."""
        from ..models import RayCluster
        from ..utils import constants as C
        from ..utils import names
        rc = self.client.try_get(RayCluster, namespace or "default",
                                 cluster_name)
        if rc is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"cluster {cluster_name} not found")
        if self.dashboard_factory is not None:
            return self.dashboard_factory(
                f"{cluster_name}.{namespace}:{C.DEFAULT_DASHBOARD_PORT}")
        from ..utils.dashboard_client import RayDashboardClient
        head_svc = names.head_service_name(C.KIND_RAYCLUSTER, rc.spec,
                                           cluster_name)
        return RayDashboardClient(
            f"{head_svc}.{namespace}.svc.{names.cluster_domain_name()}:"
            f"{C.DEFAULT_DASHBOARD_PORT}")

    def _templates(self, namespace: str) -> Dict[str, Dict[str, Any]]:
        from ..kube import objects as k8s
        from . import converters as conv
        out = {}
        for cm in self.client.list(k8s.ConfigMap, namespace):
            if (cm.metadata.labels or {}).get(conv.COMPUTE_TEMPLATE_LABEL):
                t = conv.configmap_to_compute_template(cm)
                out[t["name"]] = t
        return out

    # -- clusters ------------------------------------------------------
    def create_cluster(self, request, context):
        from ..models import RayCluster
        from . import converters as conv
        ns = request.namespace or "default"
        body = {"name": request.name, "version": request.version,
                "user": request.user or None}
        if request.spec_json:
            body["clusterSpec"] = json.loads(request.spec_json)
        rc = conv.api_cluster_to_raycluster(ns, body, self._templates(ns))
        created = self.client.create(rc)
        return self._cluster_msg(created)

    def _cluster_msg(self, rc):
        return Cluster(
            name=rc.metadata.name or "",
            namespace=rc.metadata.namespace or "",
            version=rc.spec.ray_version or "",
            cluster_state=rc.status.state or "",
            created_at=rc.metadata.creation_timestamp or "",
            spec_json=json.dumps(rc.spec.to_dict()),
        )

    def get_cluster(self, request, context):
        from ..models import RayCluster
        rc = self.client.try_get(RayCluster, request.namespace or "default",
                                 request.name)
        if rc is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"cluster {request.name} not found")
        return self._cluster_msg(rc)

    def list_clusters(self, request, context):
        from ..models import RayCluster
        out = ListClusterResponse()
        for rc in self.client.list(RayCluster, request.namespace or "default"):
            out.clusters.append(self._cluster_msg(rc))
        return out

    def delete_cluster(self, request, context):
        from ..kube.store import NotFoundError
        from ..models import RayCluster
        try:
            self.client.delete(RayCluster, request.namespace or "default",
                               request.name)
        except NotFoundError:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"cluster {request.name} not found")
        return Empty()

    # -- compute templates ----------------------------------------------
    def create_compute_template(self, request, context):
        from . import converters as conv
        ns = request.namespace or "default"
        body = {"name": request.name, "cpu": request.cpu or 1,
                "memory": request.memory or 1, "gpu": request.gpu,
                **({"gpuAccelerator": request.gpu_accelerator}
                   if request.gpu_accelerator else {})}
        self.client.create(conv.compute_template_to_configmap(ns, body))
        return request

    def list_compute_templates(self, request, context):
        out = ListComputeTemplateResponse()
        for t in self._templates(request.namespace or "default").values():
            out.compute_templates.append(ComputeTemplate(
                name=t["name"], namespace=t.get("namespace") or "",
                cpu=t["cpu"], memory=t["memory"], gpu=t["gpu"],
                gpu_accelerator=t["gpuAccelerator"]))
        return out

    def delete_compute_template(self, request, context):
        from ..kube import objects as k8s
        from ..kube.store import NotFoundError
        try:
            self.client.delete(k8s.ConfigMap, request.namespace or "default",
                               request.name)
        except NotFoundError:
            context.abort(grpc.StatusCode.NOT_FOUND, "not found")
        return Empty()

    # -- jobs ------------------------------------------------------------
    def create_ray_job(self, request, context):
        from . import converters as conv
        ns = request.namespace or "default"
        body = {"name": request.name, "entrypoint": request.entrypoint}
        if request.spec_json:
            body.update(json.loads(request.spec_json))
        job = conv.api_job_to_rayjob(ns, body, self._templates(ns))
        created = self.client.create(job)
        return self._job_msg(created)

    def _job_msg(self, job):
        return RayJobMsg(
            name=job.metadata.name or "",
            namespace=job.metadata.namespace or "",
            entrypoint=job.spec.entrypoint or "",
            job_status=job.status.job_status or "",
            job_deployment_status=job.status.job_deployment_status or "",
            ray_cluster_name=job.status.ray_cluster_name or "")

    def get_ray_job(self, request, context):
        from ..models import RayJob
        job = self.client.try_get(RayJob, request.namespace or "default",
                                  request.name)
        if job is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"job {request.name} not found")
        return self._job_msg(job)

    def list_ray_jobs(self, request, context):
        from ..models import RayJob
        out = ListRayJobResponse()
        for job in self.client.list(RayJob, request.namespace or "default"):
            out.jobs.append(self._job_msg(job))
        return out

    def delete_ray_job(self, request, context):
        from ..kube.store import NotFoundError
        from ..models import RayJob
        try:
            self.client.delete(RayJob, request.namespace or "default",
                               request.name)
        except NotFoundError:
            context.abort(grpc.StatusCode.NOT_FOUND, "not found")
        return Empty()

    # -- services (reference proto/serve.proto RayServeService) ---------
    def create_ray_service(self, request, context):
        from . import converters as conv
        ns = request.namespace or "default"
        body = {"name": request.name,
                "serveConfigV2": request.serve_config_v2 or None}
        if request.spec_json:
            body.update(json.loads(request.spec_json))
        svc = conv.api_service_to_rayservice(ns, body, self._templates(ns))
        created = self.client.create(svc)
        return self._service_msg(created)

    def _service_msg(self, svc):
        return RayServiceMsg(
            name=svc.metadata.name or "",
            namespace=svc.metadata.namespace or "",
            serve_config_v2=svc.spec.serve_config_v2 or "",
            service_status=svc.status.service_status or "",
            active_ray_cluster_name=
                svc.status.active_service_status.ray_cluster_name or "",
            pending_ray_cluster_name=
                svc.status.pending_service_status.ray_cluster_name or "")

    def get_ray_service(self, request, context):
        from ..models import RayService
        svc = self.client.try_get(RayService, request.namespace or "default",
                                  request.name)
        if svc is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"service {request.name} not found")
        return self._service_msg(svc)

    def list_ray_services(self, request, context):
        from ..models import RayService
        out = ListRayServiceResponse()
        for svc in self.client.list(RayService, request.namespace or "default"):
            out.services.append(self._service_msg(svc))
        return out

    def delete_ray_service(self, request, context):
        from ..kube.store import NotFoundError
        from ..models import RayService
        try:
            self.client.delete(RayService, request.namespace or "default",
                               request.name)
        except NotFoundError:
            context.abort(grpc.StatusCode.NOT_FOUND, "not found")
        return Empty()

    def update_ray_service(self, request, context):
        """serve.proto:39 UpdateRayService — replace serveConfigV2/spec."""
        from ..models import RayService
        ns = request.namespace or "default"

        def mutate(svc):
            if request.serve_config_v2:
                svc.spec.serve_config_v2 = request.serve_config_v2
            if request.spec_json:
                spec = json.loads(request.spec_json)
                fresh = RayService.from_dict(
                    {"apiVersion": "ray.io/v1", "kind": "RayService",
                     "metadata": {"name": request.name, "namespace": ns},
                     "spec": spec})
                svc.spec = fresh.spec

        from ..kube.store import NotFoundError
        try:
            updated = self.client.update_with_retry(RayService, ns,
                                                    request.name, mutate)
        except NotFoundError:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"service {request.name} not found")
        return self._service_msg(updated)

    # -- ListAll* variants (cluster.proto:50, config.proto:50, job.proto:52,
    # serve.proto:60 — cross-namespace listings) -------------------------
    def list_all_clusters(self, request, context):
        from ..models import RayCluster
        out = ListClusterResponse()
        for rc in self.client.list(RayCluster, None):
            out.clusters.append(self._cluster_msg(rc))
        return out

    def list_all_ray_jobs(self, request, context):
        from ..models import RayJob
        out = ListRayJobResponse()
        for job in self.client.list(RayJob, None):
            out.jobs.append(self._job_msg(job))
        return out

    def list_all_ray_services(self, request, context):
        from ..models import RayService
        out = ListRayServiceResponse()
        for svc in self.client.list(RayService, None):
            out.services.append(self._service_msg(svc))
        return out

    def get_compute_template(self, request, context):
        t = self._templates(request.namespace or "default").get(request.name)
        if t is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"compute template {request.name} not found")
        return ComputeTemplate(
            name=t["name"], namespace=t.get("namespace") or "",
            cpu=t["cpu"], memory=t["memory"], gpu=t["gpu"],
            gpu_accelerator=t["gpuAccelerator"])

    def list_all_compute_templates(self, request, context):
        from ..kube import objects as k8s
        from . import converters as conv
        out = ListComputeTemplateResponse()
        for cm in self.client.list(k8s.ConfigMap, None):
            if (cm.metadata.labels or {}).get(conv.COMPUTE_TEMPLATE_LABEL):
                t = conv.configmap_to_compute_template(cm)
                out.compute_templates.append(ComputeTemplate(
                    name=t["name"], namespace=t.get("namespace") or "",
                    cpu=t["cpu"], memory=t["memory"], gpu=t["gpu"],
                    gpu_accelerator=t["gpuAccelerator"]))
        return out

    # -- image templates (config.proto:138-231 ImageTemplateService) -----
    IMAGE_TEMPLATE_LABEL = "ray.io/image-template"

    def create_image_template(self, request, context):
        from ..kube import objects as k8s
        ns = request.namespace or "default"
        cm = k8s.ConfigMap(
            metadata=k8s.ObjectMeta(
                name=f"imagetpl-{request.name}", namespace=ns,
                labels={self.IMAGE_TEMPLATE_LABEL: request.name}),
            data={"name": request.name,
                  "baseImage": request.base_image,
                  "pipPackages": json.dumps(list(request.pip_packages)),
                  "environmentVariables":
                      request.environment_variables_json or "{}"})
        self.client.create(cm)
        return request

    def _image_template_msg(self, cm):
        data = cm.data or {}
        msg = ImageTemplate(
            name=data.get("name", ""),
            namespace=cm.metadata.namespace or "",
            base_image=data.get("baseImage", ""),
            environment_variables_json=data.get("environmentVariables", ""))
        for p in json.loads(data.get("pipPackages", "[]")):
            msg.pip_packages.append(p)
        return msg

    def get_image_template(self, request, context):
        from ..kube import objects as k8s
        cm = self.client.try_get(k8s.ConfigMap, request.namespace or "default",
                                 f"imagetpl-{request.name}")
        if cm is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"image template {request.name} not found")
        return self._image_template_msg(cm)

    def list_image_templates(self, request, context):
        from ..kube import objects as k8s
        out = ListImageTemplateResponse()
        for cm in self.client.list(k8s.ConfigMap,
                                   request.namespace or "default"):
            if (cm.metadata.labels or {}).get(self.IMAGE_TEMPLATE_LABEL):
                out.image_templates.append(self._image_template_msg(cm))
        return out

    def delete_image_template(self, request, context):
        from ..kube import objects as k8s
        from ..kube.store import NotFoundError
        try:
            self.client.delete(k8s.ConfigMap, request.namespace or "default",
                               f"imagetpl-{request.name}")
        except NotFoundError:
            context.abort(grpc.StatusCode.NOT_FOUND, "not found")
        return Empty()

    # -- job submissions (job_submission.proto:26-70) ---------------------
    def submit_ray_job(self, request, context):
        import yaml
        dashboard = self._dashboard_for_cluster(
            context, request.namespace, request.clustername)
        sub = request.jobsubmission
        body = {"entrypoint": sub.entrypoint}
        if sub.submission_id:
            body["submission_id"] = sub.submission_id
        if sub.runtime_env:
            body["runtime_env"] = yaml.safe_load(sub.runtime_env)
        if sub.metadata_json:
            body["metadata"] = json.loads(sub.metadata_json)
        if sub.entrypoint_num_cpus:
            body["entrypoint_num_cpus"] = sub.entrypoint_num_cpus
        if sub.entrypoint_num_gpus:
            body["entrypoint_num_gpus"] = sub.entrypoint_num_gpus
        submission_id = dashboard.submit_job(body)
        return SubmitRayJobReply(submission_id=submission_id)

    def _submission_msg(self, info: Dict[str, Any]):
        return JobSubmissionInfo(
            submission_id=str(info.get("submission_id") or ""),
            status=str(info.get("status") or ""),
            entrypoint=str(info.get("entrypoint") or ""),
            message=str(info.get("message") or ""),
            error_type=str(info.get("error_type") or ""),
            start_time=str(info.get("start_time") or ""),
            end_time=str(info.get("end_time") or ""))

    def get_job_details(self, request, context):
        dashboard = self._dashboard_for_cluster(
            context, request.namespace, request.clustername)
        info = dashboard.get_job_info(request.submissionid)
        if info is None:
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"submission {request.submissionid} not found")
        return self._submission_msg(info)

    def list_job_details(self, request, context):
        dashboard = self._dashboard_for_cluster(
            context, request.namespace, request.clustername)
        out = ListJobSubmissionInfo()
        for info in dashboard.list_jobs():
            out.submissions.append(self._submission_msg(info))
        return out

    def get_job_log(self, request, context):
        dashboard = self._dashboard_for_cluster(
            context, request.namespace, request.clustername)
        log = dashboard.get_job_log(request.submissionid)
        return GetJobLogReply(log=log or "")

    def stop_ray_job_submission(self, request, context):
        dashboard = self._dashboard_for_cluster(
            context, request.namespace, request.clustername)
        dashboard.stop_job(request.submissionid)
        return Empty()

    def delete_ray_job_submission(self, request, context):
        dashboard = self._dashboard_for_cluster(
            context, request.namespace, request.clustername)
        delete = getattr(dashboard, "delete_job", None)
        if delete is not None:
            delete(request.submissionid)
        else:
            dashboard.stop_job(request.submissionid)
        return Empty()


def _unary(handler, req_cls, resp_cls):
    return grpc.unary_unary_rpc_method_handler(
        handler,
        request_deserializer=req_cls.FromString,
        response_serializer=lambda m: m.SerializeToString())


def create_grpc_server(client=None, port: int = 8887,
                       max_workers: int = 8,
                       dashboard_factory=None) -> grpc.Server:
    """Build (not start) the gRPC server with the six v1 services
    (cluster.proto, config.proto incl. ImageTemplateService, job.proto,
    job_submission.proto, serve.proto; error.proto shapes ride gRPC
    status)."""
    svc = _Service(client, dashboard_factory=dashboard_factory)
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    handlers = {
        "kuberayamd.v1.ClusterService": {
            "CreateCluster": _unary(svc.create_cluster, Cluster, Cluster),
            "GetCluster": _unary(svc.get_cluster, GetRequest, Cluster),
            "ListCluster": _unary(svc.list_clusters, ListRequest,
                                  ListClusterResponse),
            "ListAllClusters": _unary(svc.list_all_clusters, Empty,
                                      ListClusterResponse),
            "DeleteCluster": _unary(svc.delete_cluster, DeleteRequest, Empty),
        },
        "kuberayamd.v1.ComputeTemplateService": {
            "CreateComputeTemplate": _unary(svc.create_compute_template,
                                            ComputeTemplate, ComputeTemplate),
            "GetComputeTemplate": _unary(svc.get_compute_template,
                                         GetRequest, ComputeTemplate),
            "ListComputeTemplate": _unary(svc.list_compute_templates,
                                          ListRequest,
                                          ListComputeTemplateResponse),
            "ListAllComputeTemplates": _unary(svc.list_all_compute_templates,
                                              Empty,
                                              ListComputeTemplateResponse),
            "DeleteComputeTemplate": _unary(svc.delete_compute_template,
                                            DeleteRequest, Empty),
        },
        "kuberayamd.v1.ImageTemplateService": {
            "CreateImageTemplate": _unary(svc.create_image_template,
                                          ImageTemplate, ImageTemplate),
            "GetImageTemplate": _unary(svc.get_image_template, GetRequest,
                                       ImageTemplate),
            "ListImageTemplates": _unary(svc.list_image_templates,
                                         ListRequest,
                                         ListImageTemplateResponse),
            "DeleteImageTemplate": _unary(svc.delete_image_template,
                                          DeleteRequest, Empty),
        },
        "kuberayamd.v1.RayServeService": {
            "CreateRayService": _unary(svc.create_ray_service, RayServiceMsg,
                                       RayServiceMsg),
            "UpdateRayService": _unary(svc.update_ray_service, RayServiceMsg,
                                       RayServiceMsg),
            "GetRayService": _unary(svc.get_ray_service, GetRequest,
                                    RayServiceMsg),
            "ListRayServices": _unary(svc.list_ray_services, ListRequest,
                                      ListRayServiceResponse),
            "ListAllRayServices": _unary(svc.list_all_ray_services, Empty,
                                         ListRayServiceResponse),
            "DeleteRayService": _unary(svc.delete_ray_service, DeleteRequest,
                                       Empty),
        },
        "kuberayamd.v1.RayJobService": {
            "CreateRayJob": _unary(svc.create_ray_job, RayJobMsg, RayJobMsg),
            "GetRayJob": _unary(svc.get_ray_job, GetRequest, RayJobMsg),
            "ListRayJob": _unary(svc.list_ray_jobs, ListRequest,
                                 ListRayJobResponse),
            "ListAllRayJobs": _unary(svc.list_all_ray_jobs, Empty,
                                     ListRayJobResponse),
            "DeleteRayJob": _unary(svc.delete_ray_job, DeleteRequest, Empty),
        },
        "kuberayamd.v1.RayJobSubmissionService": {
            "SubmitRayJob": _unary(svc.submit_ray_job, SubmitRayJobRequest,
                                   SubmitRayJobReply),
            "GetJobDetails": _unary(svc.get_job_details, JobSubmissionQuery,
                                    JobSubmissionInfo),
            "GetJobLog": _unary(svc.get_job_log, JobSubmissionQuery,
                                GetJobLogReply),
            "ListJobDetails": _unary(svc.list_job_details,
                                     JobSubmissionQuery,
                                     ListJobSubmissionInfo),
            "StopRayJob": _unary(svc.stop_ray_job_submission,
                                 JobSubmissionQuery, Empty),
            "DeleteRayJob": _unary(svc.delete_ray_job_submission,
                                   JobSubmissionQuery, Empty),
        },
    }
    for service_name, methods in handlers.items():
        server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler(service_name, methods),))
    server.add_insecure_port(f"[::]:{port}")
    return server
