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


# ---------------------------------------------------------------------------
# service implementation over the same converters as the HTTP surface
# ---------------------------------------------------------------------------

class _Service:
    """Shared backend for the three gRPC services."""

    def __init__(self, client):
        from ..kube.client import InMemoryClient
        self.client = client or InMemoryClient()

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


def _unary(handler, req_cls, resp_cls):
    return grpc.unary_unary_rpc_method_handler(
        handler,
        request_deserializer=req_cls.FromString,
        response_serializer=lambda m: m.SerializeToString())


def create_grpc_server(client=None, port: int = 8887,
                       max_workers: int = 8) -> grpc.Server:
    """Build (not start) the gRPC server with the three services."""
    svc = _Service(client)
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    handlers = {
        "kuberayamd.v1.ClusterService": {
            "CreateCluster": _unary(svc.create_cluster, Cluster, Cluster),
            "GetCluster": _unary(svc.get_cluster, GetRequest, Cluster),
            "ListCluster": _unary(svc.list_clusters, ListRequest,
                                  ListClusterResponse),
            "DeleteCluster": _unary(svc.delete_cluster, DeleteRequest, Empty),
        },
        "kuberayamd.v1.ComputeTemplateService": {
            "CreateComputeTemplate": _unary(svc.create_compute_template,
                                            ComputeTemplate, ComputeTemplate),
            "ListComputeTemplate": _unary(svc.list_compute_templates,
                                          ListRequest,
                                          ListComputeTemplateResponse),
            "DeleteComputeTemplate": _unary(svc.delete_compute_template,
                                            DeleteRequest, Empty),
        },
        "kuberayamd.v1.RayServeService": {
            "CreateRayService": _unary(svc.create_ray_service, RayServiceMsg,
                                       RayServiceMsg),
            "GetRayService": _unary(svc.get_ray_service, GetRequest,
                                    RayServiceMsg),
            "ListRayServices": _unary(svc.list_ray_services, ListRequest,
                                      ListRayServiceResponse),
            "DeleteRayService": _unary(svc.delete_ray_service, DeleteRequest,
                                       Empty),
        },
        "kuberayamd.v1.RayJobService": {
            "CreateRayJob": _unary(svc.create_ray_job, RayJobMsg, RayJobMsg),
            "GetRayJob": _unary(svc.get_ray_job, GetRequest, RayJobMsg),
            "ListRayJob": _unary(svc.list_ray_jobs, ListRequest,
                                 ListRayJobResponse),
            "DeleteRayJob": _unary(svc.delete_ray_job, DeleteRequest, Empty),
        },
    }
    for service_name, methods in handlers.items():
        server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler(service_name, methods),))
    server.add_insecure_port(f"[::]:{port}")
    return server
