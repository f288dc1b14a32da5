"""RayCronJob reconciler (reference: raycronjob_controller.go:58-217).

Parses the cron schedule, detects missed fire times and creates RayJobs
named ``{cron}-{minutehash}`` so a given minute fires at most once.
"""
from __future__ import annotations

import datetime as dt
import hashlib
import logging
from typing import Optional

from ..kube import objects as k8s
from ..kube.client import KubeClient
from ..kube.controller import Reconciler, Request, Result
from ..kube.events import EventRecorder, NullRecorder
from ..kube.store import AlreadyExistsError
from ..models import RayCronJob, RayJob
from ..utils import constants as C
from ..utils.cron import parse_cron
from ..utils.validation import validate_raycronjob_spec

logger = logging.getLogger("kuberay.raycronjob")


def _minute_hash(t: dt.datetime) -> str:
    return hashlib.sha1(t.strftime("%Y%m%d%H%M").encode()).hexdigest()[:8]


class RayCronJobReconciler(Reconciler):
    def __init__(self, client: KubeClient, recorder: Optional[EventRecorder] = None,
                 now_fn=None):
        self.client = client
        self.recorder = recorder or NullRecorder()
        self.now_fn = now_fn or (lambda: dt.datetime.utcnow())

    def reconcile(self, request: Request) -> Result:
        namespace, name = request
        cron = self.client.try_get(RayCronJob, namespace, name)
        if cron is None or cron.metadata.deletion_timestamp:
            return Result()

        errs = validate_raycronjob_spec(cron)
        if errs:
            self.recorder.eventf(cron, "Warning", "InvalidRayCronJobSpec", "; ".join(errs))
            return Result()
        if cron.spec.suspend:
            return Result()

        schedule = parse_cron(cron.spec.schedule)
        now = self.now_fn()

        # spec.timeZone: the schedule's wall-clock fields are interpreted in
        # that IANA zone (CronJob semantics); stored timestamps stay UTC
        tz = None
        if cron.spec.time_zone:
            from zoneinfo import ZoneInfo
            tz = ZoneInfo(cron.spec.time_zone)

        def to_local(t_utc: dt.datetime) -> dt.datetime:
            if tz is None:
                return t_utc
            return t_utc.replace(tzinfo=dt.timezone.utc).astimezone(tz) \
                .replace(tzinfo=None)

        def to_utc(t_local: dt.datetime) -> dt.datetime:
            if tz is None:
                return t_local
            return t_local.replace(tzinfo=tz).astimezone(dt.timezone.utc) \
                .replace(tzinfo=None)

        last = None
        if cron.status.last_schedule_time:
            try:
                last = dt.datetime.strptime(cron.status.last_schedule_time,
                                            "%Y-%m-%dT%H:%M:%SZ")
            except ValueError:
                last = None
        basis = last or (dt.datetime.strptime(cron.metadata.creation_timestamp,
                                              "%Y-%m-%dT%H:%M:%SZ")
                         if cron.metadata.creation_timestamp else now)

        fire_local = schedule.next_after(to_local(basis))
        if fire_local is None:
            return Result()
        fire = to_utc(fire_local)
        if fire > now:
            return Result(requeue_after=min((fire - now).total_seconds(), 300))

        # fire (catch up at most the most recent missed tick, like CronJob
        # with startingDeadline unbounded collapsed to latest)
        latest_local = fire_local
        while True:
            nxt = schedule.next_after(latest_local)
            if nxt is None or to_utc(nxt) > now:
                break
            latest_local = nxt
        latest = to_utc(latest_local)

        job_name = f"{cron.metadata.name}-{_minute_hash(latest)}"
        rayjob = RayJob(
            metadata=k8s.ObjectMeta(
                name=job_name,
                namespace=namespace,
                labels={C.RAY_CRONJOB_NAME_LABEL_KEY: cron.metadata.name},
                annotations={C.RAY_CRONJOB_TIMESTAMP_ANNOTATION_KEY:
                             latest.strftime("%Y-%m-%dT%H:%M:%SZ")},
                owner_references=[k8s.owner_reference_for(cron)],
            ),
            spec=cron.spec.job_template.clone(),
        )
        try:
            self.client.create(rayjob)
            self.recorder.eventf(cron, "Normal", "CreatedRayJob",
                                 "Created RayJob %s for schedule fire at %s",
                                 job_name, latest.isoformat())
        except AlreadyExistsError:
            pass
        cron.status.last_schedule_time = latest.strftime("%Y-%m-%dT%H:%M:%SZ")
        self.client.update_status(cron)
        nxt = schedule.next_after(now)
        delay = (nxt - now).total_seconds() if nxt else 300
        return Result(requeue_after=min(delay, 300))
