"""Mutating admission webhook.

Reference behavior (/root/reference/pkg/scheduler/webhook.go:47-83): for each
pod CREATE, if any container requests a vGPU resource (and the pod is not
privileged), set ``spec.schedulerName`` so the vanilla kube-scheduler defers
to our extender; also inject the priority env (device.mutate_admission).
Responds with a JSONPatch AdmissionReview.
"""
from __future__ import annotations

import base64
import json
import logging
from typing import Optional

from ..device import get_devices
from ..utils.types import PodInfo

log = logging.getLogger(__name__)


def handle_admission_review(review: dict, scheduler_name: str = "vgpu-scheduler") -> dict:
    req = review.get("request", {}) or {}
    uid = req.get("uid", "")
    pod_obj = (req.get("object") or {})
    resp = {
        "apiVersion": review.get("apiVersion", "admission.k8s.io/v1"),
        "kind": "AdmissionReview",
        "response": {"uid": uid, "allowed": True},
    }
    if pod_obj.get("kind", "Pod") != "Pod" and req.get("kind", {}).get("kind") != "Pod":
        return resp

    pod = PodInfo.from_k8s(pod_obj)
    patches = []
    has_resource = False
    for i, ctr in enumerate(pod.containers):
        if ctr.security_privileged:
            log.info("pod %s container %s is privileged; skipping mutation", pod.name, ctr.name)
            continue
        before_env = dict(ctr.env)
        for dev in get_devices().values():
            if dev.mutate_admission(ctr):
                has_resource = True
        # env injections made by mutate_admission become JSONPatches
        for k, v in ctr.env.items():
            if k not in before_env:
                spec_ctrs = (pod_obj.get("spec", {}) or {}).get("containers", [])
                existing_env = (spec_ctrs[i].get("env") if i < len(spec_ctrs) else None) or []
                if not existing_env:
                    patches.append(
                        {
                            "op": "add",
                            "path": f"/spec/containers/{i}/env",
                            "value": [{"name": k, "value": str(v)}],
                        }
                    )
                else:
                    patches.append(
                        {
                            "op": "add",
                            "path": f"/spec/containers/{i}/env/-",
                            "value": {"name": k, "value": str(v)},
                        }
                    )
    if has_resource:
        patches.append(
            {"op": "add", "path": "/spec/schedulerName", "value": scheduler_name}
        )
    if patches:
        resp["response"]["patchType"] = "JSONPatch"
        resp["response"]["patch"] = base64.b64encode(
            json.dumps(patches).encode()
        ).decode()
    return resp
