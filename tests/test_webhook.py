"""Mutating-webhook edge cases (reference webhook.go:47-83)."""
import base64
import json

from k8s_device_plugin_amd.scheduler.webhook import handle_admission_review


def review_for(pod_obj, uid="r1"):
    return handle_admission_review({
        "apiVersion": "admission.k8s.io/v1",
        "kind": "AdmissionReview",
        "request": {"uid": uid, "object": pod_obj, "kind": {"kind": "Pod"}},
    })


def patches_of(review):
    if "patch" not in review["response"]:
        return []
    return json.loads(base64.b64decode(review["response"]["patch"]))


def gpu_pod(extra_ctr=None, privileged=False):
    ctr = {
        "name": "main",
        "resources": {"limits": {"amd.com/gpu": "1"}},
    }
    if privileged:
        ctr["securityContext"] = {"privileged": True}
    ctrs = [ctr] + (extra_ctr or [])
    return {"kind": "Pod",
            "metadata": {"name": "p", "namespace": "default", "uid": "u"},
            "spec": {"containers": ctrs}}


def test_gpu_pod_gets_scheduler_name():
    r = review_for(gpu_pod())
    assert r["response"]["allowed"]
    assert any(p["path"] == "/spec/schedulerName" for p in patches_of(r))


def test_non_gpu_pod_untouched():
    pod = {"kind": "Pod",
           "metadata": {"name": "p", "namespace": "default", "uid": "u"},
           "spec": {"containers": [{"name": "web",
                                    "resources": {"limits": {"cpu": "1"}}}]}}
    r = review_for(pod)
    assert r["response"]["allowed"]
    assert not any(p["path"] == "/spec/schedulerName" for p in patches_of(r))


def test_privileged_container_skipped():
    """Privileged containers are not mutated (webhook.go:57-62): the pod
    keeps the default scheduler even though it requests the resource."""
    r = review_for(gpu_pod(privileged=True))
    assert r["response"]["allowed"]
    assert not any(p["path"] == "/spec/schedulerName" for p in patches_of(r))


def test_non_pod_object_allowed_untouched():
    r = handle_admission_review({
        "apiVersion": "admission.k8s.io/v1",
        "kind": "AdmissionReview",
        "request": {"uid": "x", "object": {"kind": "Deployment"},
                    "kind": {"kind": "Deployment"}},
    })
    assert r["response"]["allowed"]
    assert "patch" not in r["response"]


def test_priority_env_injected():
    pod = gpu_pod()
    pod["spec"]["containers"][0]["resources"]["limits"]["amd.com/priority"] = "1"
    r = review_for(pod)
    env_patches = [p for p in patches_of(r) if "/env" in p["path"]]
    assert env_patches, "priority limit must inject the task-priority env"
    flat = json.dumps(env_patches)
    assert "VGPU_TASK_PRIORITY" in flat


def test_uid_echoed():
    r = review_for(gpu_pod(), uid="abc-123")
    assert r["response"]["uid"] == "abc-123"
