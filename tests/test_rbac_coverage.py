"""RBAC coverage guard (reference analog: scripts/rbac-check.py — verify
the shipped ClusterRole grants every resource the operator code touches).

The operator's reachable kinds are exactly the keys of
kube/rest.py RESOURCES; the helm chart and the raw manifest must both
grant them (Events only need create/patch; everything else full CRUD for
the reconcilers' create/update/delete + owner GC)."""
import os

import yaml

from kuberay_amd.kube.rest import RESOURCES

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _rules_from(path):
    rules = []
    with open(path) as f:
        for doc in yaml.safe_load_all(f):
            if doc and doc.get("kind") in ("ClusterRole", "Role"):
                rules.extend(doc.get("rules") or [])
    return rules


def _granted(rules, group, plural, verb):
    for rule in rules:
        groups = rule.get("apiGroups") or []
        resources = rule.get("resources") or []
        verbs = rule.get("verbs") or []
        if ("*" in groups or group in groups) and \
                ("*" in resources or plural in resources) and \
                ("*" in verbs or verb in verbs):
            return True
    return False


def _group_of(prefix: str) -> str:
    if prefix == "/api/v1":
        return ""
    return prefix.split("/apis/")[1].split("/")[0]


READ_ONLY_OK = {"Node", "EndpointSlice"}        # operator only watches these
CREATE_ONLY_OK = {"Event"}                       # recorder creates/patches


def _check(path):
    rules = _rules_from(path)
    missing = []
    for kind, (prefix, plural) in RESOURCES.items():
        group = _group_of(prefix)
        if kind in CREATE_ONLY_OK:
            needed = ["create", "patch"]
        elif kind in READ_ONLY_OK:
            needed = ["get", "list", "watch"]
        else:
            needed = ["get", "list", "watch", "create", "update", "patch",
                      "delete"]
        for verb in needed:
            if not _granted(rules, group, plural, verb):
                missing.append(f"{group or 'core'}/{plural}: {verb}")
    assert not missing, f"{path} missing grants: {sorted(set(missing))}"


def test_helm_clusterrole_covers_operator_surface():
    _check(os.path.join(HERE, "deploy", "helm", "kuberay-amd-operator",
                        "templates", "rbac.yaml"))


def test_manifest_rbac_covers_operator_surface():
    _check(os.path.join(HERE, "deploy", "manifests", "operator.yaml"))


def test_committed_crds_match_generator():
    """Drift guard (generate-crd-schema.sh analog): deploy/crds must equal
    what kuberay_amd.crds generates from the pydantic models."""
    import tempfile

    from kuberay_amd import crds as crdgen

    with tempfile.TemporaryDirectory() as tmp:
        crdgen.write_crds(tmp)
        for name in os.listdir(tmp):
            generated = open(os.path.join(tmp, name)).read()
            committed_path = os.path.join(HERE, "deploy", "crds", name)
            assert os.path.exists(committed_path), f"{name} not committed"
            assert open(committed_path).read() == generated, \
                f"{name} drifted — run: python -m kuberay_amd.crds deploy/crds"
