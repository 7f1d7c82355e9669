"""Typed CRD validation (aigw/controller/crds.py) — the kube-apiserver
CRD/CEL gate analogue (reference: tests/crdcel/main_test.go + the
XValidation markers in api/v1beta1). The reference's own example bundles
must validate; each mirrored rule must reject its violation with a
field-qualified message."""

import glob
import os

import pytest
import yaml

from aigw.controller.crds import CRDValidationError, validate_bundle, validate_crd
from aigw.filterapi.config import ConfigError
from aigw.controller import translate_yaml

REF_EXAMPLES = "/root/reference/examples"


@pytest.mark.skipif(not os.path.isdir(REF_EXAMPLES),
                    reason="reference checkout not present")
def test_reference_example_bundles_validate():
    n_total = 0
    for f in glob.glob(os.path.join(REF_EXAMPLES, "*", "*.yaml")):
        docs = [d for d in yaml.safe_load_all(open(f)) if isinstance(d, dict)]
        n_total += validate_bundle(docs)
    assert n_total >= 20, n_total


def test_repo_example_bundle_validates():
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    text = open(os.path.join(root, "examples", "crd_bundle.yaml")).read()
    docs = [d for d in yaml.safe_load_all(text) if isinstance(d, dict)]
    assert validate_bundle(docs) >= 3


def _route(spec):
    return {"apiVersion": "aigateway.envoyproxy.io/v1beta1",
            "kind": "AIGatewayRoute",
            "metadata": {"name": "r"}, "spec": spec}


def test_kubebuilder_rules_mirrored():
    ok = _route({
        "rules": [{"name": "a", "backendRefs": [{"name": "b", "weight": 1}]},
                  {"name": "b", "backendRefs": [{"name": "c", "priority": 1}]}],
    })
    assert validate_crd(ok) is not None

    cases = [
        # targetRefs only Gateway
        (_route({"targetRefs": [{"kind": "HTTPRoute", "name": "x"}]}),
         "only Gateway"),
        # duplicate rule names
        (_route({"rules": [{"name": "a"}, {"name": "a"}]}), "unique"),
        # reserved rule name
        (_route({"rules": [{"name": "route-not-found"}]}), "reserved"),
        # group without kind
        (_route({"rules": [{"backendRefs": [{"name": "b",
                                             "group": "inference.networking.k8s.io"}]}]}),
         "together"),
        # non-InferencePool group/kind
        (_route({"rules": [{"backendRefs": [{"name": "b", "group": "x",
                                             "kind": "Y"}]}]}),
         "InferencePool"),
        # mixed pool + service backend refs
        (_route({"rules": [{"backendRefs": [
            {"name": "p", "group": "inference.networking.k8s.io",
             "kind": "InferencePool"},
            {"name": "b"}]}]}), "cannot mix"),
        # negative weight
        (_route({"rules": [{"backendRefs": [{"name": "b", "weight": -1}]}]}),
         "weight"),
        # priority out of range
        (_route({"rules": [{"backendRefs": [{"name": "b",
                                             "priority": 70000}]}]}),
         "priority"),
    ]
    for doc, needle in cases:
        with pytest.raises(CRDValidationError) as e:
            validate_crd(doc)
        assert needle.lower() in str(e.value).lower() or needle in str(e.value), (
            needle, str(e.value))


def test_bsp_exactly_one_credential():
    def bsp(type_, extra):
        spec = {"type": type_,
                "targetRefs": [{"kind": "AIServiceBackend", "name": "b"}]}
        spec.update(extra)
        return {"kind": "BackendSecurityPolicy", "metadata": {"name": "p"},
                "spec": spec}

    assert validate_crd(bsp("APIKey", {"apiKey": {"secretRef": {"name": "s"}}}))
    # matching field missing
    with pytest.raises(CRDValidationError, match="must be set"):
        validate_crd(bsp("APIKey", {}))
    # extra field of another type
    with pytest.raises(CRDValidationError, match="only apiKey"):
        validate_crd(bsp("APIKey", {"apiKey": {"secretRef": {"name": "s"}},
                                    "awsCredentials": {"region": "us-east-1"}}))
    with pytest.raises(CRDValidationError, match="unknown"):
        validate_crd(bsp("Telepathy", {}))


def test_asb_backend_ref_kind():
    def asb(ref):
        return {"kind": "AIServiceBackend", "metadata": {"name": "b"},
                "spec": {"schema": {"name": "OpenAI"}, "backendRef": ref}}

    assert validate_crd(asb({"name": "x", "kind": "Backend",
                             "group": "gateway.envoyproxy.io"}))
    with pytest.raises(CRDValidationError, match="Backend resource"):
        validate_crd(asb({"name": "x", "kind": "HTTPRoute", "group": "g"}))
    with pytest.raises(CRDValidationError, match="schema"):
        validate_crd({"kind": "AIServiceBackend", "metadata": {"name": "b"},
                      "spec": {"schema": {"name": "Telepathy"},
                               "backendRef": {"name": "x"}}})


def test_translate_rejects_invalid_bundle():
    bad = """
apiVersion: aigateway.envoyproxy.io/v1beta1
kind: AIGatewayRoute
metadata: {name: r}
spec:
  rules:
    - name: route-not-found
"""
    with pytest.raises(ConfigError, match="reserved"):
        translate_yaml(bad)
