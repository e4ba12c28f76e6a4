"""Unit tests for the stdlib REST client's object model and selector
matching (kubeshare_amd/scheduler/kubeclient.py)."""
from kubeshare_amd.scheduler.kubeclient import (K8sObj, _camel,
                                                _match_fields,
                                                _match_labels)


def test_camel_case_mapping():
    assert _camel("node_name") == "nodeName"
    assert _camel("creation_timestamp") == "creationTimestamp"
    assert _camel("scheduler_name") == "schedulerName"
    assert _camel("name") == "name"


def test_attr_read_write_through():
    pod = {"metadata": {"name": "p", "labels": {"a": "1"}},
           "spec": {"nodeName": "n1",
                    "containers": [{"name": "c", "env": None}]}}
    o = K8sObj(pod)
    assert o.metadata.name == "p"
    assert o.spec.node_name == "n1"
    assert o.status is None                      # absent key
    # write-through: mutating a nested wrapper hits the original dict
    c = o.spec.containers[0]
    c.env = [{"name": "X", "value": "1"}]
    assert pod["spec"]["containers"][0]["env"] == [
        {"name": "X", "value": "1"}]
    o.spec.node_name = "n2"
    assert pod["spec"]["nodeName"] == "n2"
    # None deletes the key (resource_version nulling on shadow pods)
    o.metadata.resource_version = None
    assert "resourceVersion" not in pod["metadata"]


def test_dict_protocol_on_maps():
    o = K8sObj({"labels": {"a": "1", "b": "2"}, "empty": {}})
    labels = o.labels
    assert dict(labels.items()) == {"a": "1", "b": "2"}
    assert "a" in labels and labels["b"] == "2"
    assert len(labels) == 2 and sorted(labels) == ["a", "b"]
    assert not o.empty          # empty map is falsy (dict semantics)
    assert (o.empty or {}) == {}


def test_label_selector_matching():
    obj = {"metadata": {"labels": {"SharedGPU": "true", "team": "ml"}}}
    assert _match_labels(obj, "SharedGPU=true")
    assert _match_labels(obj, "SharedGPU=true,team=ml")
    assert not _match_labels(obj, "SharedGPU=false")
    assert not _match_labels(obj, "missing=1")
    assert _match_labels(obj, "team!=infra")
    assert not _match_labels(obj, "team!=ml")
    assert _match_labels(obj, "SharedGPU")       # existence
    assert _match_labels(obj, "")                # no selector


def test_field_selector_matching():
    obj = {"status": {"phase": "Pending"}, "spec": {"nodeName": "a"}}
    assert _match_fields(obj, "status.phase=Pending")
    assert not _match_fields(obj, "status.phase=Running")
    assert _match_fields(obj, "spec.nodeName=a,status.phase=Pending")
    assert not _match_fields(obj, "spec.missing=x")
