"""Property-based invariants (hypothesis): whatever random workload the
scheduler sees, deleting every pod must restore the pristine cluster —
no leaked availability, memory, or ports — and the label parser must
never crash on arbitrary input."""
import string

from hypothesis import given, settings, strategies as st

from kubeshare_amd.scheduler.harness import FakeCluster
from kubeshare_amd.utils import constants as C
from kubeshare_amd.utils.labels import LabelError, parse_pod

FRACS = ["0.1", "0.2", "0.25", "0.3", "0.33", "0.5", "0.75", "1.0"]


def pod_labels_strategy():
    shared = st.builds(
        lambda req, lim, prio: {
            C.POD_GPU_REQUEST: req, C.POD_GPU_LIMIT: lim,
            **({C.POD_PRIORITY: prio} if prio else {})},
        st.sampled_from(FRACS), st.sampled_from(["1.0"]),
        st.sampled_from(["", "0", "50", "100"]))
    multi = st.builds(
        lambda n: {C.POD_GPU_REQUEST: f"{n}.0", C.POD_GPU_LIMIT: f"{n}.0"},
        st.integers(min_value=2, max_value=4))
    limit_only = st.just({C.POD_GPU_LIMIT: "1.0"})
    return st.one_of(shared, multi, limit_only)


@settings(max_examples=40, deadline=None)
@given(st.lists(pod_labels_strategy(), min_size=1, max_size=16),
       st.randoms())
def test_schedule_then_delete_restores_pristine_state(workload, rnd):
    fc = FakeCluster(nodes={"node-a": {"gpus": 4}, "node-b": {"gpus": 4}})
    tree = fc.scheduler.tree
    pods = [fc.add_pod("p", f"pod{i}", labels)
            for i, labels in enumerate(workload)]
    fc.schedule_pending()
    # delete in random order, interleaved with rescheduling attempts
    order = list(pods)
    rnd.shuffle(order)
    for i, pod in enumerate(order):
        fc.delete_pod(pod.key)
        if i % 3 == 0:
            fc.schedule_pending()
    for node in ("node-a", "node-b"):
        for leaf in tree.leaves_on_node(node):
            assert leaf.available == 1.0, (leaf.id, leaf.available)
            assert leaf.available_whole == 1
            assert leaf.free_memory == leaf.full_memory, leaf.id
        pool = fc.scheduler.ports[node]
        assert all(not u for u in pool.used), "leaked manager port"
    assert fc.scheduler.waiting == {}


@settings(max_examples=40, deadline=None)
@given(st.lists(pod_labels_strategy(), min_size=1, max_size=12))
def test_never_oversubscribe_requests(workload):
    """However pods are packed, the sum of reserved fractions per leaf
    never exceeds 1.0 (requests are guarantees)."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    for i, labels in enumerate(workload):
        fc.add_pod("q", f"pod{i}", labels)
    fc.schedule_pending()
    for leaf in fc.scheduler.tree.leaves_on_node("node-a"):
        assert leaf.available >= 0.0, (leaf.id, leaf.available)
        assert leaf.free_memory >= 0


_label_text = st.text(
    alphabet=string.ascii_letters + string.digits + "./-_", max_size=12)


@settings(max_examples=200, deadline=None)
@given(st.dictionaries(
    st.sampled_from([C.POD_GPU_REQUEST, C.POD_GPU_LIMIT, C.POD_GPU_MEMORY,
                     C.POD_PRIORITY, C.POD_GPU_MODEL, C.POD_GROUP_NAME,
                     C.POD_GROUP_HEADCOUNT, C.POD_GROUP_THRESHOLD,
                     C.POD_MIN_AVAILABLE]),
    _label_text, max_size=6))
def test_label_parser_never_crashes(labels):
    """Arbitrary label soup: parse_pod returns a spec, None, or raises
    LabelError — never anything else; valid specs obey the invariants."""
    try:
        spec = parse_pod("ns", "p", labels)
    except LabelError:
        return
    if spec is None:
        return
    assert spec.limit > 0
    assert 0 <= spec.request <= spec.limit
    if spec.limit > 1.0:
        assert spec.limit == spec.request
        assert float(spec.limit).is_integer()
    assert spec.memory >= 0
    assert -1 <= spec.priority <= 100


_k8s_name = st.from_regex(r"[a-z0-9]([a-z0-9.-]{0,10}[a-z0-9])?",
                          fullmatch=True)


@settings(max_examples=60, deadline=None)
@given(st.lists(
    st.builds(lambda ns, n, lim, req, mem, grp, lease: (f"{ns}/{n}", lim,
                                                        req, mem, grp,
                                                        lease),
              _k8s_name, _k8s_name,
              st.floats(min_value=0.01, max_value=1.0),
              st.floats(min_value=0.0, max_value=1.0),
              st.integers(min_value=0, max_value=2**48),
              st.one_of(st.just(""), _k8s_name),
              st.integers(min_value=0, max_value=10000)),
    min_size=0, max_size=8))
def test_config_file_roundtrip_property(tmp_path_factory, entries):
    """Arbitrary (k8s-legal) pods/groups/values survive the per-UUID
    file write/read round trip bit-exactly enough for scheduling
    (request/limit to 1e-6, memory/group/lease exactly)."""
    from kubeshare_amd.configdaemon import files as F
    tmp = tmp_path_factory.mktemp("cfgprop")
    # PodQuota(pod, limit, request, memory, group, lease); request<=limit
    quotas = [F.PodQuota(pod, lim, min(req, lim), mem, group=grp,
                         lease_ms=lease)
              for pod, lim, req, mem, grp, lease in entries]
    path = F.write_gpu_config(str(tmp), "GPU-prop", quotas)
    back = F.read_gpu_config(path)
    assert len(back) == len(quotas)
    for a, b in zip(quotas, back):
        assert a.pod == b.pod
        assert abs(a.limit - b.limit) < 1e-6
        assert abs(a.request - b.request) < 1e-6
        assert a.memory == b.memory
        assert a.group == b.group
        assert a.lease_ms == b.lease_ms


_env_name = st.from_regex(r"[A-Z][A-Z0-9_]{0,12}", fullmatch=True)
_container = st.builds(
    lambda name, env, mounts: {
        "name": name,
        **({"env": env} if env is not None else {}),
        **({"volumeMounts": mounts} if mounts is not None else {})},
    _k8s_name,
    st.one_of(st.none(), st.lists(st.builds(
        lambda n, v: {"name": n, "value": v}, _env_name, _k8s_name),
        max_size=4)),
    st.one_of(st.none(), st.lists(st.builds(
        lambda n: {"name": n, "mountPath": "/" + n}, _k8s_name),
        max_size=2)))


@settings(max_examples=60, deadline=None)
@given(st.lists(_container, min_size=1, max_size=3),
       st.sampled_from(["50050", "50300", ""]),
       st.one_of(st.none(), st.lists(st.builds(
           lambda n: {"name": n}, _k8s_name), max_size=2)))
def test_webhook_patch_idempotent_property(containers, port, volumes):
    """For ANY container/env/volume shape: the webhook patch applies
    cleanly (RFC-6902 paths valid against the doc) and a second
    invocation is a no-op."""
    from kubeshare_amd.testing.fake_apiserver import apply_json_patch
    from kubeshare_amd.webhook import build_patch
    pod = {"metadata": {"name": "p", "namespace": "ns",
                        "annotations": {C.POD_GPU_UUID: "GPU-1",
                                        C.POD_GPU_INDEX: "0",
                                        C.POD_GPU_MEMORY: "5",
                                        **({C.POD_MANAGER_PORT: port}
                                           if port else {})}},
           "spec": {"containers": containers,
                    **({"volumes": volumes} if volumes is not None
                       else {})}}
    patch = build_patch(pod)
    mutated = apply_json_patch(pod, patch)     # must not raise
    assert build_patch(mutated) == []          # idempotent
    for c in mutated["spec"]["containers"]:
        names = [e["name"] for e in c["env"]]
        assert names.count(C.ENV_ROCR_VISIBLE_DEVICES) == 1
        assert C.ENV_INJECTED in names
        if port:
            assert names.count("POD_MANAGER_UDS") == 1


# ---------------------------------------------------------------- aggregator
_junk = st.text(string.printable, max_size=12)


@given(labels=st.dictionaries(st.sampled_from([
           C.POD_GPU_LIMIT, C.POD_GPU_REQUEST, C.POD_MIN_AVAILABLE,
           C.POD_LEASE_MS, C.POD_GROUP_NAME]), _junk, max_size=5),
       ann=st.dictionaries(st.sampled_from([
           C.POD_GPU_UUID, C.POD_GPU_MEMORY, C.POD_MANAGER_PORT,
           C.POD_CELL_ID]), _junk, max_size=4))
@settings(max_examples=200, deadline=None)
def test_demand_from_pod_never_raises(labels, ann):
    """Scrape safety: arbitrary junk in labels/annotations yields a
    PodDemand or None, never an exception (one bad pod must not 500
    the aggregator endpoint)."""
    from kubeshare_amd.aggregator import PodDemand, demand_from_pod

    class P:
        namespace, name, uid, node = "ns", "p", "u", "n"
        env = {}

    P.labels, P.annotations = labels, ann
    d = demand_from_pod(P())
    assert d is None or isinstance(d, PodDemand)
