"""Label parsing/validation parity with the reference's accept/reject
behavior (pkg/scheduler/pod.go:175-327) — the cases mirror the e2e pods
test/pod1..16.yaml the reference uses as its manual test suite."""
import pytest

from kubeshare_amd.utils import constants as C
from kubeshare_amd.utils.labels import LabelError, parse_pod


def mk(labels):
    return parse_pod("default", "p", labels)


def test_integer_whole_gpu():  # test/pod1.yaml
    s = mk({C.POD_GPU_LIMIT: "2.0", C.POD_GPU_REQUEST: "2.0"})
    assert s.is_multi_gpu and s.limit == 2.0 and s.request == 2.0


def test_fractional():  # test/pod4.yaml
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_GPU_REQUEST: "0.3"})
    assert s.is_shared and s.request == pytest.approx(0.3)


def test_limit_below_request_rejected():  # test/pod8.yaml
    with pytest.raises(LabelError):
        mk({C.POD_GPU_LIMIT: "0.3", C.POD_GPU_REQUEST: "0.5"})


def test_multi_gpu_requires_equal_limit_request():
    with pytest.raises(LabelError):
        mk({C.POD_GPU_LIMIT: "3.0", C.POD_GPU_REQUEST: "2.0"})


def test_non_integer_above_one_rejected():
    with pytest.raises(LabelError):
        mk({C.POD_GPU_LIMIT: "1.5", C.POD_GPU_REQUEST: "1.5"})


def test_regular_pod_no_labels():
    assert mk({}) is None


def test_regular_pod_zero_values():  # pod.go:303-305
    assert mk({C.POD_GPU_LIMIT: "0.0", C.POD_GPU_REQUEST: "0.0"}) is None


def test_request_without_limit_rejected():  # pod.go:294-300
    with pytest.raises(LabelError):
        mk({C.POD_GPU_REQUEST: "0.5"})


def test_priority_domain():  # pod.go:179-199
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_GPU_REQUEST: "0.5",
            C.POD_PRIORITY: "100"})
    assert not s.is_opportunistic
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_GPU_REQUEST: "0.5"})
    assert s.is_opportunistic
    with pytest.raises(LabelError):
        mk({C.POD_GPU_LIMIT: "1.0", C.POD_PRIORITY: "101"})
    with pytest.raises(LabelError):
        mk({C.POD_GPU_LIMIT: "1.0", C.POD_PRIORITY: "-2"})
    # -1 is inside the reference's accepted range (pod.go:192) and is
    # opportunistic
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_PRIORITY: "-1"})
    assert s.is_opportunistic


def test_garbage_values_rejected():
    for bad in ["abc", "0.5x", ".5", "00", "1.", "2.5"]:
        with pytest.raises(LabelError):
            mk({C.POD_GPU_LIMIT: bad})


def test_memory_label():
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_GPU_REQUEST: "0.5",
            C.POD_GPU_MEMORY: str(10 * 2**30)})
    assert s.memory == 10 * 2**30
    with pytest.raises(LabelError):
        mk({C.POD_GPU_LIMIT: "1.0", C.POD_GPU_MEMORY: "-5"})


def test_default_memory_is_request_share():  # pod.go:419-421
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_GPU_REQUEST: "0.5"})
    assert s.default_memory() == C.MI355X_HBM_BYTES // 2


def test_gang_labels_two_vocabularies():
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_GROUP_NAME: "g",
            C.POD_GROUP_HEADCOUNT: "4", C.POD_GROUP_THRESHOLD: "0.6"})
    assert s.pod_group == "g" and s.min_available == 2  # floor(2.4+0.5)
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_GROUP_NAME: "g",
            C.POD_MIN_AVAILABLE: "3"})
    assert s.min_available == 3


def test_model_pinning():
    s = mk({C.POD_GPU_LIMIT: "1.0", C.POD_GPU_MODEL: C.MI355X_MODEL})
    assert s.model == C.MI355X_MODEL
