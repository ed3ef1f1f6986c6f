"""Golden wire-format tests: pin the runtime-built descriptors to the
exact proto3 bytes a kubelet's generated code produces (field numbers and
types from k8s.io/kubelet/pkg/apis/dra/v1beta1/api.proto)."""

from k8s_dra_driver_amd.plugin.proto import REGISTRATION, V1ALPHA4, V1BETA1


def test_claim_wire_bytes():
    c = V1BETA1.Claim(namespace="ns", uid="u1", name="c1")
    # field 1 (namespace) LEN "ns"; field 2 (uid) LEN "u1"; field 3 (name)
    assert c.SerializeToString() == b"\x0a\x02ns\x12\x02u1\x1a\x02c1"


def test_device_wire_bytes():
    d = V1BETA1.Device(
        request_names=["r"],
        pool_name="p",
        device_name="d",
        cdi_device_ids=["x=y"],
    )
    assert (
        d.SerializeToString()
        == b"\x0a\x01r\x12\x01p\x1a\x01d\x22\x03x=y"
    )


def test_prepare_response_map_encoding():
    resp = V1BETA1.NodePrepareResourcesResponse()
    one = V1BETA1.NodePrepareResourceResponse(error="boom")
    resp.claims["uid"].CopyFrom(one)
    data = resp.SerializeToString()
    # map<string, M> encodes as repeated entries: field 1 LEN(13), then
    # key (field 1, "uid") and value (field 2: submessage with error
    # field 2 = "boom")
    assert data == b"\x0a\x0d\x0a\x03uid\x12\x06\x12\x04boom"
    # round trip through the OTHER version's class: identical wire format
    again = V1ALPHA4.NodePrepareResourcesResponse.FromString(data)
    assert again.claims["uid"].error == "boom"


def test_registration_info_wire():
    info = REGISTRATION.PluginInfo(
        type="DRAPlugin",
        name="gpu.amd.com",
        endpoint="/s.sock",
        supported_versions=["v1beta1"],
    )
    data = info.SerializeToString()
    assert data == (
        b"\x0a\x09DRAPlugin"
        b"\x12\x0bgpu.amd.com"
        b"\x1a\x07/s.sock"
        b"\x22\x07v1beta1"
    )


def test_cross_version_compatibility():
    """v1alpha4 and v1beta1 messages are wire-identical (only the package
    and service names differ), so a kubelet speaking either decodes us."""
    req = V1BETA1.NodePrepareResourcesRequest()
    c = req.claims.add()
    c.namespace, c.name, c.uid = "a", "b", "c"
    alpha = V1ALPHA4.NodePrepareResourcesRequest.FromString(
        req.SerializeToString()
    )
    assert alpha.claims[0].namespace == "a"
    assert alpha.claims[0].uid == "c"
