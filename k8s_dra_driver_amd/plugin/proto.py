"""Runtime-constructed protobuf messages for the kubelet DRA + registration APIs.

The wire contracts are Kubernetes' published protos (shapes documented in
the reference's vendor tree: ``k8s.io/kubelet/pkg/apis/dra/v1beta1/api.proto``
and ``pluginregistration/v1/api.proto``; v1alpha4 is identical to v1beta1
except package ``v1alpha3`` / service ``Node``). No ``grpc_tools`` protoc is
available in this environment, so the descriptors are built at runtime with
``google.protobuf.descriptor_pb2`` — the wire format is identical to
generated code.

Exposes message classes plus the gRPC method paths for both DRA service
versions; the plugin registers both, like the reference's vendored
kubeletplugin does (``draplugin.go:342-350``).
"""

from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

DRA_V1_PKG = "k8s.io.kubelet.pkg.apis.dra.v1"
DRA_V1BETA1_PKG = "k8s.io.kubelet.pkg.apis.dra.v1beta1"
DRA_V1ALPHA_PKG = "v1alpha3"
REG_PKG = "pluginregistration"

_pool = descriptor_pool.DescriptorPool()


def _add_dra_file(pkg: str, filename: str) -> None:
    f = descriptor_pb2.FileDescriptorProto()
    f.name = filename
    f.package = pkg
    f.syntax = "proto3"

    TYPE_STRING = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
    TYPE_MESSAGE = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
    LABEL_REPEATED = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
    LABEL_OPTIONAL = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL

    def add_field(msg, name, number, ftype, label=LABEL_OPTIONAL, type_name=""):
        fld = msg.field.add()
        fld.name = name
        fld.number = number
        fld.type = ftype
        fld.label = label
        if type_name:
            fld.type_name = f".{pkg}.{type_name}"
        return fld

    # Claim
    claim = f.message_type.add()
    claim.name = "Claim"
    add_field(claim, "namespace", 1, TYPE_STRING)
    add_field(claim, "uid", 2, TYPE_STRING)
    add_field(claim, "name", 3, TYPE_STRING)

    # Device
    device = f.message_type.add()
    device.name = "Device"
    add_field(device, "request_names", 1, TYPE_STRING, LABEL_REPEATED)
    add_field(device, "pool_name", 2, TYPE_STRING)
    add_field(device, "device_name", 3, TYPE_STRING)
    add_field(device, "cdi_device_ids", 4, TYPE_STRING, LABEL_REPEATED)

    # NodePrepareResourcesRequest
    prep_req = f.message_type.add()
    prep_req.name = "NodePrepareResourcesRequest"
    add_field(prep_req, "claims", 1, TYPE_MESSAGE, LABEL_REPEATED, "Claim")

    # NodePrepareResourceResponse
    prep_one = f.message_type.add()
    prep_one.name = "NodePrepareResourceResponse"
    add_field(prep_one, "devices", 1, TYPE_MESSAGE, LABEL_REPEATED, "Device")
    add_field(prep_one, "error", 2, TYPE_STRING)

    # NodePrepareResourcesResponse with map<string, NodePrepareResourceResponse>
    prep_resp = f.message_type.add()
    prep_resp.name = "NodePrepareResourcesResponse"
    entry = prep_resp.nested_type.add()
    entry.name = "ClaimsEntry"
    entry.options.map_entry = True
    add_field(entry, "key", 1, TYPE_STRING)
    v = entry.field.add()
    v.name = "value"
    v.number = 2
    v.type = TYPE_MESSAGE
    v.label = LABEL_OPTIONAL
    v.type_name = f".{pkg}.NodePrepareResourceResponse"
    claims_f = prep_resp.field.add()
    claims_f.name = "claims"
    claims_f.number = 1
    claims_f.type = TYPE_MESSAGE
    claims_f.label = LABEL_REPEATED
    claims_f.type_name = f".{pkg}.NodePrepareResourcesResponse.ClaimsEntry"

    # NodeUnprepareResourceResponse
    unprep_one = f.message_type.add()
    unprep_one.name = "NodeUnprepareResourceResponse"
    add_field(unprep_one, "error", 1, TYPE_STRING)

    # NodeUnprepareResourcesRequest
    unprep_req = f.message_type.add()
    unprep_req.name = "NodeUnprepareResourcesRequest"
    add_field(unprep_req, "claims", 1, TYPE_MESSAGE, LABEL_REPEATED, "Claim")

    # NodeUnprepareResourcesResponse map
    unprep_resp = f.message_type.add()
    unprep_resp.name = "NodeUnprepareResourcesResponse"
    uentry = unprep_resp.nested_type.add()
    uentry.name = "ClaimsEntry"
    uentry.options.map_entry = True
    add_field(uentry, "key", 1, TYPE_STRING)
    uv = uentry.field.add()
    uv.name = "value"
    uv.number = 2
    uv.type = TYPE_MESSAGE
    uv.label = LABEL_OPTIONAL
    uv.type_name = f".{pkg}.NodeUnprepareResourceResponse"
    uclaims_f = unprep_resp.field.add()
    uclaims_f.name = "claims"
    uclaims_f.number = 1
    uclaims_f.type = TYPE_MESSAGE
    uclaims_f.label = LABEL_REPEATED
    uclaims_f.type_name = f".{pkg}.NodeUnprepareResourcesResponse.ClaimsEntry"

    _pool.Add(f)


def _add_registration_file() -> None:
    f = descriptor_pb2.FileDescriptorProto()
    f.name = "pluginregistration.proto"
    f.package = REG_PKG
    f.syntax = "proto3"

    TYPE_STRING = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
    TYPE_BOOL = descriptor_pb2.FieldDescriptorProto.TYPE_BOOL
    LABEL_REPEATED = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
    LABEL_OPTIONAL = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL

    info = f.message_type.add()
    info.name = "PluginInfo"
    for name, num, label in (
        ("type", 1, LABEL_OPTIONAL),
        ("name", 2, LABEL_OPTIONAL),
        ("endpoint", 3, LABEL_OPTIONAL),
        ("supported_versions", 4, LABEL_REPEATED),
    ):
        fld = info.field.add()
        fld.name, fld.number, fld.type, fld.label = name, num, TYPE_STRING, label

    status = f.message_type.add()
    status.name = "RegistrationStatus"
    fld = status.field.add()
    fld.name, fld.number, fld.type, fld.label = (
        "plugin_registered",
        1,
        TYPE_BOOL,
        LABEL_OPTIONAL,
    )
    fld = status.field.add()
    fld.name, fld.number, fld.type, fld.label = ("error", 2, TYPE_STRING, LABEL_OPTIONAL)

    f.message_type.add().name = "RegistrationStatusResponse"
    f.message_type.add().name = "InfoRequest"

    _pool.Add(f)


_add_dra_file(DRA_V1_PKG, "dra_v1.proto")
_add_dra_file(DRA_V1BETA1_PKG, "dra_v1beta1.proto")
_add_dra_file(DRA_V1ALPHA_PKG, "dra_v1alpha4.proto")
_add_registration_file()


def _msg(full_name: str):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(full_name))


class DraMessages:
    """Message classes for one DRA API version."""

    def __init__(self, pkg: str, service: str):
        self.package = pkg
        self.service = service
        self.Claim = _msg(f"{pkg}.Claim")
        self.Device = _msg(f"{pkg}.Device")
        self.NodePrepareResourcesRequest = _msg(f"{pkg}.NodePrepareResourcesRequest")
        self.NodePrepareResourcesResponse = _msg(f"{pkg}.NodePrepareResourcesResponse")
        self.NodePrepareResourceResponse = _msg(f"{pkg}.NodePrepareResourceResponse")
        self.NodeUnprepareResourcesRequest = _msg(f"{pkg}.NodeUnprepareResourcesRequest")
        self.NodeUnprepareResourcesResponse = _msg(
            f"{pkg}.NodeUnprepareResourcesResponse"
        )
        self.NodeUnprepareResourceResponse = _msg(
            f"{pkg}.NodeUnprepareResourceResponse"
        )

    @property
    def service_name(self) -> str:
        return f"{self.package}.{self.service}"


V1 = DraMessages(DRA_V1_PKG, "DRAPlugin")
V1BETA1 = DraMessages(DRA_V1BETA1_PKG, "DRAPlugin")
V1ALPHA4 = DraMessages(DRA_V1ALPHA_PKG, "Node")

#: kubelet DRA API version strings advertised at registration
DRA_VERSION_V1 = "v1"
DRA_VERSION_V1BETA1 = "v1beta1"
DRA_VERSION_V1ALPHA4 = "v1alpha4"


class RegistrationMessages:
    PluginInfo = _msg(f"{REG_PKG}.PluginInfo")
    RegistrationStatus = _msg(f"{REG_PKG}.RegistrationStatus")
    RegistrationStatusResponse = _msg(f"{REG_PKG}.RegistrationStatusResponse")
    InfoRequest = _msg(f"{REG_PKG}.InfoRequest")
    service_name = f"{REG_PKG}.Registration"


REGISTRATION = RegistrationMessages()
