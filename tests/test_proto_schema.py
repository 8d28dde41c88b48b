"""Cross-validate the hand-rolled v1beta1 wire encodings against the real
protobuf runtime: messages are defined here as dynamic descriptors with the
upstream field numbers (k8s.io/kubelet deviceplugin/v1beta1/api.proto) and
must parse our bytes exactly."""

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

import pb_v1beta1 as pb


def build_pool():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "v1beta1_test.proto"
    fdp.package = "v1beta1"

    def msg(name, fields):
        m = fdp.message_type.add()
        m.name = name
        for fname, num, ftype, label, type_name in fields:
            f = m.field.add()
            f.name = fname
            f.number = num
            f.type = ftype
            f.label = label
            if type_name:
                f.type_name = type_name
        return m

    T = descriptor_pb2.FieldDescriptorProto
    msg("DevicePluginOptions", [
        ("pre_start_required", 1, T.TYPE_BOOL, T.LABEL_OPTIONAL, None),
        ("get_preferred_allocation_available", 2, T.TYPE_BOOL,
         T.LABEL_OPTIONAL, None),
    ])
    msg("RegisterRequest", [
        ("version", 1, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("endpoint", 2, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("resource_name", 3, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("options", 4, T.TYPE_MESSAGE, T.LABEL_OPTIONAL,
         ".v1beta1.DevicePluginOptions"),
    ])
    msg("NUMANode", [("ID", 1, T.TYPE_INT64, T.LABEL_OPTIONAL, None)])
    msg("TopologyInfo", [("nodes", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED,
                          ".v1beta1.NUMANode")])
    msg("Device", [
        ("ID", 1, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("health", 2, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("topology", 3, T.TYPE_MESSAGE, T.LABEL_OPTIONAL,
         ".v1beta1.TopologyInfo"),
    ])
    msg("ListAndWatchResponse", [
        ("devices", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".v1beta1.Device"),
    ])
    msg("Mount", [
        ("container_path", 1, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("host_path", 2, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("read_only", 3, T.TYPE_BOOL, T.LABEL_OPTIONAL, None),
    ])
    msg("DeviceSpec", [
        ("container_path", 1, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("host_path", 2, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("permissions", 3, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
    ])
    # map<string,string> = repeated MapEntry{key=1,value=2} submessage
    entry = msg("EnvsEntry", [
        ("key", 1, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
        ("value", 2, T.TYPE_STRING, T.LABEL_OPTIONAL, None),
    ])
    entry.options.map_entry = True
    msg("ContainerAllocateResponse", [
        ("envs", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".v1beta1.EnvsEntry"),
        ("mounts", 2, T.TYPE_MESSAGE, T.LABEL_REPEATED, ".v1beta1.Mount"),
        ("devices", 3, T.TYPE_MESSAGE, T.LABEL_REPEATED,
         ".v1beta1.DeviceSpec"),
    ])
    msg("AllocateResponse", [
        ("container_responses", 1, T.TYPE_MESSAGE, T.LABEL_REPEATED,
         ".v1beta1.ContainerAllocateResponse"),
    ])

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    return pool


POOL = build_pool()


def make(name):
    return message_factory.GetMessageClass(
        POOL.FindMessageTypeByName(f"v1beta1.{name}"))()


def test_register_request_parses_with_protobuf(tmp_path):
    # encode with the C++-mirroring helpers, parse with real protobuf
    out = bytearray()
    pb.put_str(out, 1, "v1beta1")
    pb.put_str(out, 2, "amd-gpu.sock")
    pb.put_str(out, 3, "amd.com/gpu")
    opts = bytearray()
    pb.put_tag(opts, 2, 0)
    pb.put_varint(opts, 1)
    pb.put_bytes(out, 4, bytes(opts))

    m = make("RegisterRequest")
    m.ParseFromString(bytes(out))
    assert m.version == "v1beta1"
    assert m.endpoint == "amd-gpu.sock"
    assert m.resource_name == "amd.com/gpu"
    assert m.options.get_preferred_allocation_available is True


def test_list_and_watch_roundtrip_via_protobuf():
    # real protobuf encodes; our decoder must read it
    law = make("ListAndWatchResponse")
    d = law.devices.add()
    d.ID = "amdgpu-0001::2"
    d.health = "Healthy"
    d.topology.nodes.add().ID = 1
    data = law.SerializeToString()
    devs = pb.decode_list_and_watch(data)
    assert devs == [{"id": "amdgpu-0001::2", "health": "Healthy", "numa": 1}]


def test_allocate_response_parses_with_protobuf():
    """The C++ plugin's AllocateResponse bytes must be readable by real
    protobuf with the upstream schema (this is what kubelet does)."""
    # reproduce the plugin's encoding through the python mirror helpers
    cr = bytearray()
    entry = bytearray()
    pb.put_str(entry, 1, "K3SAMD_VISIBLE_DEVICES")
    pb.put_str(entry, 2, "amdgpu-0001")
    pb.put_bytes(cr, 1, bytes(entry))
    dev = bytearray()
    pb.put_str(dev, 1, "/dev/kfd")
    pb.put_str(dev, 2, "/dev/kfd")
    pb.put_str(dev, 3, "rw")
    pb.put_bytes(cr, 3, bytes(dev))
    out = bytearray()
    pb.put_bytes(out, 1, bytes(cr))

    m = make("AllocateResponse")
    m.ParseFromString(bytes(out))
    assert len(m.container_responses) == 1
    r = m.container_responses[0]
    assert r.envs["K3SAMD_VISIBLE_DEVICES"] == "amdgpu-0001"
    assert r.devices[0].host_path == "/dev/kfd"
    assert r.devices[0].permissions == "rw"
