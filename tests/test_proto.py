"""Proto subsystem tests: wire-format byte-compatibility is cross-checked
against google.protobuf by building the same schema dynamically (no protoc)."""

import numpy as np
import pytest

from poseidon_amd.proto import Message, parse_text, to_text, spec


# ---------------------------------------------------------------------------
# Dynamic google.protobuf twin of our schema
# ---------------------------------------------------------------------------

def _build_google_pool():
    from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "caffe_twin.proto"
    fdp.package = "caffetwin"
    fdp.syntax = "proto2"

    # proto2 scopes enum VALUES to the enclosing namespace, and our enums
    # share symbols (MAX appears in EltwiseOp and PoolMethod) -- nest each
    # enum inside a wrapper message to give it its own scope.
    for ename, table in spec.ENUMS.items():
        wrapper = fdp.message_type.add()
        wrapper.name = f"E_{ename}"
        e = wrapper.enum_type.add()
        e.name = ename
        for sym, val in sorted(table.items(), key=lambda kv: kv[1]):
            v = e.value.add()
            v.name = sym
            v.number = val

    KIND2TYPE = {
        "int32": descriptor_pb2.FieldDescriptorProto.TYPE_INT32,
        "int64": descriptor_pb2.FieldDescriptorProto.TYPE_INT64,
        "uint32": descriptor_pb2.FieldDescriptorProto.TYPE_UINT32,
        "uint64": descriptor_pb2.FieldDescriptorProto.TYPE_UINT64,
        "bool": descriptor_pb2.FieldDescriptorProto.TYPE_BOOL,
        "float": descriptor_pb2.FieldDescriptorProto.TYPE_FLOAT,
        "double": descriptor_pb2.FieldDescriptorProto.TYPE_DOUBLE,
        "string": descriptor_pb2.FieldDescriptorProto.TYPE_STRING,
        "bytes": descriptor_pb2.FieldDescriptorProto.TYPE_BYTES,
    }

    for mname, fields in spec.MESSAGES.items():
        m = fdp.message_type.add()
        m.name = mname
        for fname, (num, kind, label, default) in fields.items():
            f = m.field.add()
            f.name = fname
            f.number = num
            f.label = (descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
                       if label in ("rep", "packed")
                       else descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL)
            if kind.startswith("msg:"):
                f.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
                f.type_name = f".caffetwin.{kind[4:]}"
            elif kind.startswith("enum:"):
                f.type = descriptor_pb2.FieldDescriptorProto.TYPE_ENUM
                f.type_name = f".caffetwin.E_{kind[5:]}.{kind[5:]}"
            else:
                f.type = KIND2TYPE[kind]
            if label == "packed":
                f.options.packed = True

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    classes = {}
    for mname in spec.MESSAGES:
        desc = pool.FindMessageTypeByName(f"caffetwin.{mname}")
        classes[mname] = message_factory.GetMessageClass(desc)
    return classes


@pytest.fixture(scope="module")
def gclasses():
    return _build_google_pool()


def _sample_net():
    net = Message("NetParameter", name="testnet")
    l1 = net.add("layers", name="data", type="DUMMY_DATA")
    l1.top.append("data")
    dp = l1.ensure("dummy_data_param")
    dp.num.append(4)
    dp.channels.append(3)
    dp.height.append(8)
    dp.width.append(8)
    l2 = net.add("layers", name="conv1", type="CONVOLUTION")
    l2.bottom.append("data")
    l2.top.append("conv1")
    l2.blobs_lr.extend([1.0, 2.0])
    cp = l2.ensure("convolution_param")
    cp.num_output = 16
    cp.kernel_size = 3
    cp.stride = 2
    cp.pad = 1
    wf = cp.ensure("weight_filler")
    wf.type = "gaussian"
    wf.std = 0.01
    blob = l2.add("blobs", num=16, channels=3, height=3, width=3)
    blob.data = np.arange(16 * 3 * 3 * 3, dtype=np.float32) * 0.125
    blob.blob_mode = "GLOBAL"
    blob.global_id = 7
    return net


def _fill_google_from_ours(gmsg, ours):
    from poseidon_amd.proto import spec as s
    for name, (num, kind, label, default) in s.MESSAGES[ours.type_name].items():
        if not ours.has(name):
            continue
        v = getattr(ours, name)
        if label in ("rep", "packed"):
            items = list(np.asarray(v).tolist()) if isinstance(v, np.ndarray) else v
            if kind.startswith("msg:"):
                for item in items:
                    _fill_google_from_ours(getattr(gmsg, name).add(), item)
            else:
                getattr(gmsg, name).extend(items)
        elif kind.startswith("msg:"):
            _fill_google_from_ours(getattr(gmsg, name), v)
        else:
            setattr(gmsg, name, v)


def test_wire_bytes_match_google(gclasses):
    net = _sample_net()
    gnet = gclasses["NetParameter"]()
    _fill_google_from_ours(gnet, net)
    ours = net.encode()
    theirs = gnet.SerializeToString(deterministic=True)
    assert ours == theirs


def test_decode_google_bytes(gclasses):
    net = _sample_net()
    gnet = gclasses["NetParameter"]()
    _fill_google_from_ours(gnet, net)
    raw = gnet.SerializeToString(deterministic=True)
    back = Message.decode("NetParameter", raw)
    assert back.name == "testnet"
    assert len(back.layers) == 2
    conv = back.layers[1]
    assert conv.enum_name("type") == "CONVOLUTION"
    assert conv.convolution_param.num_output == 16
    assert conv.convolution_param.weight_filler.std == pytest.approx(0.01)
    blob = conv.blobs[0]
    assert blob.enum_name("blob_mode") == "GLOBAL"
    np.testing.assert_allclose(
        np.asarray(blob.data), np.arange(16 * 27, dtype=np.float32) * 0.125)


def test_roundtrip_binary():
    net = _sample_net()
    back = Message.decode("NetParameter", net.encode())
    assert back.encode() == net.encode()


def test_negative_int32_varint(gclasses):
    b = Message("BlobProto", global_id=-1, num=2)
    g = gclasses["BlobProto"]()
    g.global_id = -1
    g.num = 2
    assert b.encode() == g.SerializeToString(deterministic=True)
    back = Message.decode("BlobProto", b.encode())
    assert back.global_id == -1


PROTOTXT = """
# a comment
name: "LeNet-ish"
layers {
  name: "data"
  type: DATA
  top: "data"
  top: "label"
  data_param { source: "/tmp/db" batch_size: 64 backend: LMDB }
  transform_param { scale: 0.00390625 }
  include: { phase: TRAIN }
}
layers {
  name: "ip1"
  type: INNER_PRODUCT
  bottom: "data"
  top: "ip1"
  blobs_lr: 1
  blobs_lr: 2
  inner_product_param {
    num_output: 500
    weight_filler { type: "xavier" }
    bias_filler { type: "constant" }
  }
}
"""


def test_parse_prototxt():
    net = parse_text("NetParameter", PROTOTXT)
    assert net.name == "LeNet-ish"
    assert len(net.layers) == 2
    data = net.layers[0]
    assert data.enum_name("type") == "DATA"
    assert data.data_param.batch_size == 64
    assert data.data_param.enum_name("backend") == "LMDB"
    assert data.transform_param.scale == pytest.approx(0.00390625)
    assert data.include[0].enum_name("phase") == "TRAIN"
    ip = net.layers[1]
    assert list(ip.blobs_lr) == [1.0, 2.0]
    assert ip.inner_product_param.num_output == 500
    assert ip.inner_product_param.weight_filler.type == "xavier"


def test_text_roundtrip():
    net = parse_text("NetParameter", PROTOTXT)
    text = to_text(net)
    again = parse_text("NetParameter", text)
    assert again.encode() == net.encode()


def test_solver_prototxt():
    solver = parse_text("SolverParameter", """
        net: "train.prototxt"
        base_lr: 0.01
        lr_policy: "inv"
        gamma: 0.0001
        power: 0.75
        momentum: 0.9
        weight_decay: 0.0005
        max_iter: 10000
        solver_mode: GPU
        solver_type: NESTEROV
    """)
    assert solver.base_lr == pytest.approx(0.01)
    assert solver.enum_name("solver_type") == "NESTEROV"
    assert solver.enum_name("solver_mode") == "GPU"
    assert solver.regularization_type == "L2"  # default
