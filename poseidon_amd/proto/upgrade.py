"""Legacy NetParameter upgrade: V0LayerParameter -> LayerParameter.

Parity with the reference's upgrade path
(/root/reference/src/caffe/util/upgrade_proto.cpp:
NetNeedsUpgrade/UpgradeV0Net): pre-2014 prototxts put everything in a
string-typed V0LayerParameter under `layers { layer { ... } }`; this maps
the string types onto the LayerType enum and scatters the flat fields into
the typed per-layer params. Padding layers are fused into the following
convolution, as the reference does.
"""

from __future__ import annotations

from .message import Message

_V0_TYPE_MAP = {
    "accuracy": "ACCURACY", "bnll": "BNLL", "concat": "CONCAT",
    "conv": "CONVOLUTION", "data": "DATA", "dropout": "DROPOUT",
    "euclidean_loss": "EUCLIDEAN_LOSS", "flatten": "FLATTEN",
    "hdf5_data": "HDF5_DATA", "hdf5_output": "HDF5_OUTPUT",
    "im2col": "IM2COL", "images": "IMAGE_DATA",
    "infogain_loss": "INFOGAIN_LOSS", "innerproduct": "INNER_PRODUCT",
    "lrn": "LRN", "multinomial_logistic_loss": "MULTINOMIAL_LOGISTIC_LOSS",
    "pool": "POOLING", "relu": "RELU", "sigmoid": "SIGMOID",
    "softmax": "SOFTMAX", "softmax_loss": "SOFTMAX_LOSS", "split": "SPLIT",
    "tanh": "TANH", "window_data": "WINDOW_DATA",
}


def net_needs_upgrade(net: Message) -> bool:
    return any(lp.has("layer") for lp in net.layers)


def upgrade_v0_net(net: Message) -> Message:
    """Returns an upgraded copy; non-V0 layers pass through unchanged."""
    if not net_needs_upgrade(net):
        return net
    out = Message("NetParameter")
    if net.has("name"):
        out.name = net.name
    out.input.extend(net.input)
    out.input_dim.extend(net.input_dim)
    if net.has("force_backward"):
        out.force_backward = net.force_backward

    pending_pad = 0
    for lp in net.layers:
        if not lp.has("layer"):
            out.layers.append(lp)
            continue
        v0 = lp.layer
        t = (v0.type or "").lower()
        if t == "padding":
            pending_pad = int(v0.pad)
            continue  # fused into the next conv (upgrade_proto.cpp semantics)
        new = Message("LayerParameter", name=v0.name or lp.name or "")
        new.bottom.extend(lp.bottom)
        new.top.extend(lp.top)
        if t not in _V0_TYPE_MAP:
            raise ValueError(f"unknown V0 layer type {v0.type!r}")
        new.type = _V0_TYPE_MAP[t]
        for b in v0.blobs:
            new.blobs.append(b)
        new.blobs_lr.extend(v0.blobs_lr)
        new.weight_decay.extend(v0.weight_decay)

        tt = new.enum_name("type")
        if tt == "CONVOLUTION":
            cp = new.ensure("convolution_param")
            cp.num_output = int(v0.num_output)
            cp.bias_term = bool(v0.biasterm)
            if v0.has("kernelsize"):
                cp.kernel_size = int(v0.kernelsize)
            cp.group = int(v0.group)
            cp.stride = int(v0.stride)
            cp.pad = pending_pad if pending_pad else int(v0.pad)
            if v0.has("weight_filler"):
                cp.weight_filler = v0.weight_filler
            if v0.has("bias_filler"):
                cp.bias_filler = v0.bias_filler
            pending_pad = 0
        elif tt == "INNER_PRODUCT":
            ip = new.ensure("inner_product_param")
            ip.num_output = int(v0.num_output)
            ip.bias_term = bool(v0.biasterm)
            if v0.has("weight_filler"):
                ip.weight_filler = v0.weight_filler
            if v0.has("bias_filler"):
                ip.bias_filler = v0.bias_filler
        elif tt == "POOLING":
            pp = new.ensure("pooling_param")
            pp.pool = int(v0.pool)
            if v0.has("kernelsize"):
                pp.kernel_size = int(v0.kernelsize)
            pp.stride = int(v0.stride)
        elif tt == "DROPOUT":
            new.ensure("dropout_param").dropout_ratio = float(v0.dropout_ratio)
        elif tt == "LRN":
            p = new.ensure("lrn_param")
            p.local_size = int(v0.local_size)
            p.alpha = float(v0.alpha)
            p.beta = float(v0.beta)
        elif tt in ("DATA", "IMAGE_DATA", "WINDOW_DATA", "HDF5_DATA"):
            dp = new.ensure("data_param" if tt == "DATA" else
                            "image_data_param" if tt == "IMAGE_DATA" else
                            "window_data_param" if tt == "WINDOW_DATA" else
                            "hdf5_data_param")
            if v0.has("source"):
                dp.source = v0.source
            if v0.has("batchsize"):
                dp.batch_size = int(v0.batchsize)
            if tt in ("DATA", "IMAGE_DATA"):
                tp = new.ensure("transform_param")
                if v0.has("scale"):
                    tp.scale = float(v0.scale)
                if v0.has("meanfile"):
                    tp.mean_file = v0.meanfile
                if v0.has("cropsize"):
                    tp.crop_size = int(v0.cropsize)
                if v0.has("mirror"):
                    tp.mirror = bool(v0.mirror)
        elif tt == "CONCAT":
            new.ensure("concat_param").concat_dim = int(v0.concat_dim)
        out.layers.append(new)
    return out
