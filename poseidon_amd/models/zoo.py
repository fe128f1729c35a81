"""Model zoo: programmatic NetParameter builders for the architectures the
reference ships prototxts for (models/bvlc_alexnet, models/bvlc_googlenet,
examples/mnist, examples/cifar10) plus VGG-16. Generating the graphs in code
keeps one source of truth; dump_prototxt() writes the equivalent .prototxt.

Synthetic-data variants feed DummyData tops shaped like ImageNet/CIFAR/MNIST
batches (gaussian images, uniform labels) -- the benchmarked path, since the
image has no network access for datasets (BASELINE.md north star).
"""

from __future__ import annotations

from typing import Optional, Sequence, Tuple

from ..proto import Message, to_text


class NetBuilder:
    def __init__(self, name: str):
        self.net = Message("NetParameter", name=name)

    def _layer(self, type_: str, name: str, bottoms, tops, **extra) -> Message:
        lp = self.net.add("layers", name=name, type=type_)
        for b in ([bottoms] if isinstance(bottoms, str) else bottoms or []):
            lp.bottom.append(b)
        for t in ([tops] if isinstance(tops, str) else tops or []):
            lp.top.append(t)
        return lp

    # -- data --------------------------------------------------------------
    def dummy_data(self, name: str, tops: Sequence[str],
                   shapes: Sequence[Tuple[int, ...]],
                   fillers: Optional[Sequence[Tuple[str, dict]]] = None):
        lp = self._layer("DUMMY_DATA", name, [], list(tops))
        dp = lp.ensure("dummy_data_param")
        for shape in shapes:
            n, c, h, w = (tuple(shape) + (1, 1, 1, 1))[:4]
            dp.num.append(n)
            dp.channels.append(c)
            dp.height.append(h)
            dp.width.append(w)
        if fillers:
            for ftype, kw in fillers:
                f = dp.add("data_filler", type=ftype)
                for k, v in kw.items():
                    setattr(f, k, v)
        return lp

    def synthetic_images(self, batch: int, channels: int, size: int,
                         num_classes: int, train: bool = True):
        """Gaussian images + uniform integer-ish labels, filled once."""
        self.dummy_data(
            "data", ["data", "label"],
            [(batch, channels, size, size), (batch, 1, 1, 1)],
            fillers=[("gaussian", {"std": 1.0}),
                     ("uniform", {"min": 0.0, "max": num_classes - 0.01})])

    # -- compute layers ------------------------------------------------------
    def conv(self, name, bottom, top, num_output, kernel, stride=1, pad=0,
             group=1, w_std=0.01, w_type="gaussian", bias=0.0,
             lr=(1.0, 2.0), decay=(1.0, 0.0)):
        lp = self._layer("CONVOLUTION", name, bottom, top)
        lp.blobs_lr.extend(lr)
        lp.weight_decay.extend(decay)
        cp = lp.ensure("convolution_param")
        cp.num_output = num_output
        cp.kernel_size = kernel
        if stride != 1:
            cp.stride = stride
        if pad:
            cp.pad = pad
        if group != 1:
            cp.group = group
        wf = cp.ensure("weight_filler")
        wf.type = w_type
        if w_type == "gaussian":
            wf.std = w_std
        bf = cp.ensure("bias_filler")
        bf.type = "constant"
        bf.value = bias
        return lp

    def ip(self, name, bottom, top, num_output, w_std=0.01, w_type="gaussian",
           bias=0.0, lr=(1.0, 2.0), decay=(1.0, 0.0)):
        lp = self._layer("INNER_PRODUCT", name, bottom, top)
        lp.blobs_lr.extend(lr)
        lp.weight_decay.extend(decay)
        ip = lp.ensure("inner_product_param")
        ip.num_output = num_output
        wf = ip.ensure("weight_filler")
        wf.type = w_type
        if w_type == "gaussian":
            wf.std = w_std
        bf = ip.ensure("bias_filler")
        bf.type = "constant"
        bf.value = bias
        return lp

    def relu(self, name, blob):
        return self._layer("RELU", name, blob, blob)  # in-place

    def pool(self, name, bottom, top, method, kernel, stride, pad=0):
        lp = self._layer("POOLING", name, bottom, top)
        pp = lp.ensure("pooling_param")
        pp.pool = method
        pp.kernel_size = kernel
        pp.stride = stride
        if pad:
            pp.pad = pad
        return lp

    def lrn(self, name, bottom, top, local_size=5, alpha=1e-4, beta=0.75):
        lp = self._layer("LRN", name, bottom, top)
        p = lp.ensure("lrn_param")
        p.local_size = local_size
        p.alpha = alpha
        p.beta = beta
        return lp

    def dropout(self, name, blob, ratio=0.5):
        lp = self._layer("DROPOUT", name, blob, blob)
        lp.ensure("dropout_param").dropout_ratio = ratio
        return lp

    def concat(self, name, bottoms, top):
        return self._layer("CONCAT", name, list(bottoms), top)

    def softmax_loss(self, name, logits, label, top="loss", weight=None):
        lp = self._layer("SOFTMAX_LOSS", name, [logits, label], top)
        if weight is not None:
            lp.loss_weight.append(weight)
        return lp

    def accuracy(self, name, logits, label, top="accuracy", top_k=None):
        lp = self._layer("ACCURACY", name, [logits, label], top)
        if top_k:
            lp.ensure("accuracy_param").top_k = top_k
        return lp

    def build(self) -> Message:
        return self.net


# ---------------------------------------------------------------------------


def lenet(batch: int = 64, num_classes: int = 10) -> Message:
    """LeNet (examples/mnist/lenet_train.prototxt architecture)."""
    b = NetBuilder("LeNet")
    b.synthetic_images(batch, 1, 28, num_classes)
    b.conv("conv1", "data", "conv1", 20, 5, w_type="xavier")
    b.pool("pool1", "conv1", "pool1", "MAX", 2, 2)
    b.conv("conv2", "pool1", "conv2", 50, 5, w_type="xavier")
    b.pool("pool2", "conv2", "pool2", "MAX", 2, 2)
    b.ip("ip1", "pool2", "ip1", 500, w_type="xavier")
    b.relu("relu1", "ip1")
    b.ip("ip2", "ip1", "ip2", num_classes, w_type="xavier")
    b.softmax_loss("loss", "ip2", "label")
    return b.build()


def cifar10_quick(batch: int = 100, num_classes: int = 10) -> Message:
    """CIFAR-10 quick (examples/cifar10/cifar10_quick_train_test.prototxt)."""
    b = NetBuilder("CIFAR10_quick")
    b.synthetic_images(batch, 3, 32, num_classes)
    b.conv("conv1", "data", "conv1", 32, 5, pad=2, w_std=0.0001)
    b.pool("pool1", "conv1", "pool1", "MAX", 3, 2)
    b.relu("relu1", "pool1")
    b.conv("conv2", "pool1", "conv2", 32, 5, pad=2, w_std=0.01)
    b.relu("relu2", "conv2")
    b.pool("pool2", "conv2", "pool2", "AVE", 3, 2)
    b.conv("conv3", "pool2", "conv3", 64, 5, pad=2, w_std=0.01)
    b.relu("relu3", "conv3")
    b.pool("pool3", "conv3", "pool3", "AVE", 3, 2)
    b.ip("ip1", "pool3", "ip1", 64, w_std=0.1)
    b.ip("ip2", "ip1", "ip2", num_classes, w_std=0.1)
    b.softmax_loss("loss", "ip2", "label")
    return b.build()


def alexnet(batch: int = 256, num_classes: int = 1000) -> Message:
    """AlexNet (models/bvlc_alexnet/train_val.prototxt architecture:
    227x227 input, grouped conv2/4/5, LRN after conv1/conv2)."""
    b = NetBuilder("AlexNet")
    b.synthetic_images(batch, 3, 227, num_classes)
    b.conv("conv1", "data", "conv1", 96, 11, stride=4, w_std=0.01, bias=0.0)
    b.relu("relu1", "conv1")
    b.lrn("norm1", "conv1", "norm1")
    b.pool("pool1", "norm1", "pool1", "MAX", 3, 2)
    b.conv("conv2", "pool1", "conv2", 256, 5, pad=2, group=2, w_std=0.01, bias=0.1)
    b.relu("relu2", "conv2")
    b.lrn("norm2", "conv2", "norm2")
    b.pool("pool2", "norm2", "pool2", "MAX", 3, 2)
    b.conv("conv3", "pool2", "conv3", 384, 3, pad=1, w_std=0.01)
    b.relu("relu3", "conv3")
    b.conv("conv4", "conv3", "conv4", 384, 3, pad=1, group=2, w_std=0.01, bias=0.1)
    b.relu("relu4", "conv4")
    b.conv("conv5", "conv4", "conv5", 256, 3, pad=1, group=2, w_std=0.01, bias=0.1)
    b.relu("relu5", "conv5")
    b.pool("pool5", "conv5", "pool5", "MAX", 3, 2)
    b.ip("fc6", "pool5", "fc6", 4096, w_std=0.005, bias=0.1)
    b.relu("relu6", "fc6")
    b.dropout("drop6", "fc6", 0.5)
    b.ip("fc7", "fc6", "fc7", 4096, w_std=0.005, bias=0.1)
    b.relu("relu7", "fc7")
    b.dropout("drop7", "fc7", 0.5)
    b.ip("fc8", "fc7", "fc8", num_classes, w_std=0.01)
    b.softmax_loss("loss", "fc8", "label")
    return b.build()


def caffenet(batch: int = 256, num_classes: int = 1000) -> Message:
    """CaffeNet (models/bvlc_reference_caffenet/train_val.prototxt): the
    AlexNet variant the reference also ships -- pooling BEFORE the LRN
    (pool1 -> norm1) instead of AlexNet's norm -> pool order."""
    b = NetBuilder("CaffeNet")
    b.synthetic_images(batch, 3, 227, num_classes)
    b.conv("conv1", "data", "conv1", 96, 11, stride=4, w_std=0.01, bias=0.0)
    b.relu("relu1", "conv1")
    b.pool("pool1", "conv1", "pool1", "MAX", 3, 2)
    b.lrn("norm1", "pool1", "norm1")
    b.conv("conv2", "norm1", "conv2", 256, 5, pad=2, group=2, w_std=0.01,
           bias=0.1)
    b.relu("relu2", "conv2")
    b.pool("pool2", "conv2", "pool2", "MAX", 3, 2)
    b.lrn("norm2", "pool2", "norm2")
    b.conv("conv3", "norm2", "conv3", 384, 3, pad=1, w_std=0.01)
    b.relu("relu3", "conv3")
    b.conv("conv4", "conv3", "conv4", 384, 3, pad=1, group=2, w_std=0.01,
           bias=0.1)
    b.relu("relu4", "conv4")
    b.conv("conv5", "conv4", "conv5", 256, 3, pad=1, group=2, w_std=0.01,
           bias=0.1)
    b.relu("relu5", "conv5")
    b.pool("pool5", "conv5", "pool5", "MAX", 3, 2)
    b.ip("fc6", "pool5", "fc6", 4096, w_std=0.005, bias=0.1)
    b.relu("relu6", "fc6")
    b.dropout("drop6", "fc6", 0.5)
    b.ip("fc7", "fc6", "fc7", 4096, w_std=0.005, bias=0.1)
    b.relu("relu7", "fc7")
    b.dropout("drop7", "fc7", 0.5)
    b.ip("fc8", "fc7", "fc8", num_classes, w_std=0.01)
    b.softmax_loss("loss", "fc8", "label")
    return b.build()


def _inception(b: NetBuilder, name: str, bottom: str, c1, c3r, c3, c5r, c5, cp):
    b.conv(f"{name}/1x1", bottom, f"{name}/1x1", c1, 1, w_type="xavier")
    b.relu(f"{name}/relu_1x1", f"{name}/1x1")
    b.conv(f"{name}/3x3_reduce", bottom, f"{name}/3x3_reduce", c3r, 1, w_type="xavier")
    b.relu(f"{name}/relu_3x3_reduce", f"{name}/3x3_reduce")
    b.conv(f"{name}/3x3", f"{name}/3x3_reduce", f"{name}/3x3", c3, 3, pad=1, w_type="xavier")
    b.relu(f"{name}/relu_3x3", f"{name}/3x3")
    b.conv(f"{name}/5x5_reduce", bottom, f"{name}/5x5_reduce", c5r, 1, w_type="xavier")
    b.relu(f"{name}/relu_5x5_reduce", f"{name}/5x5_reduce")
    b.conv(f"{name}/5x5", f"{name}/5x5_reduce", f"{name}/5x5", c5, 5, pad=2, w_type="xavier")
    b.relu(f"{name}/relu_5x5", f"{name}/5x5")
    b.pool(f"{name}/pool", bottom, f"{name}/pool", "MAX", 3, 1, pad=1)
    b.conv(f"{name}/pool_proj", f"{name}/pool", f"{name}/pool_proj", cp, 1, w_type="xavier")
    b.relu(f"{name}/relu_pool_proj", f"{name}/pool_proj")
    out = f"{name}/output"
    b.concat(f"{name}/concat",
             [f"{name}/1x1", f"{name}/3x3", f"{name}/5x5", f"{name}/pool_proj"], out)
    return out


def _gnet_aux(b: NetBuilder, name: str, bottom: str, num_classes: int):
    b.pool(f"{name}/ave_pool", bottom, f"{name}/ave_pool", "AVE", 5, 3)
    b.conv(f"{name}/conv", f"{name}/ave_pool", f"{name}/conv", 128, 1, w_type="xavier")
    b.relu(f"{name}/relu_conv", f"{name}/conv")
    b.ip(f"{name}/fc", f"{name}/conv", f"{name}/fc", 1024, w_type="xavier")
    b.relu(f"{name}/relu_fc", f"{name}/fc")
    b.dropout(f"{name}/drop_fc", f"{name}/fc", 0.7)
    b.ip(f"{name}/classifier", f"{name}/fc", f"{name}/classifier", num_classes,
         w_type="xavier")
    b.softmax_loss(f"{name}/loss", f"{name}/classifier", "label",
                   top=f"{name}/loss", weight=0.3)


def googlenet(batch: int = 32, num_classes: int = 1000,
              aux_heads: bool = True) -> Message:
    """GoogLeNet / Inception-v1 (models/bvlc_googlenet architecture, 224x224,
    with the two 0.3-weight auxiliary heads of the training prototxt)."""
    b = NetBuilder("GoogLeNet")
    b.synthetic_images(batch, 3, 224, num_classes)
    b.conv("conv1/7x7_s2", "data", "conv1/7x7_s2", 64, 7, stride=2, pad=3,
           w_type="xavier")
    b.relu("conv1/relu_7x7", "conv1/7x7_s2")
    b.pool("pool1/3x3_s2", "conv1/7x7_s2", "pool1/3x3_s2", "MAX", 3, 2)
    b.lrn("pool1/norm1", "pool1/3x3_s2", "pool1/norm1")
    b.conv("conv2/3x3_reduce", "pool1/norm1", "conv2/3x3_reduce", 64, 1,
           w_type="xavier")
    b.relu("conv2/relu_3x3_reduce", "conv2/3x3_reduce")
    b.conv("conv2/3x3", "conv2/3x3_reduce", "conv2/3x3", 192, 3, pad=1,
           w_type="xavier")
    b.relu("conv2/relu_3x3", "conv2/3x3")
    b.lrn("conv2/norm2", "conv2/3x3", "conv2/norm2")
    b.pool("pool2/3x3_s2", "conv2/norm2", "pool2/3x3_s2", "MAX", 3, 2)

    o = _inception(b, "inception_3a", "pool2/3x3_s2", 64, 96, 128, 16, 32, 32)
    o = _inception(b, "inception_3b", o, 128, 128, 192, 32, 96, 64)
    b.pool("pool3/3x3_s2", o, "pool3/3x3_s2", "MAX", 3, 2)
    o = _inception(b, "inception_4a", "pool3/3x3_s2", 192, 96, 208, 16, 48, 64)
    if aux_heads:
        _gnet_aux(b, "loss1", o, num_classes)
    o = _inception(b, "inception_4b", o, 160, 112, 224, 24, 64, 64)
    o = _inception(b, "inception_4c", o, 128, 128, 256, 24, 64, 64)
    o = _inception(b, "inception_4d", o, 112, 144, 288, 32, 64, 64)
    if aux_heads:
        _gnet_aux(b, "loss2", o, num_classes)
    o = _inception(b, "inception_4e", o, 256, 160, 320, 32, 128, 128)
    b.pool("pool4/3x3_s2", o, "pool4/3x3_s2", "MAX", 3, 2)
    o = _inception(b, "inception_5a", "pool4/3x3_s2", 256, 160, 320, 32, 128, 128)
    o = _inception(b, "inception_5b", o, 384, 192, 384, 48, 128, 128)
    b.pool("pool5/7x7_s1", o, "pool5/7x7_s1", "AVE", 7, 1)
    b.dropout("pool5/drop_7x7_s1", "pool5/7x7_s1", 0.4)
    b.ip("loss3/classifier", "pool5/7x7_s1", "loss3/classifier", num_classes,
         w_type="xavier")
    b.softmax_loss("loss3/loss3", "loss3/classifier", "label", top="loss3/loss3",
                   weight=1.0)
    return b.build()


def vgg16(batch: int = 32, num_classes: int = 1000) -> Message:
    """VGG-16 (stresses SFB: fc6 is a 102M-param FC)."""
    b = NetBuilder("VGG16")
    b.synthetic_images(batch, 3, 224, num_classes)
    cfg = [(2, 64), (2, 128), (3, 256), (3, 512), (3, 512)]
    prev = "data"
    for bi, (reps, ch) in enumerate(cfg, start=1):
        for ri in range(1, reps + 1):
            name = f"conv{bi}_{ri}"
            b.conv(name, prev, name, ch, 3, pad=1, w_std=0.01)
            b.relu(f"relu{bi}_{ri}", name)
            prev = name
        b.pool(f"pool{bi}", prev, f"pool{bi}", "MAX", 2, 2)
        prev = f"pool{bi}"
    b.ip("fc6", prev, "fc6", 4096, w_std=0.005, bias=0.1)
    b.relu("relu6", "fc6")
    b.dropout("drop6", "fc6", 0.5)
    b.ip("fc7", "fc6", "fc7", 4096, w_std=0.005, bias=0.1)
    b.relu("relu7", "fc7")
    b.dropout("drop7", "fc7", 0.5)
    b.ip("fc8", "fc7", "fc8", num_classes, w_std=0.01)
    b.softmax_loss("loss", "fc8", "label")
    return b.build()


MODEL_ZOO = {
    "lenet": lenet,
    "cifar10_quick": cifar10_quick,
    "alexnet": alexnet,
    "caffenet": caffenet,
    "googlenet": googlenet,
    "vgg16": vgg16,
}


def build_net(name: str, **kw) -> Message:
    return MODEL_ZOO[name](**kw)


def dump_prototxt(name: str, path: str, **kw) -> None:
    with open(path, "w") as f:
        f.write(to_text(build_net(name, **kw)) + "\n")
