"""Rewrite a NetParameter to insert SPLIT layers where one blob *instance*
feeds multiple consumers (semantics of
/root/reference/src/caffe/util/insert_splits.cpp).

Blob names are versioned by producer: an in-place layer (top == bottom)
produces a NEW instance of the name, so `ip1 -> relu(in-place) -> ip2`
needs no split even though the name "ip1" appears as a bottom twice.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

from ..proto import Message

Instance = Tuple[str, int]  # (producer layer name, top index)


def _split_layer_name(blob: str, producer: str, idx: int) -> str:
    return f"{blob}_{producer}_{idx}_split"


def _split_blob_name(blob: str, producer: str, idx: int, k: int) -> str:
    return f"{_split_layer_name(blob, producer, idx)}_{k}"


def insert_splits(net: Message) -> Message:
    # pass 1: resolve each bottom to its producing instance, count consumers
    latest: Dict[str, Instance] = {}
    consume: Dict[Instance, int] = {}
    bottom_inst: Dict[Tuple[int, int], Instance] = {}

    for i, name in enumerate(net.input):
        inst = ("input", i)
        latest[name] = inst
        consume[inst] = 0
    for li, layer in enumerate(net.layers):
        for bi, b in enumerate(layer.bottom):
            if b not in latest:
                raise ValueError(f"layer {layer.name}: unknown bottom {b!r}")
            inst = latest[b]
            bottom_inst[(li, bi)] = inst
            consume[inst] = consume.get(inst, 0) + 1
        for ti, t in enumerate(layer.top):
            inst = (layer.name or f"#layer{li}", ti)
            latest[t] = inst
            consume.setdefault(inst, 0)

    if not any(c > 1 for c in consume.values()):
        return net

    out = Message.decode("NetParameter", net.encode())  # deep copy
    new_layers: List[Message] = []
    handed: Dict[Instance, int] = {}

    def emit_split(blob_name: str, inst: Instance) -> None:
        prod, ti = inst
        sl = Message("LayerParameter",
                     name=_split_layer_name(blob_name, prod, ti), type="SPLIT")
        sl.bottom.append(blob_name)
        for k in range(consume[inst]):
            sl.top.append(_split_blob_name(blob_name, prod, ti, k))
        new_layers.append(sl)

    for i, name in enumerate(out.input):
        inst = ("input", i)
        if consume.get(inst, 0) > 1:
            emit_split(name, inst)

    for li, layer in enumerate(out.layers):
        in_place = {ti: layer.bottom[ti] == layer.top[ti]
                    for ti in range(min(len(layer.bottom), len(layer.top)))}
        for bi in range(len(layer.bottom)):
            inst = bottom_inst[(li, bi)]
            if consume[inst] > 1:
                if in_place.get(bi, False):
                    raise ValueError(
                        f"layer {layer.name}: blob {layer.bottom[bi]!r} is "
                        "consumed in-place AND by another layer -- ambiguous "
                        "(rename the in-place top)")
                k = handed.get(inst, 0)
                handed[inst] = k + 1
                layer.bottom[bi] = _split_blob_name(layer.bottom[bi],
                                                    inst[0], inst[1], k)
        new_layers.append(layer)
        for ti, t in enumerate(layer.top):
            inst = (layer.name or f"#layer{li}", ti)
            if consume.get(inst, 0) > 1:
                emit_split(t, inst)

    del out.layers[:]
    out.layers.extend(new_layers)
    return out
