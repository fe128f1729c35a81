"""Protobuf schema + wire/text codecs for the caffe.proto message set.

Parity: /root/reference/src/caffe/util/io.cpp:38-77
(ReadProtoFromTextFile / WriteProtoToTextFile / ReadProtoFromBinaryFile /
WriteProtoToBinaryFile, incl. the reference's large coded-stream limits --
irrelevant here, the codec is stream-size-agnostic) over the schema in
spec.py (field numbers byte-compatible with src/caffe/proto/caffe.proto)."""

from . import spec
from .message import Message
from .text_format import parse_text, to_text


def read_proto_text(path: str, type_name: str) -> Message:
    with open(path, "r") as f:
        return parse_text(type_name, f.read())


def write_proto_text(msg: Message, path: str) -> None:
    with open(path, "w") as f:
        f.write(to_text(msg) + "\n")


def read_proto_binary(path: str, type_name: str) -> Message:
    with open(path, "rb") as f:
        return Message.decode(type_name, f.read())


def write_proto_binary(msg: Message, path: str) -> None:
    with open(path, "wb") as f:
        f.write(msg.encode())


__all__ = [
    "spec", "Message", "parse_text", "to_text",
    "read_proto_text", "write_proto_text",
    "read_proto_binary", "write_proto_binary",
]
