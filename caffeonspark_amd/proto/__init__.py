"""Protobuf schema + codec for Caffe-compatible configs and checkpoints."""

from . import caffe_pb, text_format
from .pbcodec import Message


def read_solver_prototxt(path: str) -> "caffe_pb.SolverParameter":
    return text_format.parse_file(path, caffe_pb.SolverParameter)


def read_net_prototxt(path: str) -> "caffe_pb.NetParameter":
    return text_format.parse_file(path, caffe_pb.NetParameter)


def read_binary_proto(path: str, msg_cls):
    with open(path, "rb") as fh:
        return msg_cls.FromString(fh.read())


def write_binary_proto(path: str, msg: Message) -> None:
    with open(path, "wb") as fh:
        fh.write(msg.SerializeToString())
