"""Config: every knob of the reference's Config.scala (CLI flag surface is
API — SURVEY.md §5 "Config / flag system"), adapted to the Spark-free
one-process-per-GPU runtime.  Flags keep the reference's names
(Config.scala:403-499)."""

from __future__ import annotations

import argparse
import os
from typing import List, Optional

import torch

from ..proto import caffe_pb, text_format


class Config:
    def __init__(self, args: Optional[List[str]] = None, **kw):
        p = argparse.ArgumentParser(prog="caffeonspark_amd", add_help=True)
        a = p.add_argument
        a("-conf", "--conf", dest="protoFile", default="",
          help="solver prototxt")
        a("-train", dest="isTraining", action="store_true")
        a("-test", dest="isTest", action="store_true")
        a("-features", dest="features", default="",
          help="comma-separated blob names to extract")
        a("-label", dest="label", default="", help="label blob name")
        a("-outputFormat", dest="outputFormat", default="json")
        a("-model", dest="modelPath", default="")
        a("-output", dest="outputPath", default="")
        a("-devices", dest="devices", type=int, default=1)
        a("-persistent", dest="isRddPersistent", action="store_true")
        a("-snapshot", dest="snapshotStateFile", default="")
        a("-weights", dest="modelWeightFile", default="")
        a("-connection", dest="connection", default="rccl",
          help="rccl|gloo|ethernet (reference: RDMA|SOCKET)")
        a("-resize", dest="resize", action="store_true")
        a("-clusterSize", dest="clusterSize", type=int, default=0)
        a("-lmdb_partitions", dest="lmdbPartitions", type=int, default=0)
        a("-imageRoot", dest="imageRoot", default="")
        a("-labelFile", dest="labelFile", default="")
        a("-captionFile", dest="captionFile", default="")
        a("-captionLength", dest="captionLength", type=int, default=20)
        a("-vocabSize", dest="vocabSize", type=int, default=10000)
        a("-imageCaptionDFDir", dest="imageCaptionDFDir", default="")
        a("-vocabDir", dest="vocabDir", default="")
        a("-embeddingDFDir", dest="embeddingDFDir", default="")
        a("-transform_thread_per_device", dest="transform_threads",
          type=int, default=1)
        a("-dtype", dest="dtype_name", default="",
          help="compute dtype: bf16|fp32 (default: bf16 on GPU)")
        ns, _ = p.parse_known_args(args or [])
        self.raw_args = list(args or [])   # re-parse on Spark executors
        self.__dict__.update(vars(ns))
        self.__dict__.update(kw)

        self.solver_param: Optional[caffe_pb.SolverParameter] = None
        self.net_param: Optional[caffe_pb.NetParameter] = None
        self.proto_dir = "."
        if self.protoFile:
            self.load_proto(self.protoFile)

        ws = int(os.environ.get("WORLD_SIZE", "1"))
        if not self.clusterSize:
            self.clusterSize = ws
        self.rank = int(os.environ.get("RANK", "0"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        if torch.cuda.is_available():
            self.device = torch.device("cuda", self.local_rank)
            default_dtype = torch.bfloat16
        else:
            self.device = torch.device("cpu")
            default_dtype = torch.float32
        self.dtype = {"bf16": torch.bfloat16, "fp32": torch.float32,
                      "": default_dtype}[self.dtype_name]

    # -- reference getter names ---------------------------------------------
    @property
    def weights(self) -> str:
        return self.modelWeightFile

    @property
    def snapshot_state(self) -> str:
        return self.snapshotStateFile

    @property
    def model_path(self) -> str:
        return self.modelPath

    def load_proto(self, path: str) -> None:
        self.protoFile = path
        self.proto_dir = os.path.dirname(os.path.abspath(path))
        self.solver_param = text_format.parse_file(
            path, caffe_pb.SolverParameter)
        net_path = self.solver_param.net or self.solver_param.train_net
        if self.solver_param.has_field("net_param"):
            self.net_param = self.solver_param.net_param
        else:
            if not os.path.exists(net_path):
                alt = os.path.join(self.proto_dir,
                                   os.path.basename(net_path))
                if os.path.exists(alt):
                    net_path = alt
            self.net_param = text_format.parse_file(
                net_path, caffe_pb.NetParameter)

    # convenience accessors mirroring Config.scala getters
    @property
    def train_data_layer_id(self) -> int:
        from ..data.source import find_data_layer
        return find_data_layer(self.net_param, True)[0]

    @property
    def test_data_layer_id(self) -> int:
        from ..data.source import find_data_layer
        return find_data_layer(self.net_param, False)[0]
