"""Caffe protobuf schema (proto2 semantics) with BVLC-compatible field numbers.

Field numbers follow the public BVLC `caffe.proto` so that `.caffemodel` /
`.solverstate` files and prototxt configs interoperate with upstream Caffe
(the reference consumes the same schema through its absent `caffe-public`
submodule — SURVEY.md §2.5).  CaffeOnSpark extensions (`source_class` on
LayerParameter, `CoSDataParameter`, the MemoryDataParameter extras) live in
a high field-number range; their exact upstream numbers are not recoverable
from the reference mount, so we claim 150+ / 100+ ranges and keep them stable
as *our* format.
"""

from .pbcodec import EnumType, Field, Message

# --------------------------------------------------------------------------- enums

Phase = EnumType("Phase", {"TRAIN": 0, "TEST": 1})


class BlobShape(Message):
    FIELDS = [Field(1, "dim", "int64", repeated=True, packed=True)]


class BlobProto(Message):
    FIELDS = [
        Field(7, "shape", "message", msg_type=BlobShape),
        Field(5, "data", "float", repeated=True, packed=True),
        Field(6, "diff", "float", repeated=True, packed=True),
        Field(8, "double_data", "double", repeated=True, packed=True),
        Field(9, "double_diff", "double", repeated=True, packed=True),
        # 4D legacy dimensions
        Field(1, "num", "int32"),
        Field(2, "channels", "int32"),
        Field(3, "height", "int32"),
        Field(4, "width", "int32"),
        # our extension: raw little-endian tensor bytes (fast path; dtype tag)
        Field(160, "raw_data", "bytes"),
        Field(161, "raw_dtype", "string"),
    ]


class BlobProtoVector(Message):
    FIELDS = [Field(1, "blobs", "message", msg_type=BlobProto, repeated=True)]


class Datum(Message):
    FIELDS = [
        Field(1, "channels", "int32"),
        Field(2, "height", "int32"),
        Field(3, "width", "int32"),
        Field(4, "data", "bytes"),
        Field(5, "label", "int32"),
        Field(6, "float_data", "float", repeated=True),
        Field(7, "encoded", "bool"),
    ]


VarianceNorm = EnumType("VarianceNorm", {"FAN_IN": 0, "FAN_OUT": 1, "AVERAGE": 2})


class FillerParameter(Message):
    FIELDS = [
        Field(1, "type", "string", default="constant"),
        Field(2, "value", "float"),
        Field(3, "min", "float"),
        Field(4, "max", "float", default=1.0),
        Field(5, "mean", "float"),
        Field(6, "std", "float", default=1.0),
        Field(7, "sparse", "int32", default=-1),
        Field(8, "variance_norm", "enum", enum_type=VarianceNorm),
    ]


class NetState(Message):
    FIELDS = [
        Field(1, "phase", "enum", enum_type=Phase, default=Phase.TEST),
        Field(2, "level", "int32"),
        Field(3, "stage", "string", repeated=True),
    ]


class NetStateRule(Message):
    FIELDS = [
        Field(1, "phase", "enum", enum_type=Phase),
        Field(2, "min_level", "int32"),
        Field(3, "max_level", "int32"),
        Field(4, "stage", "string", repeated=True),
        Field(5, "not_stage", "string", repeated=True),
    ]


DimCheckMode = EnumType("DimCheckMode", {"STRICT": 0, "PERMISSIVE": 1})


class ParamSpec(Message):
    FIELDS = [
        Field(1, "name", "string"),
        Field(2, "share_mode", "enum", enum_type=DimCheckMode),
        Field(3, "lr_mult", "float", default=1.0),
        Field(4, "decay_mult", "float", default=1.0),
    ]


class TransformationParameter(Message):
    FIELDS = [
        Field(1, "scale", "float", default=1.0),
        Field(2, "mirror", "bool"),
        Field(3, "crop_size", "uint32"),
        Field(4, "mean_file", "string"),
        Field(5, "mean_value", "float", repeated=True),
        Field(6, "force_color", "bool"),
        Field(7, "force_gray", "bool"),
    ]


class LossParameter(Message):
    Normalization = EnumType("Normalization",
                             {"FULL": 0, "VALID": 1, "BATCH_SIZE": 2, "NONE": 3})
    FIELDS = [
        Field(1, "ignore_label", "int32"),
        Field(3, "normalization", "enum", enum_type=Normalization, default=1),
        Field(2, "normalize", "bool"),
    ]


class AccuracyParameter(Message):
    FIELDS = [
        Field(1, "top_k", "uint32", default=1),
        Field(2, "axis", "int32", default=1),
        Field(3, "ignore_label", "int32"),
    ]


class ConcatParameter(Message):
    FIELDS = [
        Field(2, "axis", "int32", default=1),
        Field(1, "concat_dim", "uint32", default=1),
    ]


class ConvolutionParameter(Message):
    Engine = EnumType("Engine", {"DEFAULT": 0, "CAFFE": 1, "CUDNN": 2})
    FIELDS = [
        Field(1, "num_output", "uint32"),
        Field(2, "bias_term", "bool", default=True),
        Field(3, "pad", "uint32", repeated=True),
        Field(4, "kernel_size", "uint32", repeated=True),
        Field(6, "stride", "uint32", repeated=True),
        Field(18, "dilation", "uint32", repeated=True),
        Field(9, "pad_h", "uint32"),
        Field(10, "pad_w", "uint32"),
        Field(11, "kernel_h", "uint32"),
        Field(12, "kernel_w", "uint32"),
        Field(13, "stride_h", "uint32"),
        Field(14, "stride_w", "uint32"),
        Field(5, "group", "uint32", default=1),
        Field(7, "weight_filler", "message", msg_type=FillerParameter),
        Field(8, "bias_filler", "message", msg_type=FillerParameter),
        Field(15, "engine", "enum", enum_type=Engine),
        Field(16, "axis", "int32", default=1),
        Field(17, "force_nd_im2col", "bool"),
    ]


class DataParameter(Message):
    DB = EnumType("DB", {"LEVELDB": 0, "LMDB": 1})
    FIELDS = [
        Field(1, "source", "string"),
        Field(4, "batch_size", "uint32"),
        Field(7, "rand_skip", "uint32"),
        Field(8, "backend", "enum", enum_type=DB),
        Field(2, "scale", "float", default=1.0),
        Field(3, "mean_file", "string"),
        Field(5, "crop_size", "uint32"),
        Field(6, "mirror", "bool"),
        Field(9, "force_encoded_color", "bool"),
        Field(10, "prefetch", "uint32", default=4),
    ]


class DropoutParameter(Message):
    FIELDS = [Field(1, "dropout_ratio", "float", default=0.5)]


class DummyDataParameter(Message):
    FIELDS = [
        Field(1, "data_filler", "message", msg_type=FillerParameter, repeated=True),
        Field(6, "shape", "message", msg_type=BlobShape, repeated=True),
        Field(2, "num", "uint32", repeated=True),
        Field(3, "channels", "uint32", repeated=True),
        Field(4, "height", "uint32", repeated=True),
        Field(5, "width", "uint32", repeated=True),
    ]


class EltwiseParameter(Message):
    EltwiseOp = EnumType("EltwiseOp", {"PROD": 0, "SUM": 1, "MAX": 2})
    FIELDS = [
        Field(1, "operation", "enum", enum_type=EltwiseOp, default=1),
        Field(2, "coeff", "float", repeated=True),
        Field(3, "stable_prod_grad", "bool", default=True),
    ]


class EmbedParameter(Message):
    FIELDS = [
        Field(1, "num_output", "uint32"),
        Field(2, "input_dim", "uint32"),
        Field(3, "bias_term", "bool", default=True),
        Field(4, "weight_filler", "message", msg_type=FillerParameter),
        Field(5, "bias_filler", "message", msg_type=FillerParameter),
    ]


class ExpParameter(Message):
    FIELDS = [
        Field(1, "base", "float", default=-1.0),
        Field(2, "scale", "float", default=1.0),
        Field(3, "shift", "float"),
    ]


class FlattenParameter(Message):
    FIELDS = [
        Field(1, "axis", "int32", default=1),
        Field(2, "end_axis", "int32", default=-1),
    ]


class HDF5DataParameter(Message):
    FIELDS = [
        Field(1, "source", "string"),
        Field(2, "batch_size", "uint32"),
        Field(3, "shuffle", "bool"),
    ]


class HDF5OutputParameter(Message):
    FIELDS = [Field(1, "file_name", "string")]


class InnerProductParameter(Message):
    FIELDS = [
        Field(1, "num_output", "uint32"),
        Field(2, "bias_term", "bool", default=True),
        Field(3, "weight_filler", "message", msg_type=FillerParameter),
        Field(4, "bias_filler", "message", msg_type=FillerParameter),
        Field(5, "axis", "int32", default=1),
        Field(6, "transpose", "bool"),
    ]


class InputParameter(Message):
    FIELDS = [Field(1, "shape", "message", msg_type=BlobShape, repeated=True)]


class LRNParameter(Message):
    NormRegion = EnumType("NormRegion", {"ACROSS_CHANNELS": 0, "WITHIN_CHANNEL": 1})
    FIELDS = [
        Field(1, "local_size", "uint32", default=5),
        Field(2, "alpha", "float", default=1.0),
        Field(3, "beta", "float", default=0.75),
        Field(4, "norm_region", "enum", enum_type=NormRegion),
        Field(5, "k", "float", default=1.0),
    ]


class MemoryDataParameter(Message):
    FIELDS = [
        Field(1, "batch_size", "uint32"),
        Field(2, "channels", "uint32"),
        Field(3, "height", "uint32"),
        Field(4, "width", "uint32"),
        # CaffeOnSpark extensions (reference reads these via its yahoo/caffe
        # fork; numbers are ours — see module docstring)
        Field(100, "source", "string"),
        Field(101, "share_in_parallel", "bool"),
        Field(102, "dataframe_format", "string"),
        Field(103, "dataframe_column_select", "string", repeated=True),
        Field(104, "image_encoded", "bool"),
    ]


class PoolingParameter(Message):
    PoolMethod = EnumType("PoolMethod", {"MAX": 0, "AVE": 1, "STOCHASTIC": 2})
    FIELDS = [
        Field(1, "pool", "enum", enum_type=PoolMethod),
        Field(4, "pad", "uint32"),
        Field(9, "pad_h", "uint32"),
        Field(10, "pad_w", "uint32"),
        Field(2, "kernel_size", "uint32"),
        Field(5, "kernel_h", "uint32"),
        Field(6, "kernel_w", "uint32"),
        Field(3, "stride", "uint32", default=1),
        Field(7, "stride_h", "uint32"),
        Field(8, "stride_w", "uint32"),
        Field(12, "global_pooling", "bool"),
    ]


class PowerParameter(Message):
    FIELDS = [
        Field(1, "power", "float", default=1.0),
        Field(2, "scale", "float", default=1.0),
        Field(3, "shift", "float"),
    ]


class ReLUParameter(Message):
    FIELDS = [Field(1, "negative_slope", "float")]


class ReshapeParameter(Message):
    FIELDS = [
        Field(1, "shape", "message", msg_type=BlobShape),
        Field(2, "axis", "int32"),
        Field(3, "num_axes", "int32", default=-1),
    ]


class ScaleParameter(Message):
    FIELDS = [
        Field(1, "axis", "int32", default=1),
        Field(2, "num_axes", "int32", default=1),
        Field(3, "filler", "message", msg_type=FillerParameter),
        Field(4, "bias_term", "bool"),
        Field(5, "bias_filler", "message", msg_type=FillerParameter),
    ]


class BiasParameter(Message):
    FIELDS = [
        Field(1, "axis", "int32", default=1),
        Field(2, "num_axes", "int32", default=1),
        Field(3, "filler", "message", msg_type=FillerParameter),
    ]


class BatchNormParameter(Message):
    FIELDS = [
        Field(1, "use_global_stats", "bool"),
        Field(2, "moving_average_fraction", "float", default=0.999),
        Field(3, "eps", "float", default=1e-5),
    ]


class SliceParameter(Message):
    FIELDS = [
        Field(3, "axis", "int32", default=1),
        Field(2, "slice_point", "uint32", repeated=True),
        Field(1, "slice_dim", "uint32", default=1),
    ]


class SoftmaxParameter(Message):
    FIELDS = [Field(2, "axis", "int32", default=1)]


class TanHParameter(Message):
    FIELDS = []


class SigmoidParameter(Message):
    FIELDS = []


class ThresholdParameter(Message):
    FIELDS = [Field(1, "threshold", "float")]


class TileParameter(Message):
    FIELDS = [
        Field(1, "axis", "int32", default=1),
        Field(2, "tiles", "int32"),
    ]


class RecurrentParameter(Message):
    FIELDS = [
        Field(1, "num_output", "uint32"),
        Field(2, "weight_filler", "message", msg_type=FillerParameter),
        Field(3, "bias_filler", "message", msg_type=FillerParameter),
        Field(4, "debug_info", "bool"),
        Field(5, "expose_hidden", "bool"),
    ]


class ArgMaxParameter(Message):
    FIELDS = [
        Field(1, "out_max_val", "bool"),
        Field(2, "top_k", "uint32", default=1),
        Field(3, "axis", "int32"),
    ]


class ReductionParameter(Message):
    ReductionOp = EnumType("ReductionOp", {"SUM": 1, "ASUM": 2, "SUMSQ": 3, "MEAN": 4})
    FIELDS = [
        Field(1, "operation", "enum", enum_type=ReductionOp, default=1),
        Field(2, "axis", "int32"),
        Field(3, "coeff", "float", default=1.0),
    ]


class ELUParameter(Message):
    FIELDS = [Field(1, "alpha", "float", default=1.0)]


class PReLUParameter(Message):
    FIELDS = [
        Field(1, "filler", "message", msg_type=FillerParameter),
        Field(2, "channel_shared", "bool"),
    ]


class CropParameter(Message):
    FIELDS = [
        Field(1, "axis", "int32", default=2),
        Field(2, "offset", "uint32", repeated=True),
    ]


class LogParameter(Message):
    FIELDS = [
        Field(1, "base", "float", default=-1.0),
        Field(2, "scale", "float", default=1.0),
        Field(3, "shift", "float"),
    ]


class MVNParameter(Message):
    FIELDS = [
        Field(1, "normalize_variance", "bool", default=True),
        Field(2, "across_channels", "bool"),
        Field(3, "eps", "float", default=1e-9),
    ]


class ParameterParameter(Message):
    FIELDS = [Field(1, "shape", "message", msg_type=BlobShape)]


class PythonParameter(Message):
    FIELDS = [
        Field(1, "module", "string"),
        Field(2, "layer", "string"),
        Field(3, "param_str", "string"),
        Field(4, "share_in_parallel", "bool"),
    ]


# --- CaffeOnSpark CoSData extension (field names from reference
#     data/lrcn_cos.prototxt; numbers are ours) -------------------------------

CoSTopType = EnumType("CoSTopType", {
    "STRING": 0, "INT": 1, "FLOAT": 2, "INT_ARRAY": 3, "FLOAT_ARRAY": 4,
    "RAW_IMAGE": 5, "ENCODED_IMAGE": 6, "ENCODED_IMAGE_WITH_DIM": 7,
})


class CoSTopParameter(Message):
    FIELDS = [
        Field(1, "name", "string"),
        Field(2, "type", "enum", enum_type=CoSTopType),
        Field(3, "channels", "uint32", default=1),
        Field(4, "height", "uint32", default=1),
        Field(5, "width", "uint32", default=1),
        Field(6, "out_channels", "uint32"),
        Field(7, "out_height", "uint32"),
        Field(8, "out_width", "uint32"),
        Field(9, "sample_num_axes", "uint32", default=3),
        Field(10, "transpose", "bool"),
        Field(11, "transform_param", "message", msg_type=TransformationParameter),
    ]


class CoSDataParameter(Message):
    FIELDS = [
        Field(1, "source", "string"),
        Field(2, "batch_size", "uint32"),
        Field(3, "top", "message", msg_type=CoSTopParameter, repeated=True),
    ]


# --------------------------------------------------------------------------- layer

class LayerParameter(Message):
    FIELDS = [
        Field(1, "name", "string"),
        Field(2, "type", "string"),
        Field(3, "bottom", "string", repeated=True),
        Field(4, "top", "string", repeated=True),
        Field(10, "phase", "enum", enum_type=Phase),
        Field(5, "loss_weight", "float", repeated=True),
        Field(6, "param", "message", msg_type=ParamSpec, repeated=True),
        Field(7, "blobs", "message", msg_type=BlobProto, repeated=True),
        Field(11, "propagate_down", "bool", repeated=True),
        Field(8, "include", "message", msg_type=NetStateRule, repeated=True),
        Field(9, "exclude", "message", msg_type=NetStateRule, repeated=True),
        Field(100, "transform_param", "message", msg_type=TransformationParameter),
        Field(101, "loss_param", "message", msg_type=LossParameter),
        Field(102, "accuracy_param", "message", msg_type=AccuracyParameter),
        Field(103, "argmax_param", "message", msg_type=ArgMaxParameter),
        Field(104, "concat_param", "message", msg_type=ConcatParameter),
        Field(106, "convolution_param", "message", msg_type=ConvolutionParameter),
        Field(107, "data_param", "message", msg_type=DataParameter),
        Field(108, "dropout_param", "message", msg_type=DropoutParameter),
        Field(109, "dummy_data_param", "message", msg_type=DummyDataParameter),
        Field(110, "eltwise_param", "message", msg_type=EltwiseParameter),
        Field(111, "exp_param", "message", msg_type=ExpParameter),
        Field(112, "hdf5_data_param", "message", msg_type=HDF5DataParameter),
        Field(113, "hdf5_output_param", "message", msg_type=HDF5OutputParameter),
        Field(117, "inner_product_param", "message", msg_type=InnerProductParameter),
        Field(118, "lrn_param", "message", msg_type=LRNParameter),
        Field(119, "memory_data_param", "message", msg_type=MemoryDataParameter),
        Field(120, "mvn_param", "message", msg_type=MVNParameter),
        Field(121, "pooling_param", "message", msg_type=PoolingParameter),
        Field(122, "power_param", "message", msg_type=PowerParameter),
        Field(123, "relu_param", "message", msg_type=ReLUParameter),
        Field(124, "sigmoid_param", "message", msg_type=SigmoidParameter),
        Field(125, "softmax_param", "message", msg_type=SoftmaxParameter),
        Field(126, "slice_param", "message", msg_type=SliceParameter),
        Field(127, "tanh_param", "message", msg_type=TanHParameter),
        Field(128, "threshold_param", "message", msg_type=ThresholdParameter),
        Field(130, "python_param", "message", msg_type=PythonParameter),
        Field(131, "prelu_param", "message", msg_type=PReLUParameter),
        Field(133, "reshape_param", "message", msg_type=ReshapeParameter),
        Field(134, "log_param", "message", msg_type=LogParameter),
        Field(135, "flatten_param", "message", msg_type=FlattenParameter),
        Field(136, "reduction_param", "message", msg_type=ReductionParameter),
        Field(137, "embed_param", "message", msg_type=EmbedParameter),
        Field(138, "tile_param", "message", msg_type=TileParameter),
        Field(139, "batch_norm_param", "message", msg_type=BatchNormParameter),
        Field(140, "elu_param", "message", msg_type=ELUParameter),
        Field(141, "bias_param", "message", msg_type=BiasParameter),
        Field(142, "scale_param", "message", msg_type=ScaleParameter),
        Field(143, "input_param", "message", msg_type=InputParameter),
        Field(144, "crop_param", "message", msg_type=CropParameter),
        Field(145, "parameter_param", "message", msg_type=ParameterParameter),
        Field(146, "recurrent_param", "message", msg_type=RecurrentParameter),
        # CaffeOnSpark extensions
        Field(150, "source_class", "string"),
        Field(151, "cos_data_param", "message", msg_type=CoSDataParameter),
    ]


class NetParameter(Message):
    FIELDS = [
        Field(1, "name", "string"),
        Field(3, "input", "string", repeated=True),
        Field(8, "input_shape", "message", msg_type=BlobShape, repeated=True),
        Field(4, "input_dim", "int32", repeated=True),
        Field(5, "force_backward", "bool"),
        Field(6, "state", "message", msg_type=NetState),
        Field(7, "debug_info", "bool"),
        Field(100, "layer", "message", msg_type=LayerParameter, repeated=True),
    ]


# --------------------------------------------------------------------------- solver

SolverMode = EnumType("SolverMode", {"CPU": 0, "GPU": 1})
SnapshotFormat = EnumType("SnapshotFormat", {"HDF5": 0, "BINARYPROTO": 1})
SolverType = EnumType("SolverType", {"SGD": 0, "NESTEROV": 1, "ADAGRAD": 2,
                                     "RMSPROP": 3, "ADADELTA": 4, "ADAM": 5})


class SolverParameter(Message):
    FIELDS = [
        Field(24, "net", "string"),
        Field(25, "net_param", "message", msg_type=NetParameter),
        Field(1, "train_net", "string"),
        Field(2, "test_net", "string", repeated=True),
        Field(21, "train_net_param", "message", msg_type=NetParameter),
        Field(22, "test_net_param", "message", msg_type=NetParameter, repeated=True),
        Field(26, "train_state", "message", msg_type=NetState),
        Field(27, "test_state", "message", msg_type=NetState, repeated=True),
        Field(3, "test_iter", "int32", repeated=True),
        Field(4, "test_interval", "int32"),
        Field(19, "test_compute_loss", "bool"),
        Field(32, "test_initialization", "bool", default=True),
        Field(5, "base_lr", "float"),
        Field(6, "display", "int32"),
        Field(33, "average_loss", "int32", default=1),
        Field(7, "max_iter", "int32"),
        Field(36, "iter_size", "int32", default=1),
        Field(8, "lr_policy", "string"),
        Field(9, "gamma", "float"),
        Field(10, "power", "float"),
        Field(11, "momentum", "float"),
        Field(12, "weight_decay", "float"),
        Field(29, "regularization_type", "string", default="L2"),
        Field(13, "stepsize", "int32"),
        Field(34, "stepvalue", "int32", repeated=True),
        Field(35, "clip_gradients", "float", default=-1.0),
        Field(14, "snapshot", "int32"),
        Field(15, "snapshot_prefix", "string"),
        Field(16, "snapshot_diff", "bool"),
        Field(37, "snapshot_format", "enum", enum_type=SnapshotFormat, default=1),
        Field(17, "solver_mode", "enum", enum_type=SolverMode, default=1),
        Field(18, "device_id", "int32"),
        Field(20, "random_seed", "int64", default=-1),
        Field(40, "type", "string", default="SGD"),
        Field(31, "delta", "float", default=1e-8),
        Field(39, "momentum2", "float", default=0.999),
        Field(38, "rms_decay", "float", default=0.99),
        Field(23, "debug_info", "bool"),
        Field(28, "snapshot_after_train", "bool", default=True),
        Field(30, "solver_type", "enum", enum_type=SolverType),
        Field(41, "layer_wise_reduce", "bool", default=True),
    ]


class SolverState(Message):
    FIELDS = [
        Field(1, "iter", "int32"),
        Field(2, "learned_net", "string"),
        Field(3, "history", "message", msg_type=BlobProto, repeated=True),
        Field(4, "current_step", "int32"),
    ]
