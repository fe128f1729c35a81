"""Declarative schema for the Caffe/Poseidon protobuf message set.

This is a from-scratch, schema-driven protobuf implementation: the field
numbers, labels and defaults below mirror the reference schema
(/root/reference/src/caffe/proto/caffe.proto) so that .caffemodel /
.solverstate files are byte-compatible, but the implementation (wire.py,
message.py, text_format.py) is our own.

Field kinds:
    'int32' 'int64' 'uint32' 'uint64' 'bool'  -> varint
    'float'                                   -> 32-bit fixed
    'double'                                  -> 64-bit fixed
    'string' 'bytes'                          -> length-delimited
    'enum:<EnumName>'                         -> varint (named enum)
    'msg:<MessageName>'                       -> length-delimited submessage
Labels: 'opt' (optional), 'rep' (repeated), 'packed' (repeated packed).
"""

# ---------------------------------------------------------------------------
# Enums (name -> {symbol: value})
# ---------------------------------------------------------------------------

ENUMS = {
    "Phase": {"TRAIN": 0, "TEST": 1},
    "BlobMode": {"GLOBAL": 0, "LOCAL": 1},
    "SolverMode": {"CPU": 0, "GPU": 1},
    "SolverType": {"SGD": 0, "NESTEROV": 1, "ADAGRAD": 2},
    "DimCheckMode": {"STRICT": 0, "PERMISSIVE": 1},
    "Engine": {"DEFAULT": 0, "CAFFE": 1, "CUDNN": 2},
    "DB": {"LEVELDB": 0, "LMDB": 1},
    "EltwiseOp": {"PROD": 0, "SUM": 1, "MAX": 2},
    "HingeNorm": {"L1": 1, "L2": 2},
    "NormRegion": {"ACROSS_CHANNELS": 0, "WITHIN_CHANNEL": 1},
    "PoolMethod": {"MAX": 0, "AVE": 1, "STOCHASTIC": 2},
    # LayerParameter.LayerType (caffe.proto:244-286) -- values must match
    # exactly for .caffemodel compatibility.
    "LayerType": {
        "NONE": 0,
        "ABSVAL": 35,
        "ACCURACY": 1,
        "ARGMAX": 30,
        "BNLL": 2,
        "CONCAT": 3,
        "CONTRASTIVE_LOSS": 37,
        "CONVOLUTION": 4,
        "DATA": 5,
        "DROPOUT": 6,
        "DUMMY_DATA": 32,
        "EUCLIDEAN_LOSS": 7,
        "ELTWISE": 25,
        "FLATTEN": 8,
        "HDF5_DATA": 9,
        "HDF5_OUTPUT": 10,
        "HINGE_LOSS": 28,
        "IM2COL": 11,
        "IMAGE_DATA": 12,
        "INFOGAIN_LOSS": 13,
        "INNER_PRODUCT": 14,
        "LRN": 15,
        "MEMORY_DATA": 29,
        "MULTINOMIAL_LOGISTIC_LOSS": 16,
        "MVN": 34,
        "POOLING": 17,
        "POWER": 26,
        "RELU": 18,
        "SIGMOID": 19,
        "SIGMOID_CROSS_ENTROPY_LOSS": 27,
        "SILENCE": 36,
        "SOFTMAX": 20,
        "SOFTMAX_LOSS": 21,
        "SPLIT": 22,
        "SLICE": 33,
        "TANH": 23,
        "WINDOW_DATA": 24,
        "THRESHOLD": 31,
    },
}

# ---------------------------------------------------------------------------
# Messages: name -> {field_name: (number, kind, label, default)}
# default=None means "unset" (proto2 optional semantics; getters fall back to
# the declared default where one exists).
# ---------------------------------------------------------------------------

MESSAGES = {
    "SVProto": {
        "layer_id": (1, "int32", "opt", None),
        "a": (2, "float", "rep", None),
        "b": (3, "float", "rep", None),
    },
    "BlobProto": {
        "num": (1, "int32", "opt", 0),
        "channels": (2, "int32", "opt", 0),
        "height": (3, "int32", "opt", 0),
        "width": (4, "int32", "opt", 0),
        "data": (5, "float", "packed", None),
        "diff": (6, "float", "packed", None),
        "blob_mode": (7, "enum:BlobMode", "opt", 1),   # LOCAL
        "global_id": (8, "int32", "opt", -1),
    },
    "BlobProtoVector": {
        "blobs": (1, "msg:BlobProto", "rep", None),
    },
    "Datum": {
        "channels": (1, "int32", "opt", None),
        "height": (2, "int32", "opt", None),
        "width": (3, "int32", "opt", None),
        "data": (4, "bytes", "opt", None),
        "label": (5, "int32", "opt", None),
        "float_data": (6, "float", "rep", None),
    },
    "FillerParameter": {
        "type": (1, "string", "opt", "constant"),
        "value": (2, "float", "opt", 0.0),
        "min": (3, "float", "opt", 0.0),
        "max": (4, "float", "opt", 1.0),
        "mean": (5, "float", "opt", 0.0),
        "std": (6, "float", "opt", 1.0),
        "sparse": (7, "int32", "opt", -1),
    },
    "NetParameter": {
        "name": (1, "string", "opt", None),
        "layers": (2, "msg:LayerParameter", "rep", None),
        "input": (3, "string", "rep", None),
        "input_dim": (4, "int32", "rep", None),
        "force_backward": (5, "bool", "opt", False),
        "state": (6, "msg:NetState", "opt", None),
    },
    "SolverParameter": {
        "net": (24, "string", "opt", None),
        "net_param": (25, "msg:NetParameter", "opt", None),
        "train_net": (1, "string", "opt", None),
        "test_net": (2, "string", "rep", None),
        "train_net_param": (21, "msg:NetParameter", "opt", None),
        "test_net_param": (22, "msg:NetParameter", "rep", None),
        "train_state": (26, "msg:NetState", "opt", None),
        "test_state": (27, "msg:NetState", "rep", None),
        "test_iter": (3, "int32", "rep", None),
        "test_interval": (4, "int32", "opt", 0),
        "test_compute_loss": (19, "bool", "opt", False),
        "test_initialization": (32, "bool", "opt", True),
        "base_lr": (5, "float", "opt", None),
        "display": (6, "int32", "opt", None),
        "max_iter": (7, "int32", "opt", None),
        "lr_policy": (8, "string", "opt", None),
        "gamma": (9, "float", "opt", None),
        "power": (10, "float", "opt", None),
        "momentum": (11, "float", "opt", None),
        "weight_decay": (12, "float", "opt", None),
        "regularization_type": (29, "string", "opt", "L2"),
        "stepsize": (13, "int32", "opt", None),
        "snapshot": (14, "int32", "opt", 0),
        "snapshot_prefix": (15, "string", "opt", None),
        "snapshot_diff": (16, "bool", "opt", False),
        "solver_mode": (17, "enum:SolverMode", "opt", 1),
        "device_id": (18, "string", "opt", "0"),
        "random_seed": (20, "int64", "opt", -1),
        "solver_type": (30, "enum:SolverType", "opt", 0),
        "delta": (31, "float", "opt", 1e-8),
        "debug_info": (23, "bool", "opt", False),
        "snapshot_after_train": (28, "bool", "opt", True),
        "layer_blobs_global_idx": (33, "msg:LayerPSTablePair", "rep", None),
    },
    "SolverState": {
        "iter": (1, "int32", "opt", None),
        "learned_net": (2, "string", "opt", None),
        "history": (3, "msg:BlobProto", "rep", None),
    },
    "NetState": {
        "phase": (1, "enum:Phase", "opt", 1),  # TEST
        "level": (2, "int32", "opt", 0),
        "stage": (3, "string", "rep", None),
    },
    "NetStateRule": {
        "phase": (1, "enum:Phase", "opt", None),
        "min_level": (2, "int32", "opt", None),
        "max_level": (3, "int32", "opt", None),
        "stage": (4, "string", "rep", None),
        "not_stage": (5, "string", "rep", None),
    },
    "LayerPSTablePair": {
        "layer_name": (1, "string", "opt", None),
        "table_id": (2, "int32", "rep", None),
    },
    "LayerParameter": {
        "bottom": (2, "string", "rep", None),
        "top": (3, "string", "rep", None),
        "name": (4, "string", "opt", None),
        "include": (32, "msg:NetStateRule", "rep", None),
        "exclude": (33, "msg:NetStateRule", "rep", None),
        "type": (5, "enum:LayerType", "opt", 0),
        "blobs": (6, "msg:BlobProto", "rep", None),
        "param": (1001, "string", "rep", None),
        "blob_share_mode": (1002, "enum:DimCheckMode", "rep", None),
        "blobs_lr": (7, "float", "rep", None),
        "weight_decay": (8, "float", "rep", None),
        "loss_weight": (35, "float", "rep", None),
        "accuracy_param": (27, "msg:AccuracyParameter", "opt", None),
        "argmax_param": (23, "msg:ArgMaxParameter", "opt", None),
        "concat_param": (9, "msg:ConcatParameter", "opt", None),
        "contrastive_loss_param": (40, "msg:ContrastiveLossParameter", "opt", None),
        "convolution_param": (10, "msg:ConvolutionParameter", "opt", None),
        "data_param": (11, "msg:DataParameter", "opt", None),
        "dropout_param": (12, "msg:DropoutParameter", "opt", None),
        "dummy_data_param": (26, "msg:DummyDataParameter", "opt", None),
        "eltwise_param": (24, "msg:EltwiseParameter", "opt", None),
        "hdf5_data_param": (13, "msg:HDF5DataParameter", "opt", None),
        "hdf5_output_param": (14, "msg:HDF5OutputParameter", "opt", None),
        "hinge_loss_param": (29, "msg:HingeLossParameter", "opt", None),
        "image_data_param": (15, "msg:ImageDataParameter", "opt", None),
        "infogain_loss_param": (16, "msg:InfogainLossParameter", "opt", None),
        "inner_product_param": (17, "msg:InnerProductParameter", "opt", None),
        "lrn_param": (18, "msg:LRNParameter", "opt", None),
        "memory_data_param": (22, "msg:MemoryDataParameter", "opt", None),
        "mvn_param": (34, "msg:MVNParameter", "opt", None),
        "pooling_param": (19, "msg:PoolingParameter", "opt", None),
        "power_param": (21, "msg:PowerParameter", "opt", None),
        "relu_param": (30, "msg:ReLUParameter", "opt", None),
        "sigmoid_param": (38, "msg:SigmoidParameter", "opt", None),
        "softmax_param": (39, "msg:SoftmaxParameter", "opt", None),
        "slice_param": (31, "msg:SliceParameter", "opt", None),
        "tanh_param": (37, "msg:TanHParameter", "opt", None),
        "threshold_param": (25, "msg:ThresholdParameter", "opt", None),
        "window_data_param": (20, "msg:WindowDataParameter", "opt", None),
        "transform_param": (36, "msg:TransformationParameter", "opt", None),
    },
    "TransformationParameter": {
        "scale": (1, "float", "opt", 1.0),
        "mirror": (2, "bool", "opt", False),
        "crop_size": (3, "uint32", "opt", 0),
        "mean_file": (4, "string", "opt", None),
        "mean_value": (5, "float", "rep", None),
    },
    "AccuracyParameter": {
        "top_k": (1, "uint32", "opt", 1),
    },
    "ArgMaxParameter": {
        "out_max_val": (1, "bool", "opt", False),
        "top_k": (2, "uint32", "opt", 1),
    },
    "ConcatParameter": {
        "concat_dim": (1, "uint32", "opt", 1),
    },
    "ContrastiveLossParameter": {
        "margin": (1, "float", "opt", 1.0),
    },
    "ConvolutionParameter": {
        "num_output": (1, "uint32", "opt", None),
        "bias_term": (2, "bool", "opt", True),
        "pad": (3, "uint32", "opt", 0),
        "pad_h": (9, "uint32", "opt", 0),
        "pad_w": (10, "uint32", "opt", 0),
        "kernel_size": (4, "uint32", "opt", None),
        "kernel_h": (11, "uint32", "opt", None),
        "kernel_w": (12, "uint32", "opt", None),
        "group": (5, "uint32", "opt", 1),
        "stride": (6, "uint32", "opt", 1),
        "stride_h": (13, "uint32", "opt", None),
        "stride_w": (14, "uint32", "opt", None),
        "weight_filler": (7, "msg:FillerParameter", "opt", None),
        "bias_filler": (8, "msg:FillerParameter", "opt", None),
        "engine": (15, "enum:Engine", "opt", 0),
    },
    "DataParameter": {
        "source": (1, "string", "opt", None),
        "batch_size": (4, "uint32", "opt", None),
        "rand_skip": (7, "uint32", "opt", 0),
        "backend": (8, "enum:DB", "opt", 0),
        "shared_file_system": (9, "bool", "opt", False),
        "scale": (2, "float", "opt", 1.0),
        "mean_file": (3, "string", "opt", None),
        "crop_size": (5, "uint32", "opt", 0),
        "mirror": (6, "bool", "opt", False),
    },
    "DropoutParameter": {
        "dropout_ratio": (1, "float", "opt", 0.5),
    },
    "DummyDataParameter": {
        "data_filler": (1, "msg:FillerParameter", "rep", None),
        "num": (2, "uint32", "rep", None),
        "channels": (3, "uint32", "rep", None),
        "height": (4, "uint32", "rep", None),
        "width": (5, "uint32", "rep", None),
    },
    "EltwiseParameter": {
        "operation": (1, "enum:EltwiseOp", "opt", 1),
        "coeff": (2, "float", "rep", None),
        "stable_prod_grad": (3, "bool", "opt", True),
    },
    "ThresholdParameter": {
        "threshold": (1, "float", "opt", 0.0),
    },
    "HDF5DataParameter": {
        "source": (1, "string", "opt", None),
        "batch_size": (2, "uint32", "opt", None),
    },
    "HDF5OutputParameter": {
        "file_name": (1, "string", "opt", None),
    },
    "HingeLossParameter": {
        "norm": (1, "enum:HingeNorm", "opt", 1),
    },
    "ImageDataParameter": {
        "source": (1, "string", "opt", None),
        "batch_size": (4, "uint32", "opt", None),
        "rand_skip": (7, "uint32", "opt", 0),
        "shuffle": (8, "bool", "opt", False),
        "new_height": (9, "uint32", "opt", 0),
        "new_width": (10, "uint32", "opt", 0),
        "shared_file_system": (11, "bool", "opt", False),
        "scale": (2, "float", "opt", 1.0),
        "mean_file": (3, "string", "opt", None),
        "crop_size": (5, "uint32", "opt", 0),
        "mirror": (6, "bool", "opt", False),
    },
    "InfogainLossParameter": {
        "source": (1, "string", "opt", None),
    },
    "InnerProductParameter": {
        "num_output": (1, "uint32", "opt", None),
        "bias_term": (2, "bool", "opt", True),
        "weight_filler": (3, "msg:FillerParameter", "opt", None),
        "bias_filler": (4, "msg:FillerParameter", "opt", None),
    },
    "LRNParameter": {
        "local_size": (1, "uint32", "opt", 5),
        "alpha": (2, "float", "opt", 1.0),
        "beta": (3, "float", "opt", 0.75),
        "norm_region": (4, "enum:NormRegion", "opt", 0),
    },
    "MemoryDataParameter": {
        "batch_size": (1, "uint32", "opt", None),
        "channels": (2, "uint32", "opt", None),
        "height": (3, "uint32", "opt", None),
        "width": (4, "uint32", "opt", None),
    },
    "MVNParameter": {
        "normalize_variance": (1, "bool", "opt", True),
        "across_channels": (2, "bool", "opt", False),
    },
    "PoolingParameter": {
        "pool": (1, "enum:PoolMethod", "opt", 0),
        "pad": (4, "uint32", "opt", 0),
        "pad_h": (9, "uint32", "opt", 0),
        "pad_w": (10, "uint32", "opt", 0),
        "kernel_size": (2, "uint32", "opt", None),
        "kernel_h": (5, "uint32", "opt", None),
        "kernel_w": (6, "uint32", "opt", None),
        "stride": (3, "uint32", "opt", 1),
        "stride_h": (7, "uint32", "opt", None),
        "stride_w": (8, "uint32", "opt", None),
        "engine": (11, "enum:Engine", "opt", 0),
    },
    "PowerParameter": {
        "power": (1, "float", "opt", 1.0),
        "scale": (2, "float", "opt", 1.0),
        "shift": (3, "float", "opt", 0.0),
    },
    "ReLUParameter": {
        "negative_slope": (1, "float", "opt", 0.0),
        "engine": (2, "enum:Engine", "opt", 0),
    },
    "SigmoidParameter": {
        "engine": (1, "enum:Engine", "opt", 0),
    },
    "SliceParameter": {
        "slice_dim": (1, "uint32", "opt", 1),
        "slice_point": (2, "uint32", "rep", None),
    },
    "SoftmaxParameter": {
        "engine": (1, "enum:Engine", "opt", 0),
    },
    "TanHParameter": {
        "engine": (1, "enum:Engine", "opt", 0),
    },
    "WindowDataParameter": {
        "source": (1, "string", "opt", None),
        "scale": (2, "float", "opt", 1.0),
        "mean_file": (3, "string", "opt", None),
        "batch_size": (4, "uint32", "opt", None),
        "crop_size": (5, "uint32", "opt", 0),
        "mirror": (6, "bool", "opt", False),
        "fg_threshold": (7, "float", "opt", 0.5),
        "bg_threshold": (8, "float", "opt", 0.5),
        "fg_fraction": (9, "float", "opt", 0.25),
        "context_pad": (10, "uint32", "opt", 0),
        "crop_mode": (11, "string", "opt", "warp"),
    },
}

# Enum fields whose text-format symbols live in a nested scope; text parsing
# accepts symbols from the referenced enum table regardless of nesting.
