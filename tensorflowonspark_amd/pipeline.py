"""ML-pipeline layer: ``TFEstimator``/``TFModel`` (parity: reference
``pipeline.py``).

``TFEstimator(train_fn, tf_args).fit(df)`` launches a TFCluster over the
DataFrame's context, feeds the selected input columns (sorted by name,
reference ``pipeline.py:415-416``), shuts down with ``grace_secs`` so the chief
can export, and returns a ``TFModel``. ``TFModel.transform(df)`` runs
independent per-partition batch inference with a per-python-worker model cache
(reference ``pipeline.py:495-499``), loading the TorchScript export produced by
``TFNode.export_saved_model``.

Works with real Spark ML (the param mixins subclass pyspark Params when
pyspark is importable) or standalone with the built-in lightweight Params."""

import argparse
import copy
import logging
import os

from . import TFCluster

logger = logging.getLogger(__name__)


class Namespace(object):
    """Convert dicts / argv lists / argparse.Namespace into an attr object
    (parity: reference ``pipeline.py:299-339``)."""

    ARGV = "ARGV"

    def __init__(self, d=None):
        if d is None:
            return
        if isinstance(d, list):
            self.__dict__[Namespace.ARGV] = d
        elif isinstance(d, dict):
            self.__dict__.update(d)
        elif isinstance(d, Namespace):
            self.__dict__.update(d.__dict__)
        elif isinstance(d, argparse.Namespace):
            self.__dict__.update(vars(d))
        else:
            raise TypeError("unsupported Namespace init: {}".format(type(d)))

    def __getattr__(self, name):
        if name.startswith("__") and name.endswith("__"):
            raise AttributeError(name)  # keep copy/pickle protocols working
        return None  # undefined args read as None

    def __repr__(self):
        return "Namespace({})".format(self.__dict__)


# ---------------------------------------------------------------------------
# Params layer: real pyspark.ml.param.Params when pyspark is importable
# (so TFEstimator/TFModel compose inside a pyspark.ml.Pipeline — VERDICT r01
# item 5 / reference pipeline.py:52-296), lightweight stand-in otherwise.
# ---------------------------------------------------------------------------

try:  # pragma: no cover - exercised only under a real pyspark install
    from pyspark.ml.param import Param as _SparkParam
    from pyspark.ml.param import Params as _SparkParams
    _HAVE_PYSPARK_ML = True
except Exception:
    _HAVE_PYSPARK_ML = False

if _HAVE_PYSPARK_ML:  # pragma: no cover - needs pyspark
    class _Params(_SparkParams):
        """pyspark Params with the snake_case kwarg _set/_get the local
        implementation provides (pyspark's _set already accepts param-name
        keywords, so only _get needs a shim)."""

        def _get(self, name, default=None):
            if self.hasParam(name):
                p = self.getParam(name)
                if self.isSet(p) or self.hasDefault(p):
                    return self.getOrDefault(p)
            return default

    def _mixin(name, param, default=None):
        """Has<Name> mixin carrying a real pyspark Param named ``param``
        (snake_case, matching the tf_args attribute it overlays)."""
        cap = name[0].upper() + name[1:]

        def setter(self, value):
            return self._set(**{param: value})

        def getter(self):
            return self._get(param, default)

        attrs = {
            param: _SparkParam(_SparkParams._dummy(), param,
                               "TFoS pipeline param " + param),
            "set" + cap: setter,
            "get" + cap: getter,
        }
        return type("Has" + cap, (_SparkParams,), attrs)
else:
    class _Params(object):
        def __init__(self):
            self._paramMap = {}

        def _set(self, **kwargs):
            self._paramMap.update(kwargs)
            return self

        def _get(self, name, default=None):
            return self._paramMap.get(name, default)

    def _mixin(name, param, default=None):
        """Generate a Has<Name> mixin with set<Name>/get<Name>."""
        cap = name[0].upper() + name[1:]

        def setter(self, value):
            return self._set(**{param: value})

        def getter(self):
            return self._get(param, default)

        return type("Has" + cap, (object,),
                    {"set" + cap: setter, "get" + cap: getter})


HasBatchSize = _mixin("batchSize", "batch_size", 100)
HasClusterSize = _mixin("clusterSize", "cluster_size", 1)
HasEpochs = _mixin("epochs", "epochs", 1)
HasGraceSecs = _mixin("graceSecs", "grace_secs", 30)
HasInputMapping = _mixin("inputMapping", "input_mapping")
HasInputMode = _mixin("inputMode", "input_mode", TFCluster.InputMode.SPARK)
HasMasterNode = _mixin("masterNode", "master_node", "chief")
HasModelDir = _mixin("modelDir", "model_dir")
HasNumPS = _mixin("numPS", "num_ps", 0)
HasDriverPSNodes = _mixin("driverPSNodes", "driver_ps_nodes", False)
HasOutputMapping = _mixin("outputMapping", "output_mapping")
HasProtocol = _mixin("protocol", "protocol", "xgmi")
HasReaders = _mixin("readers", "readers", 1)
HasSteps = _mixin("steps", "steps", 1000)
HasTensorboard = _mixin("tensorboard", "tensorboard", False)
HasTFRecordDir = _mixin("tfrecordDir", "tfrecord_dir")
HasExportDir = _mixin("exportDir", "export_dir")
HasSignatureDefKey = _mixin("signatureDefKey", "signature_def_key")
HasTagSet = _mixin("tagSet", "tag_set")
HasNumGPUs = _mixin("numGPUs", "num_gpus", 1)
HasInputShapes = _mixin("inputShapes", "input_shapes")


class TFParams(_Params, HasBatchSize, HasClusterSize, HasEpochs, HasGraceSecs,
               HasInputMapping, HasInputMode, HasMasterNode, HasModelDir,
               HasNumPS, HasDriverPSNodes, HasOutputMapping, HasProtocol,
               HasReaders, HasSteps, HasTensorboard, HasTFRecordDir,
               HasExportDir, HasSignatureDefKey, HasTagSet, HasNumGPUs,
               HasInputShapes):
    """Merged args+params holder (parity: reference ``pipeline.py:342-351``)."""

    def __init__(self, tf_args=None):
        _Params.__init__(self)
        self.args = Namespace(tf_args) if tf_args is not None else Namespace({})

    def merge_args_params(self):
        """Overlay ML params onto a copy of args."""
        args = copy.copy(self.args)
        for k, v in self._paramMap.items():
            # pyspark stores {Param: value}; the local layer {name: value}
            setattr(args, k.name if hasattr(k, "name") else k, v)
        return args


if _HAVE_PYSPARK_ML:  # pragma: no cover - pyspark Pipeline integration
    from pyspark.ml import Estimator as _PipelineEstimator
    from pyspark.ml import Model as _PipelineModel
else:
    class _PipelineEstimator(object):
        pass

    class _PipelineModel(object):
        pass


class TFEstimator(TFParams, _PipelineEstimator):
    """Spark-ML-style Estimator around a distributed train function;
    a real ``pyspark.ml.Estimator`` (usable inside ``pyspark.ml.Pipeline``)
    when pyspark is installed."""

    def __init__(self, train_fn, tf_args=None, export_fn=None):
        super().__init__(tf_args)
        self.train_fn = train_fn
        self.export_fn = export_fn
        self._set(batch_size=100, epochs=1, steps=1000, grace_secs=30,
                  cluster_size=1, num_ps=0, master_node="chief",
                  input_mode=TFCluster.InputMode.SPARK)

    def fit(self, dataset):
        return self._fit(dataset)

    def _fit(self, dataset):
        args = self.merge_args_params()
        logger.info("fit: args = %s", args)
        assert args.input_mode == TFCluster.InputMode.SPARK, \
            "TFEstimator requires InputMode.SPARK (reference pipeline.py:124-127)"
        sc = dataset.sc if hasattr(dataset, "sc") else dataset.rdd.context
        input_cols = sorted(args.input_mapping) if args.input_mapping \
            else list(dataset.columns)
        cluster = TFCluster.run(sc, self.train_fn, args, args.cluster_size,
                                num_ps=args.num_ps,
                                tensorboard=bool(args.tensorboard),
                                input_mode=args.input_mode,
                                master_node=args.master_node,
                                num_gpus=args.num_gpus or 0)
        cluster.train(dataset.select(input_cols).rdd, args.epochs)
        cluster.shutdown(grace_secs=args.grace_secs or 30)
        # Early-fail the export->transform contract: if this pipeline's model
        # will be loaded by TFModel.transform (export_dir set), the chief must
        # have produced a TorchScript export — a state_dict-only fallback
        # would otherwise break the pipeline late, inside transform.
        if getattr(args, "export_dir", None):
            script_path = os.path.join(args.export_dir, "model.pt")
            sd_path = os.path.join(args.export_dir, "state_dict.pt")
            if not os.path.exists(script_path) and os.path.exists(sd_path):
                raise RuntimeError(
                    "fit() produced only a state_dict export at {} — "
                    "torch.jit.script failed for this model, and "
                    "TFModel.transform requires TorchScript. Make the model "
                    "scriptable or export with require_script=True to see "
                    "the scripting error at export time.".format(sd_path))
        return TFModel(self.merge_args_params())


class TFModel(TFParams, _PipelineModel):
    """Spark-ML-style Model: independent per-partition batch inference;
    a real ``pyspark.ml.Model`` when pyspark is installed."""

    def __init__(self, tf_args):
        super().__init__(tf_args)
        self._set(batch_size=getattr(tf_args, "batch_size", None) or 100)

    def transform(self, dataset):
        return self._transform(dataset)

    def _transform(self, dataset):
        args = self.merge_args_params()
        logger.info("transform: args = %s", args)
        assert args.export_dir or args.model_dir, \
            "TFModel needs export_dir (TorchScript) or model_dir (state_dict)"
        input_cols = sorted(args.input_mapping) if args.input_mapping \
            else list(dataset.columns)
        output_cols = [c for _t, c in
                       sorted((args.output_mapping or {"output": "prediction"}).items())]
        rdd_out = dataset.select(input_cols).rdd.mapPartitions(
            lambda it: _run_model(it, args))
        if hasattr(dataset, "sc"):  # LocalDataFrame
            rows = rdd_out.collect()
            return dataset.sc.createDataFrame(rows, output_cols)
        # real pyspark  # pragma: no cover
        from pyspark.sql import SparkSession
        spark = SparkSession.builder.getOrCreate()
        return spark.createDataFrame(rdd_out, output_cols)


# per-python-worker cache of the loaded model (reference pipeline.py:495-499)
_model_cache = {"key": None, "model": None}


def _load_model(args):
    import torch
    key = (args.export_dir, args.model_dir)
    if _model_cache["key"] == key and _model_cache["model"] is not None:
        return _model_cache["model"]
    if args.export_dir:
        path = os.path.join(args.export_dir, "model.pt")
        model = torch.jit.load(path, map_location="cpu")
    else:
        raise ValueError("model_dir-only transform requires an export; "
                         "save a TorchScript export via ctx.export_saved_model")
    if torch.cuda.is_available():
        model = model.to("cuda")
    model.eval()
    _model_cache["key"] = key
    _model_cache["model"] = model
    return model


def yield_batch(iterable, batch_size, num_tensors=1):
    """Group rows into batches of lists-per-tensor (parity: reference
    ``pipeline.py:691-713``; bytearray values are converted to bytes)."""
    tensors = [[] for _ in range(num_tensors)]
    for row in iterable:
        if row is None:
            break
        items = row if isinstance(row, (list, tuple)) else [row]
        for i in range(num_tensors):
            v = items[i]
            if isinstance(v, bytearray):
                v = bytes(v)
            tensors[i].append(v)
        if len(tensors[0]) >= batch_size:
            yield tensors
            tensors = [[] for _ in range(num_tensors)]
    if len(tensors[0]) > 0:
        yield tensors


def _run_model(iterator, args):
    import torch
    model = _load_model(args)
    device = next(model.parameters()).device if any(
        True for _ in model.parameters()) else torch.device("cpu")
    input_cols = sorted(args.input_mapping) if args.input_mapping else None
    n_inputs = len(input_cols) if input_cols else 1
    batch_size = args.batch_size or 100
    shapes = args.input_shapes or {}
    for tensors in yield_batch(iterator, batch_size, n_inputs):
        inputs = []
        for i, col_vals in enumerate(tensors):
            t = torch.as_tensor(col_vals, dtype=torch.float32, device=device)
            # coerce flat row arrays to the expected tensor shape (the
            # reference reshaped from the saved_model signature,
            # pipeline.py:618-630; TorchScript has no shape metadata, so the
            # shape comes from the input_shapes param)
            col = input_cols[i] if input_cols else None
            if col in shapes:
                t = t.reshape([t.shape[0]] + list(shapes[col]))
            inputs.append(t)
        with torch.no_grad():
            out = model(*inputs)
        if isinstance(out, (list, tuple)):
            outs = [o.cpu() for o in out]
        else:
            outs = [out.cpu()]
        n = len(tensors[0])
        for o in outs:
            assert o.shape[0] == n, \
                "output batch size {} != input {}".format(o.shape[0], n)
        for j in range(n):
            yield tuple(outs[k][j].tolist() for k in range(len(outs)))
