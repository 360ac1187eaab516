"""SparkAsyncDL / SparkAsyncDLModel — the Spark-ML Estimator/Transformer
API, parameter-for-parameter compatible with reference
sparkflow/tensorflow_async.py (same 20 estimator Params, same defaults
tensorflow_async.py:176-182, same 6 model Params), backed by the
MI355X-native engine instead of TF+Flask.
"""
from __future__ import annotations

import json

import numpy as np

from pyspark import SparkContext, keyword_only
from pyspark.ml import Model
from pyspark.ml.base import Estimator
from pyspark.ml.param import Param, Params, TypeConverters
from pyspark.ml.param.shared import HasInputCol, HasLabelCol, HasPredictionCol
from pyspark.ml.util import Identifiable, MLReadable, MLWritable

from sparkflow_amd.ml_util import convert_weights_to_json, predict_func
from sparkflow_amd.parallel.hogwild import HogwildSparkModel
from sparkflow_amd.parallel.optimizers import REGISTRY
from sparkflow_amd.parallel.ps import OptimizerSpec
from sparkflow_amd.spark.pipeline_util import PysparkReaderWriter


def build_optimizer(optimizer_name, learning_rate, optimizer_options):
    """Name + options -> OptimizerSpec for the PS (parity:
    tensorflow_async.py:17-42 — same 10 registry names, same default
    options, unknown names fall back to gradient_descent)."""
    name = optimizer_name if optimizer_name in REGISTRY else "gradient_descent"
    return OptimizerSpec(name=name, learning_rate=learning_rate,
                         options=optimizer_options)


def handle_data(data, inp_col, label_col):
    if label_col is None:
        return np.asarray(data[inp_col])
    return np.asarray(data[inp_col]), data[label_col]


class SparkAsyncDLModel(Model, HasInputCol, HasPredictionCol,
                        PysparkReaderWriter, MLReadable, MLWritable,
                        Identifiable):
    modelJson = Param(Params._dummy(), "modelJson", "",
                      typeConverter=TypeConverters.toString)
    modelWeights = Param(Params._dummy(), "modelWeights", "",
                         typeConverter=TypeConverters.toString)
    tfOutput = Param(Params._dummy(), "tfOutput", "",
                     typeConverter=TypeConverters.toString)
    tfInput = Param(Params._dummy(), "tfInput", "",
                    typeConverter=TypeConverters.toString)
    tfDropout = Param(Params._dummy(), "tfDropout", "",
                      typeConverter=TypeConverters.toString)
    toKeepDropout = Param(Params._dummy(), "toKeepDropout", "",
                          typeConverter=TypeConverters.toBoolean)

    @keyword_only
    def __init__(self, inputCol=None, modelJson=None, modelWeights=None,
                 tfInput=None, tfOutput=None, tfDropout=None,
                 toKeepDropout=None, predictionCol=None):
        super(SparkAsyncDLModel, self).__init__()
        self._setDefault(modelJson=None, inputCol="encoded",
                         predictionCol="predicted", tfOutput=None,
                         tfInput=None, modelWeights=None, tfDropout=None,
                         toKeepDropout=False)
        kwargs = self._input_kwargs
        self.setParams(**kwargs)

    @keyword_only
    def setParams(self, inputCol=None, modelJson=None, modelWeights=None,
                  tfInput=None, tfOutput=None, tfDropout=None,
                  toKeepDropout=None, predictionCol=None):
        kwargs = self._input_kwargs
        return self._set(**kwargs)

    def _transform(self, dataset):
        inp = self.getOrDefault(self.inputCol)
        out = self.getOrDefault(self.predictionCol)
        mod_json = self.getOrDefault(self.modelJson)
        mod_weights = self.getOrDefault(self.modelWeights)
        tf_input = self.getOrDefault(self.tfInput)
        tf_output = self.getOrDefault(self.tfOutput)
        tf_dropout = self.getOrDefault(self.tfDropout)
        keep = self.getOrDefault(self.toKeepDropout)
        return dataset.rdd.mapPartitions(
            lambda rows: predict_func(rows, mod_json, out, mod_weights, inp,
                                      tf_output, tf_input, tf_dropout, keep)
        ).toDF()


class SparkAsyncDL(Estimator, HasInputCol, HasPredictionCol, HasLabelCol,
                   PysparkReaderWriter, MLReadable, MLWritable, Identifiable):
    tensorflowGraph = Param(Params._dummy(), "tensorflowGraph", "",
                            typeConverter=TypeConverters.toString)
    tfInput = Param(Params._dummy(), "tfInput", "",
                    typeConverter=TypeConverters.toString)
    tfOutput = Param(Params._dummy(), "tfOutput", "",
                     typeConverter=TypeConverters.toString)
    tfLabel = Param(Params._dummy(), "tfLabel", "",
                    typeConverter=TypeConverters.toString)
    tfOptimizer = Param(Params._dummy(), "tfOptimizer", "",
                        typeConverter=TypeConverters.toString)
    tfLearningRate = Param(Params._dummy(), "tfLearningRate", "",
                           typeConverter=TypeConverters.toFloat)
    iters = Param(Params._dummy(), "iters", "",
                  typeConverter=TypeConverters.toInt)
    partitions = Param(Params._dummy(), "partitions", "",
                       typeConverter=TypeConverters.toInt)
    miniBatchSize = Param(Params._dummy(), "miniBatchSize", "",
                          typeConverter=TypeConverters.toInt)
    miniStochasticIters = Param(Params._dummy(), "miniStochasticIters", "",
                                typeConverter=TypeConverters.toInt)
    verbose = Param(Params._dummy(), "verbose", "",
                    typeConverter=TypeConverters.toInt)
    acquireLock = Param(Params._dummy(), "acquireLock", "",
                        typeConverter=TypeConverters.toBoolean)
    shufflePerIter = Param(Params._dummy(), "shufflePerIter", "",
                           typeConverter=TypeConverters.toBoolean)
    tfDropout = Param(Params._dummy(), "tfDropout", "",
                      typeConverter=TypeConverters.toString)
    toKeepDropout = Param(Params._dummy(), "toKeepDropout", "",
                          typeConverter=TypeConverters.toBoolean)
    partitionShuffles = Param(Params._dummy(), "partitionShuffles", "",
                              typeConverter=TypeConverters.toInt)
    optimizerOptions = Param(Params._dummy(), "optimizerOptions", "",
                             typeConverter=TypeConverters.toString)
    port = Param(Params._dummy(), "port", "",
                 typeConverter=TypeConverters.toInt)

    @keyword_only
    def __init__(self, inputCol=None, tensorflowGraph=None, tfInput=None,
                 tfLabel=None, tfOutput=None, tfOptimizer=None,
                 tfLearningRate=None, iters=None, predictionCol=None,
                 partitions=None, miniBatchSize=None,
                 miniStochasticIters=None, acquireLock=None,
                 shufflePerIter=None, tfDropout=None, toKeepDropout=None,
                 verbose=None, labelCol=None, partitionShuffles=None,
                 optimizerOptions=None, port=None):
        """Same Param surface and semantics as the reference estimator
        (tensorflow_async.py:123-184); ``tensorflowGraph`` carries the
        ``build_graph`` JSON artifact of a PyTorch module."""
        super(SparkAsyncDL, self).__init__()
        self._setDefault(inputCol="transformed", tensorflowGraph="",
                         tfInput="x:0", tfLabel=None,
                         tfOutput="out/Sigmoid:0", tfOptimizer="adam",
                         tfLearningRate=.01, partitions=5, miniBatchSize=128,
                         miniStochasticIters=-1, shufflePerIter=True,
                         tfDropout=None, acquireLock=False, verbose=0,
                         iters=1000, toKeepDropout=False,
                         predictionCol="predicted", labelCol=None,
                         partitionShuffles=1, optimizerOptions=None,
                         port=5000)
        kwargs = self._input_kwargs
        self.setParams(**kwargs)

    @keyword_only
    def setParams(self, inputCol=None, tensorflowGraph=None, tfInput=None,
                  tfLabel=None, tfOutput=None, tfOptimizer=None,
                  tfLearningRate=None, iters=None, predictionCol=None,
                  partitions=None, miniBatchSize=None,
                  miniStochasticIters=None, acquireLock=None,
                  shufflePerIter=None, tfDropout=None, toKeepDropout=None,
                  verbose=None, labelCol=None, partitionShuffles=None,
                  optimizerOptions=None, port=None):
        kwargs = self._input_kwargs
        return self._set(**kwargs)

    def getTensorflowGraph(self):
        return self.getOrDefault(self.tensorflowGraph)

    def getIters(self):
        return self.getOrDefault(self.iters)

    def getTfInput(self):
        return self.getOrDefault(self.tfInput)

    def getTfLabel(self):
        return self.getOrDefault(self.tfLabel)

    def getTfOutput(self):
        return self.getOrDefault(self.tfOutput)

    def getTfOptimizer(self):
        return self.getOrDefault(self.tfOptimizer)

    def getTfLearningRate(self):
        return self.getOrDefault(self.tfLearningRate)

    def getPartitions(self):
        return self.getOrDefault(self.partitions)

    def getMiniBatchSize(self):
        return self.getOrDefault(self.miniBatchSize)

    def getMiniStochasticIters(self):
        return self.getOrDefault(self.miniStochasticIters)

    def getVerbose(self):
        return self.getOrDefault(self.verbose)

    def getAcquireLock(self):
        return self.getOrDefault(self.acquireLock)

    # reference-compatible misspelled getter (tensorflow_async.py:238)
    getAqcuireLock = getAcquireLock

    def getShufflePerIter(self):
        return self.getOrDefault(self.shufflePerIter)

    def getTfDropout(self):
        return self.getOrDefault(self.tfDropout)

    def getToKeepDropout(self):
        return self.getOrDefault(self.toKeepDropout)

    def getPartitionShuffles(self):
        return self.getOrDefault(self.partitionShuffles)

    def getOptimizerOptions(self):
        return self.getOrDefault(self.optimizerOptions)

    def getPort(self):
        return self.getOrDefault(self.port)

    def _fit(self, dataset):
        inp_col = self.getInputCol()
        graph_json = self.getTensorflowGraph()
        iters = self.getIters()
        label = self.getLabelCol()
        prediction = self.getPredictionCol()
        tf_input = self.getTfInput()
        tf_label = self.getTfLabel()
        tf_output = self.getTfOutput()
        optimizer_options = self.getOptimizerOptions()
        if optimizer_options is not None:
            optimizer_options = json.loads(optimizer_options)
        optimizer = build_optimizer(self.getTfOptimizer(),
                                    self.getTfLearningRate(),
                                    optimizer_options)
        port = self.getPort()

        df = dataset.rdd.map(lambda x: handle_data(x, inp_col, label))
        if self.getPartitions() < df.getNumPartitions():
            df = df.coalesce(self.getPartitions())

        driver_host = SparkContext._active_spark_context.getConf() \
            .get("spark.driver.host")
        spark_model = HogwildSparkModel(
            tensorflowGraph=graph_json,
            iters=iters,
            tfInput=tf_input,
            tfLabel=tf_label,
            optimizer=optimizer,
            master_url=str(driver_host) + ":" + str(port),
            acquire_lock=self.getAcquireLock(),
            mini_batch=self.getMiniBatchSize(),
            mini_stochastic_iters=self.getMiniStochasticIters(),
            shuffle=self.getShufflePerIter(),
            verbose=self.getVerbose(),
            partition_shuffles=self.getPartitionShuffles(),
            port=port,
        )

        weights = spark_model.train(df)
        json_weights = convert_weights_to_json(weights)

        return SparkAsyncDLModel(
            inputCol=inp_col,
            modelJson=graph_json,
            modelWeights=json_weights,
            tfOutput=tf_output,
            tfInput=tf_input,
            tfDropout=self.getTfDropout(),
            toKeepDropout=self.getToKeepDropout(),
            predictionCol=prediction,
        )
