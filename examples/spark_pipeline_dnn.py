"""The reference's simple_dnn flow on the REAL pyspark API surface
(reference examples/simple_dnn.py:40-66): VectorAssembler -> SparkTorch in a
Spark ML Pipeline, fit on local[2] (2 genuine barrier tasks, world_size=2
torch.distributed), save the fitted pipeline in the StopWordsRemover carrier
format, load + unwrap, evaluate.

Run without a JVM against the vendored double:

    PYTHONPATH=vendor python examples/spark_pipeline_dnn.py

or with real pyspark installed, exactly as-is.
"""

import numpy as np


def main():
    import torch
    import torch.nn as nn
    from pyspark.ml import Pipeline, PipelineModel
    from pyspark.ml.evaluation import MulticlassClassificationEvaluator
    from pyspark.ml.feature import VectorAssembler
    from pyspark.sql import SparkSession

    from sparktorch_amd import PysparkPipelineWrapper, SparkTorch, serialize_torch_obj

    spark = SparkSession.builder.master("local[2]").appName("spark-pipeline-dnn").getOrCreate()

    rng = np.random.default_rng(0)
    rows = [
        tuple([float(i % 2)] + list(rng.normal(i % 2 * 2.0, 1.0, 10))) for i in range(400)
    ]
    cols = ["label"] + ["f%d" % i for i in range(10)]
    df = spark.createDataFrame(rows, cols).repartition(2)

    network = nn.Sequential(nn.Linear(10, 32), nn.ReLU(), nn.Linear(32, 2))
    torch_obj = serialize_torch_obj(network, nn.CrossEntropyLoss(), torch.optim.Adam, lr=1e-3)

    assembler = VectorAssembler(inputCols=cols[1:], outputCol="features")
    spark_model = SparkTorch(
        inputCol="features",
        labelCol="label",
        predictionCol="predictions",
        torchObj=torch_obj,
        iters=30,
        verbose=0,
        miniBatch=64,
        validationPct=0.1,
        earlyStopPatience=10,
    )

    p = Pipeline(stages=[assembler, spark_model]).fit(df)
    p.write().overwrite().save("spark_pipeline_dnn_saved")

    loaded = PysparkPipelineWrapper.unwrap(PipelineModel.load("spark_pipeline_dnn_saved"))
    predictions = loaded.transform(df)

    evaluator = MulticlassClassificationEvaluator(
        labelCol="label", predictionCol="predictions", metricName="accuracy"
    )
    print("Train accuracy = %g" % evaluator.evaluate(predictions))


if __name__ == "__main__":
    main()
