from sparktorch_amd.compat.params import HAS_PYSPARK
