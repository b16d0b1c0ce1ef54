from torched_impala_amd.algo import vtrace, losses  # noqa: F401
