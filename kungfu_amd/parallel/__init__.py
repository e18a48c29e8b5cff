from kungfu_amd.parallel.fusion import GradBucketReducer, FlatParamGroup  # noqa
