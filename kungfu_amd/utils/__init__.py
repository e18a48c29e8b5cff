from kungfu_amd.utils.dtypes import core_dtype, core_op, hip_dtype  # noqa
