from torchdistx_amd.utils.tape import (  # noqa: F401
    describe_module,
    materialization_report,
    record_info,
)
