from torchdistx_amd.utils.tape import describe_module, record_info  # noqa: F401
