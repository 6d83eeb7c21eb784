# Alias of torchdistx_amd.fake (see package __init__).
from torchdistx_amd.fake import fake_mode, is_fake, meta_like  # noqa: F401
